"""RCCL smoke on ONE MI355X: 2 ranks share cuda:0 so the real collective
path (RCCL over dmabuf IPC, not gloo) is exercised without an 8-GPU node
(VERDICT round-1 item 1). Also proves the fit/agg overlap engine produces
bit-identical results to the synchronous aggregation path under RCCL.
"""
import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu


def _worker(rank: int, world: int, port: int, results):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK="0",
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
        HSA_ENABLE_IPC_MODE_LEGACY="0",
    )
    import torch.distributed as dist

    from algorithms.fedavg import _build, make_batches
    from vantage6_amd.parallel.comm import init_from_env
    from vantage6_amd.parallel.fedavg import FedAvgConfig, FedAvgTrainer

    comm = init_from_env()  # nccl (=RCCL) backend, both ranks on cuda:0
    try:
        assert comm.backend == "nccl"
        dev = comm.device

        # --- collective sanity: all-reduce over RCCL on one device
        t = torch.full((1024,), float(rank + 1), device=dev)
        dist.all_reduce(t)
        assert torch.all(t == float(sum(range(1, world + 1)))), "all_reduce wrong"

        # --- FedAvg: overlapped aggregation must equal synchronous exactly.
        # Linear-only model → deterministic kernels → bitwise comparison.
        g = torch.Generator().manual_seed(1000 + rank)
        x = torch.randn(512, 64, generator=g)
        w = torch.randn(64, 4, generator=g)
        dataset = {"x": x, "y": (x @ w).argmax(-1)}
        mk = {"in_features": 64, "num_classes": 4}

        def run(overlap: bool) -> torch.Tensor:
            net, loss_fn = _build("logreg", mk, dev)
            tr = FedAvgTrainer(
                net, loss_fn, make_batches(dataset, 64, dev, seed=7 + rank),
                comm,
                FedAvgConfig(local_steps=2, lr=0.05, overlap=overlap,
                             bucket_bytes=4096),  # tiny buckets: many events
                n_samples=512,
            )
            for _ in range(4):
                tr.run_round()
            tr.finalize()
            if overlap:
                assert tr.overlap_stats.get("agg_comm_ms", 0) >= 0
            return tr.flat.master_flat.clone()

        m_sync = run(overlap=False)
        m_ovl = run(overlap=True)
        results[rank] = {
            "bitwise_equal": bool(torch.equal(m_sync, m_ovl)),
            "checksum": float(m_ovl.double().sum().item()),
            "max_diff": float((m_sync - m_ovl).abs().max().item()),
        }
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_rccl_two_ranks_one_gpu_overlap_equivalence():
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    port = 29561
    ctx = mp.get_context("spawn")
    mgr = ctx.Manager()
    results = mgr.dict()
    mp.spawn(_worker, args=(2, port, results), nprocs=2, join=True)
    assert set(results.keys()) == {0, 1}
    for r in (0, 1):
        assert results[r]["bitwise_equal"], (
            f"rank {r}: overlap diverged from sync aggregation "
            f"(max diff {results[r]['max_diff']})")
    # both ranks hold the identical aggregated model
    assert results[0]["checksum"] == results[1]["checksum"]
