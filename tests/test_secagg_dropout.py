"""Secure-aggregation dropout story (VERDICT round-1 item 9).

Pairwise-mask secure aggregation cannot recover a round in which a rank
dies mid-collective (its masks no longer cancel); the DEFINED recovery is
checkpoint-restart: the collective fails fast on every surviving rank, the
last checkpoint is intact, and a fresh (smaller) group resumes from it.
This test proves the whole path at 4 gloo ranks with an injected death.
"""
import os

import pytest
import torch
import torch.multiprocessing as mp


def _shard(rank):
    g = torch.Generator().manual_seed(500 + rank)
    x = torch.randn(256, 16, generator=g)
    return {"x": x, "y": (x.sum(-1) > 0).long()}


MK = {"in_features": 16, "num_classes": 2}


def _phase1(rank: int, world: int, port: int, ckpt_dir: str, results):
    """4 ranks, secure agg, checkpoint every round; rank 3 dies in round 2."""
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    import torch.distributed as dist

    from algorithms.fedavg import fit_rounds
    from vantage6_amd.parallel.comm import init_from_env

    comm = init_from_env(force_backend="gloo", timeout_s=20.0)
    ckpt = os.path.join(ckpt_dir, "secagg.safetensors")

    def killer(trainer, stats):
        if rank == 3 and stats["round"] >= 1:
            os._exit(17)  # injected fault: survivors die mid-collective next round

    try:
        fit_rounds(_shard(rank), model="logreg", model_kwargs=MK, rounds=6,
                   local_steps=2, batch_size=32, lr=0.05, secure_agg=True,
                   comm=comm, device=torch.device("cpu"), round_hook=killer,
                   checkpoint=ckpt, checkpoint_every=1)
        results[rank] = {"outcome": "completed"}  # only the dead rank skips this
    except Exception as exc:  # noqa: BLE001 — collective fails fast on death
        results[rank] = {"outcome": "aborted", "error": type(exc).__name__}
    finally:
        try:
            dist.destroy_process_group()
        except Exception:  # noqa: BLE001
            pass


def _phase2(rank: int, world: int, port: int, ckpt_dir: str, results):
    """Surviving 3 ranks restart as a fresh group from the checkpoint."""
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    import torch.distributed as dist

    from algorithms.fedavg import fit_rounds
    from vantage6_amd.parallel.comm import init_from_env

    comm = init_from_env(force_backend="gloo", timeout_s=60.0)
    ckpt = os.path.join(ckpt_dir, "secagg.safetensors")
    try:
        out = fit_rounds(_shard(rank), model="logreg", model_kwargs=MK, rounds=6,
                         local_steps=2, batch_size=32, lr=0.05, secure_agg=True,
                         comm=comm, device=torch.device("cpu"),
                         checkpoint=ckpt, checkpoint_every=2)
        results[rank] = {"outcome": "completed", "rounds": out["rounds"],
                         "checksum": out["master_checksum"],
                         "history_len": len(out["history"])}
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_secagg_rank_death_checkpoint_restart(tmp_path):
    ctx = mp.get_context("spawn")
    mgr = ctx.Manager()
    res1 = mgr.dict()
    procs = [ctx.Process(target=_phase1, args=(r, 4, 29571, str(tmp_path), res1))
             for r in range(4)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(120)
        if p.is_alive():
            p.terminate()
            p.join(10)
    # rank 3 died by design (exit 17); survivors must have aborted cleanly
    assert procs[3].exitcode == 17
    for r in range(3):
        assert res1.get(r, {}).get("outcome") == "aborted", res1.get(r)
    # the last checkpoint is intact and resumable
    ckpt = tmp_path / "secagg.safetensors"
    assert ckpt.exists() and (tmp_path / "secagg.safetensors.json").exists()

    res2 = mgr.dict()
    mp.spawn(_phase2, args=(3, 29572, str(tmp_path), res2), nprocs=3, join=True)
    assert all(res2[r]["outcome"] == "completed" for r in range(3))
    # resumed rounds only (history excludes pre-checkpoint rounds)
    assert all(res2[r]["history_len"] < 6 for r in range(3))
    # every survivor holds the identical aggregated model
    cks = [res2[r]["checksum"] for r in range(3)]
    assert cks[0] == pytest.approx(cks[1], abs=1e-9)
    assert cks[0] == pytest.approx(cks[2], abs=1e-9)
