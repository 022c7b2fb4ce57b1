"""Checkpoint/resume + metrics tests (CPU)."""
import pytest
import torch

from algorithms.fedavg import _build, make_batches
from vantage6_amd.parallel.checkpoint import load_checkpoint, save_checkpoint
from vantage6_amd.parallel.comm import CommContext
from vantage6_amd.parallel.fedavg import FedAvgConfig, FedAvgTrainer
from vantage6_amd.utils.metrics import RoundMetrics


def make_trainer(seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(128, 8, generator=g)
    dataset = {"x": x, "y": (x.sum(-1) > 0).long()}
    net, loss_fn = _build("logreg", {"in_features": 8, "num_classes": 2}, "cpu")
    batches = make_batches(dataset, 32, "cpu", seed=7)
    cfg = FedAvgConfig(local_steps=2, lr=0.05)
    return FedAvgTrainer(net, loss_fn, batches, CommContext(), cfg, n_samples=128)


def test_checkpoint_roundtrip(tmp_path):
    t1 = make_trainer()
    for _ in range(3):
        t1.run_round()
    ckpt = tmp_path / "state.safetensors"
    save_checkpoint(t1, ckpt)

    t2 = make_trainer()
    rnd = load_checkpoint(t2, ckpt)
    assert rnd == 3
    assert torch.equal(t1.flat.master_flat, t2.flat.master_flat)
    assert torch.equal(t1.flat.momentum_flat, t2.flat.momentum_flat)
    assert torch.equal(t1.flat.param_flat, t2.flat.param_flat)
    # resumed trainer continues deterministically (same data iterator state
    # is the caller's responsibility; here fresh iterators on both sides)
    s1 = t1.run_round()
    s2 = t2.run_round()
    assert s1["round"] == s2["round"] == 4


def test_checkpoint_shape_mismatch_rejected(tmp_path):
    t1 = make_trainer()
    save_checkpoint(t1, tmp_path / "c.safetensors")
    g = torch.Generator().manual_seed(0)
    x = torch.randn(64, 4, generator=g)
    net, loss_fn = _build("logreg", {"in_features": 4, "num_classes": 2}, "cpu")
    t3 = FedAvgTrainer(net, loss_fn,
                       make_batches({"x": x, "y": (x.sum(-1) > 0).long()}, 16, "cpu"),
                       CommContext(), FedAvgConfig(), n_samples=64)
    import pytest

    with pytest.raises(ValueError):
        load_checkpoint(t3, tmp_path / "c.safetensors")


def test_metrics_table():
    m = RoundMetrics()
    t = make_trainer()
    for _ in range(3):
        m.record(t.run_round())
    table = m.table()
    assert "rounds/s" in table
    assert len(table.splitlines()) >= 5


def test_prometheus_endpoint_scrape():
    """RoundMetrics with a port exposes rounds/phase metrics over HTTP."""
    import requests

    from vantage6_amd.utils.metrics import RoundMetrics

    m = RoundMetrics(prometheus_port=7791)
    m.record({"loss": 1.5, "fit_s": 0.01, "agg_s": 0.002, "total_s": 0.012})
    m.record({"loss": 1.2, "fit_s": 0.011, "agg_s": 0.002, "total_s": 0.013})
    body = requests.get("http://127.0.0.1:7791/metrics", timeout=5).text
    assert "v6amd_rounds_total" in body or "rounds" in body, body[:400]


def test_fit_rounds_checkpoint_resume(tmp_path):
    """Engine-mode checkpointing: crash after round 4, resume completes the
    remaining rounds and the master matches an uninterrupted 6-round run."""
    import torch

    from algorithms.fedavg import fit_rounds
    from vantage6_amd.parallel.comm import CommContext

    g = torch.Generator().manual_seed(5)
    dataset = {"x": torch.randn(64, 16, generator=g),
               "y": torch.randint(0, 2, (64,), generator=g)}
    kw = dict(model="logreg", model_kwargs={"in_features": 16, "num_classes": 2},
              local_steps=2, batch_size=16, lr=0.1,
              comm=CommContext(device=torch.device("cpu")))
    ckpt = str(tmp_path / "fit.ckpt")
    out4 = fit_rounds(dataset, rounds=4, checkpoint=ckpt, checkpoint_every=1, **kw)
    out6 = fit_rounds(dataset, rounds=6, checkpoint=ckpt, checkpoint_every=1, **kw)
    assert len(out6["history"]) == 2  # resumed at round 4, ran 2 more
    ref = fit_rounds(dataset, rounds=6, **kw)
    assert out6["master_checksum"] == pytest.approx(ref["master_checksum"],
                                                    rel=1e-6)


def test_checkpoint_round_trips_server_optimizer_state(tmp_path):
    """FedAvgM/FedAdam server buffers survive save/load (a resumed run
    must not restart the server recurrence)."""
    import torch

    from algorithms.fedavg import _build, make_batches
    from vantage6_amd.parallel.checkpoint import (load_checkpoint,
                                                  save_checkpoint)
    from vantage6_amd.parallel.comm import CommContext
    from vantage6_amd.parallel.fedavg import FedAvgConfig, FedAvgTrainer

    comm = CommContext()
    x = torch.randn(64, 8, generator=torch.Generator().manual_seed(11))
    ds = {"x": x, "y": (x.sum(-1) > 0).long()}
    mk = {"in_features": 8, "num_classes": 2}

    def trainer():
        net, loss = _build("logreg", mk, "cpu")
        return FedAvgTrainer(net, loss, make_batches(ds, 32, "cpu", seed=2),
                             comm, FedAvgConfig(local_steps=1, lr=0.1,
                                                server_adam=True),
                             n_samples=64)

    tr = trainer()
    tr.run_round()  # allocates srv buffers (world-1: recurrence skipped,
    tr._srv_v.fill_(0.25)  # but the state must still round-trip)
    tr._srv_m.fill_(-0.5)
    p = tmp_path / "ck.safetensors"
    save_checkpoint(tr, p)
    tr2 = trainer()
    load_checkpoint(tr2, p)
    assert torch.equal(tr2._srv_v, tr._srv_v)
    assert torch.equal(tr2._srv_m, tr._srv_m)
