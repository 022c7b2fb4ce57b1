"""Multi-process FedAvg engine tests on CPU (gloo, world_size=2).

The distributed path must be correct by construction without a GPU
(SURVEY.md §5 lesson); the same code runs over RCCL on MI355X.
"""
import os

import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank: int, world: int, port: int, results):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
    )
    import torch.distributed as dist

    from algorithms.fedavg import fit_rounds
    from vantage6_amd.parallel.comm import init_from_env

    comm = init_from_env(force_backend="gloo")
    try:
        g = torch.Generator().manual_seed(1000 + rank)
        x = torch.randn(256, 16, generator=g)
        w = torch.randn(16, 2, generator=g)
        dataset = {"x": x, "y": (x @ w).argmax(-1)}
        mk = {"in_features": 16, "num_classes": 2}

        out_plain = fit_rounds(
            dataset, model="logreg", model_kwargs=mk, rounds=3, local_steps=2,
            batch_size=32, lr=0.05, comm=comm, device=torch.device("cpu"),
        )
        # gather rank-0's view of final loss + param checksum via all_reduce
        from vantage6_amd.models import build_model

        torch.manual_seed(1234)
        net = build_model("logreg", **mk)

        out_sec = fit_rounds(
            dataset, model="logreg", model_kwargs=mk, rounds=3, local_steps=2,
            batch_size=32, lr=0.05, secure_agg=True, comm=comm,
            device=torch.device("cpu"),
        )
        # reduce+broadcast aggregation must match all-reduce exactly-ish
        from vantage6_amd.parallel.fedavg import FedAvgConfig, FedAvgTrainer
        from algorithms.fedavg import _build, make_batches

        def run_agg(algo):
            net2, loss2 = _build("logreg", mk, "cpu")
            tr = FedAvgTrainer(net2, loss2, make_batches(dataset, 32, "cpu", seed=3),
                               comm, FedAvgConfig(local_steps=1, lr=0.05,
                                                  agg_algo=algo), n_samples=256)
            for _ in range(2):
                tr.run_round()
            return float(tr.flat.master_flat.double().sum())

        ck_ar = run_agg("allreduce")
        ck_rb = run_agg("reduce_bcast")
        results[rank] = {
            "plain_loss": out_plain["final_loss"],
            "sec_loss": out_sec["final_loss"],
            "plain_ck": out_plain["master_checksum"],
            "sec_ck": out_sec["master_checksum"],
            "agg_ar": ck_ar, "agg_rb": ck_rb,
        }
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_fedavg_gloo_two_ranks_plain_and_secure():
    port = 29531
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker, args=(2, port, results), nprocs=2, join=True)
    assert set(results.keys()) == {0, 1}
    # secure-agg must match plain aggregation to quantization precision
    # (same shard, same RNG → same trajectory up to int64 fixed-point rounding)
    for r in (0, 1):
        assert results[r]["plain_loss"] == pytest.approx(
            results[r]["sec_loss"], abs=5e-3
        )
    # after aggregation every rank must hold the identical global model
    assert results[0]["plain_ck"] == pytest.approx(results[1]["plain_ck"], abs=1e-6)
    assert results[0]["sec_ck"] == pytest.approx(results[1]["sec_ck"], abs=1e-6)
    # reduce_bcast == allreduce (same math, different collective schedule)
    for r in (0, 1):
        assert results[r]["agg_ar"] == pytest.approx(results[r]["agg_rb"], abs=1e-4)


def test_fedavg_single_process_runs():
    """world_size=1 degenerates to plain local training."""
    from algorithms.fedavg import fit_rounds
    from vantage6_amd.parallel.comm import CommContext

    g = torch.Generator().manual_seed(5)
    x = torch.randn(128, 8, generator=g)
    dataset = {"x": x, "y": (x.sum(-1) > 0).long()}
    out = fit_rounds(
        dataset, model="logreg", model_kwargs={"in_features": 8, "num_classes": 2},
        rounds=2, local_steps=2, batch_size=32, comm=CommContext(),
        device=torch.device("cpu"),
    )
    assert out["rounds"] == 2
    assert out["final_loss"] > 0


def test_elastic_round_survives_station_failure(tmp_path):
    """API-mode FedAvg with fault injection: one of 3 stations crashes every
    round; the weighted mean proceeds over the 2 survivors (SURVEY.md §6.3
    elastic recovery); run statuses record the crash."""
    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    dbs = [[DatabaseConfig(label="default",
                           uri=f"synthetic://tabular?n=128&d=8&seed={i}",
                           type="synthetic")] for i in range(3)]
    with DevNetwork(n_stations=3, databases=dbs, name="elastic") as net:
        fail_org = net.org_ids[2]
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="elastic fedavg", image="algorithms.fedavg",
            input_={"method": "central_fedavg", "kwargs": {
                "model": "logreg",
                "model_kwargs": {"in_features": 8, "num_classes": 2},
                "rounds": 2, "local_steps": 1, "batch_size": 16,
                "min_stations": 2, "fail_orgs": [fail_org],
            }},
        )
        out = net.client.wait_for_results(task["id"], timeout=120)[0]
        assert len(out["history"]) == 2
        # the injected-fault station shows crashed runs in the task tree
        tasks = net.client.task.list(per_page=100)["data"]
        sub = [t for t in tasks if t["parent_id"] == task["id"]]
        assert sub, "no subtasks recorded"
        runs = net.client.run.from_task(sub[0]["id"])
        statuses = {r["organization_id"]: r["status"] for r in runs}
        assert statuses[fail_org] == "crashed"
        assert sum(1 for s in statuses.values() if s == "completed") == 2


@pytest.mark.timeout(240)
def test_fedavg_gloo_four_ranks_secure_equals_plain():
    """World size 4: the pairwise Philox masks (6 pairs) must still cancel
    exactly and match the plain weighted mean — guards the pair indexing
    beyond the trivial 2-rank case (the driver scales to 8)."""
    port = 29621
    results = mp.Manager().dict()
    mp.spawn(_worker, args=(4, port, results), nprocs=4, join=True)
    assert len(results) == 4
    cks = {r["plain_ck"] for r in results.values()}
    assert len(cks) == 1, "plain master diverged across ranks"
    sks = {r["sec_ck"] for r in results.values()}
    assert len(sks) == 1, "secure master diverged across ranks"
    for r in results.values():
        assert r["sec_ck"] == pytest.approx(r["plain_ck"], rel=1e-5)
        assert r["agg_ar"] == pytest.approx(r["agg_rb"], rel=1e-6)


def _prox_momentum_worker(rank: int, world: int, port: int, results):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
    )
    import torch.distributed as dist

    from algorithms.fedavg import _build, make_batches
    from vantage6_amd.parallel.comm import init_from_env
    from vantage6_amd.parallel.fedavg import FedAvgConfig, FedAvgTrainer

    comm = init_from_env(force_backend="gloo")
    try:
        g = torch.Generator().manual_seed(500 + rank)
        x = torch.randn(128, 8, generator=g)
        dataset = {"x": x, "y": (x.sum(-1) > 0).long()}
        mk = {"in_features": 8, "num_classes": 2}

        def run(mu=0.0, beta=0.0, slr=1.0, rounds=3):
            net, loss = _build("logreg", mk, "cpu")
            tr = FedAvgTrainer(
                net, loss, make_batches(dataset, 32, "cpu", seed=7), comm,
                FedAvgConfig(local_steps=2, lr=0.1, momentum=0.0,
                             weight_decay=0.0, fedprox_mu=mu,
                             server_momentum=beta, server_lr=slr),
                n_samples=128)
            snaps = []
            for _ in range(rounds):
                tr.run_round()
                snaps.append(tr.flat.master_flat.clone())
            return snaps

        base = run()
        prox = run(mu=10.0)  # strong prox pulls harder toward round start
        # manual FedAvgM check: replay plain rounds and apply the server
        # momentum recurrence to the plain per-round MEANS is non-trivial
        # (momentum changes the trajectory), so verify round-1 exactly:
        # with v0=0, w1 = w0 + slr*(mean_1 - w0); slr=1 -> identical to
        # plain FedAvg on round 1, and diverges after.
        favgm = run(beta=0.9, slr=1.0)
        results[rank] = {
            "base_r1": float(base[0].double().sum()),
            "favgm_r1": float(favgm[0].double().sum()),
            "base_r3": float(base[-1].double().sum()),
            "favgm_r3": float(favgm[-1].double().sum()),
            "prox_drift": float((prox[0] - base[0]).abs().max()),
        }
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_fedprox_and_server_momentum_gloo():
    """FedProx changes the local trajectory; FedAvgM round 1 equals plain
    FedAvg (v0=0, server_lr=1) and diverges afterwards; ranks agree."""
    port = 29531
    with mp.Manager() as man:
        results = man.dict()
        mp.spawn(_prox_momentum_worker, args=(2, port, results), nprocs=2,
                 join=True)
        r0, r1 = results[0], results[1]
    assert r0 == r1  # bitwise identical across ranks (same global state)
    assert abs(r0["favgm_r1"] - r0["base_r1"]) < 1e-9
    assert abs(r0["favgm_r3"] - r0["base_r3"]) > 1e-9
    assert r0["prox_drift"] > 1e-9


def test_fedprox_gradient_math_single_rank():
    """One local step with huge mu: grad = g + mu*(w - w0); with w=w0 at
    the step the prox term is ZERO -> first step matches plain exactly;
    the second step differs."""
    from algorithms.fedavg import _build, make_batches
    from vantage6_amd.parallel.comm import CommContext
    from vantage6_amd.parallel.fedavg import FedAvgConfig, FedAvgTrainer

    comm = CommContext()  # world 1, cpu
    x = torch.randn(64, 8, generator=torch.Generator().manual_seed(3))
    dataset = {"x": x, "y": (x.sum(-1) > 0).long()}
    mk = {"in_features": 8, "num_classes": 2}

    def run(mu, steps):
        net, loss = _build("logreg", mk, "cpu")
        tr = FedAvgTrainer(net, loss, make_batches(dataset, 32, "cpu", seed=5),
                           comm, FedAvgConfig(local_steps=steps, lr=0.1,
                                              momentum=0.0, weight_decay=0.0,
                                              fedprox_mu=mu), n_samples=64)
        tr.run_round()
        return tr.flat.master_flat.clone()

    assert torch.equal(run(0.0, 1), run(25.0, 1))   # prox term is 0 at w=w0
    assert not torch.equal(run(0.0, 2), run(25.0, 2))


def _fedadam_worker(rank, world, port, results):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
    )
    import torch.distributed as dist

    from algorithms.fedavg import _build, make_batches
    from vantage6_amd.parallel.comm import init_from_env
    from vantage6_amd.parallel.fedavg import FedAvgConfig, FedAvgTrainer

    comm = init_from_env(force_backend="gloo")
    try:
        g = torch.Generator().manual_seed(900 + rank)
        x = torch.randn(96, 8, generator=g)
        dataset = {"x": x, "y": (x.sum(-1) > 0).long()}
        mk = {"in_features": 8, "num_classes": 2}

        def run(adam: bool):
            net, loss = _build("logreg", mk, "cpu")
            tr = FedAvgTrainer(
                net, loss, make_batches(dataset, 32, "cpu", seed=9), comm,
                FedAvgConfig(local_steps=1, lr=0.1, momentum=0.0,
                             weight_decay=0.0, server_adam=adam,
                             server_lr=0.5), n_samples=96)
            w0 = tr.flat.master_flat.clone()
            means = []
            for _ in range(2):
                # capture the plain mean by replaying aggregation math:
                # run the round, then reconstruct from returned state
                tr.run_round()
                means.append(tr.flat.master_flat.clone())
            return w0, means

        w0p, plain = run(False)
        w0a, fedadam = run(True)
        # manual replay of FedAdam from the PLAIN means requires identical
        # local trajectories; round 1 starts from the same w0 so mean_1 is
        # shared. Verify round-1 closed form: d = mean1 - w0;
        # m = 0.1*d ... b1=0.9 -> m=(1-b1)d; v=(1-b2)d^2;
        # w1 = w0 + 0.5 * m/(sqrt(v)+tau)
        d = plain[0] - w0p
        m = 0.1 * d
        v = 0.01 * d * d
        want = w0p + 0.5 * m / (v.sqrt() + 1e-3)
        err = float((fedadam[0] - want).abs().max())
        results[rank] = {"err": err,
                         "ck": float(fedadam[-1].double().sum())}
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_fedadam_gloo_two_ranks():
    port = 29541
    with mp.Manager() as man:
        results = man.dict()
        mp.spawn(_fedadam_worker, args=(2, port, results), nprocs=2, join=True)
        r0, r1 = results[0], results[1]
    assert r0["err"] < 1e-6, r0["err"]
    assert r0["ck"] == r1["ck"]


def test_partial_fit_fedprox_math():
    """Station-mode partial_fit: with w=w_round the first prox step is
    exactly plain SGD; a second step diverges under large mu."""
    from algorithms.fedavg import _partial_fit_impl

    x = torch.randn(48, 6, generator=torch.Generator().manual_seed(21))
    db = {"x": x, "y": (x.sum(-1) > 0).long()}
    mk = {"model": "logreg",
          "model_kwargs": {"in_features": 6, "num_classes": 2},
          "batch_size": 16, "lr": 0.1, "seed": 7}

    def run(mu, steps):
        out = _partial_fit_impl(db, local_steps=steps, fedprox_mu=mu, **mk)
        return {k: v.clone() for k, v in out["params"].items()}

    a1, b1 = run(0.0, 1), run(30.0, 1)
    assert all(torch.equal(a1[k], b1[k]) for k in a1)
    a2, b2 = run(0.0, 2), run(30.0, 2)
    assert any(not torch.equal(a2[k], b2[k]) for k in a2)
