"""FedAvg algorithm package.

Two execution modes, mirroring the two planes of the framework:

1. **API mode** (``central_fedavg`` / ``partial_fit``) — the vantage6-style
   control-plane flow: a central task fans out per-round partial-fit
   subtasks via the orchestrator, partials return parameter deltas, the
   central weighted-averages them. Runs anywhere (CPU stations in tests);
   this is capability parity with how FedAvg is actually written on the
   reference.

2. **Engine mode** (``fit_rounds``) — the MI355X data plane: every station
   is a torch.distributed rank; local fit uses the hand-written HIP
   kernels; aggregation is RCCL reduce over xGMI (vantage6_amd.parallel).
   ``bench.py`` and GPU deployments use this path.
"""
from __future__ import annotations

import torch

from vantage6_amd.algorithm import algorithm_client, data
from vantage6_amd.models import build_model, model_loss
from vantage6_amd.parallel.comm import CommContext, init_from_env
from vantage6_amd.parallel.fedavg import FedAvgConfig, FedAvgTrainer


# --------------------------------------------------------------------------
# shared helpers
# --------------------------------------------------------------------------
def make_batches(dataset: dict, batch_size: int, device, dtype=torch.bfloat16,
                 seed: int = 0):
    """Infinite minibatch iterator over a resident synthetic/tensor shard."""
    keys = list(dataset.keys())
    n = len(dataset[keys[0]])
    g = torch.Generator().manual_seed(seed)

    def to_dev(t, k):
        if t.is_floating_point():
            return t.to(device=device, dtype=dtype)
        return t.to(device=device)

    resident = {k: to_dev(v, k) for k, v in dataset.items()}
    while True:
        idx = torch.randint(0, n, (batch_size,), generator=g).to(device)
        yield {k: v[idx] for k, v in resident.items()}


def _build(model_name: str, model_kwargs: dict, device, seed: int = 1234):
    torch.manual_seed(seed)  # identical init across stations
    model = build_model(model_name, device=device, **(model_kwargs or {}))
    return model, model_loss(model_name)


# --------------------------------------------------------------------------
# engine mode — one call per rank/station
# --------------------------------------------------------------------------
def fit_rounds(
    dataset: dict,
    *,
    model: str = "resnet18",
    model_kwargs: dict | None = None,
    rounds: int = 10,
    local_steps: int = 4,
    batch_size: int = 32,
    lr: float = 0.1,
    momentum: float = 0.9,
    weight_decay: float = 5e-4,
    optimizer: str = "sgd",
    secure_agg: bool = False,
    fedprox_mu: float = 0.0,
    server_momentum: float = 0.0,
    server_adam: bool = False,
    server_lr: float = 1.0,
    dp: dict | None = None,
    comm: CommContext | None = None,
    device=None,
    round_hook=None,
    checkpoint: str | None = None,
    checkpoint_every: int = 0,
) -> dict:
    """Engine mode. With ``checkpoint`` set, training resumes from the file
    if it exists and (with ``checkpoint_every`` > 0) saves every N rounds —
    the reference's long-running-task recovery story on
    the engine path."""
    comm = comm or init_from_env()
    device = device or comm.device
    net, loss_fn = _build(model, model_kwargs, device)
    batches = make_batches(dataset, batch_size, device, seed=17 + comm.rank)
    dp_cfg = None
    if dp:
        from vantage6_amd.parallel.dp import DPConfig

        dp_cfg = DPConfig(**dp)
    cfg = FedAvgConfig(
        local_steps=local_steps, lr=lr, momentum=momentum,
        weight_decay=weight_decay, optimizer=optimizer, secure_agg=secure_agg,
        fedprox_mu=fedprox_mu, server_momentum=server_momentum,
        server_adam=server_adam, server_lr=server_lr, dp=dp_cfg,
    )
    n = len(dataset[list(dataset.keys())[0]])
    trainer = FedAvgTrainer(net, loss_fn, batches, comm, cfg, n_samples=n)
    if dp_cfg is not None:
        from vantage6_amd.models.factory import model_loss_per_sample

        trainer.per_sample_loss_fn = model_loss_per_sample(model)
    start_round = 0
    if checkpoint is not None:
        import os as _os

        from vantage6_amd.parallel.checkpoint import (load_checkpoint,
                                                      save_checkpoint)

        if _os.path.exists(checkpoint):
            start_round = load_checkpoint(trainer, checkpoint)
    history = []
    for r in range(start_round, rounds):
        stats = trainer.run_round()
        history.append(stats)
        if round_hook:
            round_hook(trainer, stats)
        if checkpoint is not None and checkpoint_every and \
                (r + 1) % checkpoint_every == 0:
            save_checkpoint(trainer, checkpoint)
    trainer.finalize()  # join any in-flight overlapped aggregation
    out = {
        "rounds": rounds,
        "final_loss": history[-1]["loss"],
        "history": history,
        "n_params": trainer.flat.numel,
        # post-aggregation master checksum: identical across ranks by design
        "master_checksum": float(trainer.flat.master_flat.double().sum().item()),
    }
    if dp_cfg is not None:
        out["dp_privacy"] = trainer.dp_engine.epsilon_report(
            steps=rounds * local_steps, sample_rate=min(1.0, batch_size / n))
    return out


# --------------------------------------------------------------------------
# API mode — the vantage6-style central/partial pair
# --------------------------------------------------------------------------
@algorithm_client
def central_fedavg(
    client,
    model: str = "logreg",
    model_kwargs: dict | None = None,
    rounds: int = 3,
    local_steps: int = 2,
    batch_size: int = 16,
    lr: float = 0.1,
    min_stations: int = 1,
    fedprox_mu: float = 0.0,  # FedProx proximal term on the local steps
    fail_orgs: list | None = None,  # fault injection (tests): orgs that crash
):
    """Central part: per round, fan out partial_fit and weighted-average.

    Elastic: a station whose run crashes or is killed
    simply drops out of that round's weighted mean — FedAvg tolerates it;
    the round fails only if fewer than ``min_stations`` complete.
    """
    org_ids = [o["id"] for o in client.organization.list()]
    # round 0 params come from a deterministic seed on every station
    global_params: dict | None = None
    history = []
    for r in range(rounds):
        task = client.task.create(
            input_={
                "method": "partial_fit",
                "kwargs": {
                    "model": model, "model_kwargs": model_kwargs,
                    "params": global_params, "local_steps": local_steps,
                    "batch_size": batch_size, "lr": lr, "seed": 1234,
                    "fedprox_mu": fedprox_mu, "fail_orgs": fail_orgs,
                },
            },
            organizations=org_ids,
            name=f"fedavg round {r}",
        )
        results = client.wait_for_results(task["id"], timeout=600,
                                          allow_failures=True)
        if len(results) < min_stations:
            raise RuntimeError(
                f"round {r}: only {len(results)} stations completed "
                f"(min_stations={min_stations})"
            )
        total = sum(res["n_samples"] for res in results)
        avg = None
        for res in results:
            w = res["n_samples"] / total
            p = res["params"]
            if avg is None:
                avg = {k: v.float() * w for k, v in p.items()}
            else:
                for k, v in p.items():
                    avg[k] += v.float() * w
        global_params = {k: v.to(torch.bfloat16) for k, v in avg.items()}
        history.append({"round": r, "mean_loss":
                        sum(res["loss"] for res in results) / len(results)})
    return {"params": global_params, "history": history}


def _partial_fit_impl(
    db,
    model: str = "logreg",
    model_kwargs: dict | None = None,
    params: dict | None = None,
    local_steps: int = 2,
    batch_size: int = 16,
    lr: float = 0.1,
    seed: int = 1234,
    fedprox_mu: float = 0.0,
    fail_orgs: list | None = None,
    _v6_env: dict | None = None,
):
    """Partial: local SGD steps on this station's shard; return new params."""
    env = _v6_env or {}
    if fail_orgs and int(env.get("ORGANIZATION_ID", -1) or -1) in fail_orgs:
        raise RuntimeError("injected fault: station down mid-round")
    device = "cuda" if torch.cuda.is_available() else "cpu"
    net, loss_fn = _build(model, model_kwargs, device, seed=seed)
    if params is not None:
        state = {k: v.to(device) for k, v in params.items()}
        net.load_state_dict(state, strict=True)
    dataset = db if isinstance(db, dict) else {"x": db}
    batches = make_batches(dataset, batch_size, device)
    opt = torch.optim.SGD(net.parameters(), lr=lr, momentum=0.9)
    prox_ref = ([p.detach().clone() for p in net.parameters()]
                if fedprox_mu > 0 else None)  # w_round = the received global
    losses = []
    for _ in range(local_steps):
        batch = next(batches)
        opt.zero_grad(set_to_none=True)
        loss = loss_fn(net, batch)
        loss.backward()
        if prox_ref is not None:
            with torch.no_grad():
                for pp, r0 in zip(net.parameters(), prox_ref):
                    if pp.grad is not None:
                        pp.grad.add_((pp.detach() - r0).to(pp.grad.dtype),
                                     alpha=fedprox_mu)
        opt.step()
        losses.append(float(loss.detach()))
    n = len(dataset[list(dataset.keys())[0]])
    return {
        "params": {k: v.cpu() for k, v in net.state_dict().items()
                   if v.dtype.is_floating_point},
        "loss": sum(losses) / len(losses),
        "n_samples": n,
    }


_partial_fit_impl._v6_wants_env = True  # @data forwards the env map through
partial_fit = data(1)(_partial_fit_impl)
