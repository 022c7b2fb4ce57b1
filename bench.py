#!/usr/bin/env python3
"""Flagship benchmark — FedAvg rounds/sec (BASELINE.json metric).

Default config (headline, BASELINE config[2]): FedAvg ResNet-18 bf16 on
synthetic 224×224 shards, N stations = N GPUs (one rank per GPU over RCCL),
local_steps SGD steps per round + weighted parameter all-reduce.

Contract (driver): ``python bench.py --gpus N --steps K --warmup W``;
for N>1 launched via torch.distributed.run with one rank per GPU. Does W
untimed warmup rounds, times exactly K rounds between barrier+synchronize
pairs, takes the MAX elapsed over ranks, and rank 0 prints ONE JSON line.

A "step" here is one FedAvg ROUND (local_steps local minibatch SGD steps +
aggregation) — the reference's unit of work (one central
round-trip of subtask fan-out + fit + collect).
"""
from __future__ import annotations

import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent))

import torch

from algorithms.fedavg import make_batches, _build
from vantage6_amd.parallel.comm import init_from_env
from vantage6_amd.parallel.fedavg import FedAvgConfig, FedAvgTrainer


def synthetic_dataset(model: str, n: int, seed: int, model_kwargs: dict):
    g = torch.Generator().manual_seed(seed)
    if model in ("resnet18", "resnet50", "vit_b16"):
        return {
            "x": torch.randn(n, 3, 224, 224, generator=g),
            "y": torch.randint(0, model_kwargs.get("num_classes", 1000), (n,), generator=g),
        }
    if model == "logreg":
        d = model_kwargs.get("in_features", 128)
        c = model_kwargs.get("num_classes", 2)
        x = torch.randn(n, d, generator=g)
        w = torch.randn(d, c, generator=g)
        y = (x @ w).argmax(-1)
        return {"x": x, "y": y}
    # token models
    seq = model_kwargs.pop("_seq_len", 512)
    vocab = model_kwargs.get("vocab", 30522 if model.startswith("bert") else 128256)
    # seq+1 tokens: LM loss shifts by one, so the model still sees seq_len
    # positions and GEMM M-dims stay tile-aligned (batch*seq multiples of 256)
    return {"input_ids": torch.randint(0, vocab, (n, seq + 1), generator=g)}


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20, help="timed FedAvg rounds")
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--model", default="resnet18",
                    choices=["resnet18", "resnet50", "vit_b16", "logreg",
                             "bert_base", "llama_1b"])
    ap.add_argument("--batch-size", type=int, default=None, help="per-station batch")
    ap.add_argument("--local-steps", type=int, default=4)
    ap.add_argument("--seq-len", type=int, default=512)
    ap.add_argument("--shard-size", type=int, default=None, help="samples per station")
    ap.add_argument("--secure-agg", action="store_true")
    ap.add_argument("--dp", action="store_true", help="DP-SGD (ghost clipping)")
    ap.add_argument("--agg-algo", default="allreduce",
                    choices=["allreduce", "reduce_bcast"])
    ap.add_argument("--overlap", dest="overlap", action="store_true", default=True,
                    help="overlap round-k collective with round-k+1 forward (default)")
    ap.add_argument("--no-overlap", dest="overlap", action="store_false")
    ap.add_argument("--channels-last", dest="channels_last", action="store_true",
                    default=True, help="NHWC layout for conv models (default)")
    ap.add_argument("--no-channels-last", dest="channels_last",
                    action="store_false")
    ap.add_argument("--graph", dest="graph", action="store_true", default=True,
                    help="hipGraph-capture the local step (default)")
    ap.add_argument("--no-graph", dest="graph", action="store_false")
    ap.add_argument("--dtype", default="bf16", choices=["bf16"])
    ap.add_argument("--device", default=None, help="override (e.g. cpu for smoke)")
    ap.add_argument("--target-loss", type=float, default=None,
                    help="also report rounds/seconds until loss < target "
                         "(BASELINE metric's wall-clock-to-target-loss half; "
                         "defaults to a per-model target below random-init loss)")
    args = ap.parse_args()
    if args.target_loss is None:
        # BASELINE.json names "wall-clock-to-target-loss" as half the
        # headline metric — emit it by DEFAULT. Targets sit meaningfully
        # below the random-init loss of each model (ln(1000)=6.9,
        # ln(30522)=10.3, ln(128256)=11.8) so hitting them requires real
        # optimization progress on the synthetic shards.
        args.target_loss = {"resnet18": 6.0, "resnet50": 6.0,
                            "vit_b16": 6.5, "logreg": 0.45,
                            "bert_base": 9.5, "llama_1b": 11.0}[args.model]

    comm = init_from_env()
    device = torch.device(args.device) if args.device else comm.device
    on_gpu = device.type == "cuda"
    if on_gpu:
        torch.backends.cudnn.benchmark = True  # MIOpen find-mode
        torch.backends.cuda.matmul.allow_bf16_reduced_precision_reduction = True

    defaults = {
        "resnet18": {"batch": 32, "kwargs": {"num_classes": 1000}, "shard": 256},
        "resnet50": {"batch": 32, "kwargs": {"num_classes": 1000}, "shard": 256},
        "vit_b16": {"batch": 32, "kwargs": {"num_classes": 1000}, "shard": 256},
        "logreg": {"batch": 1024, "kwargs": {"in_features": 128, "num_classes": 2},
                   "shard": 16384},
        "bert_base": {"batch": 8, "kwargs": {}, "shard": 64},
        "llama_1b": {"batch": 2, "kwargs": {}, "shard": 16},
    }[args.model]
    batch = args.batch_size or defaults["batch"]
    shard = args.shard_size or max(defaults["shard"], batch)
    mk = dict(defaults["kwargs"])
    if args.model in ("bert_base", "llama_1b"):
        mk["_seq_len"] = args.seq_len

    dataset = synthetic_dataset(args.model, shard, seed=100 + comm.rank, model_kwargs=mk)
    mk.pop("_seq_len", None)

    net, loss_fn = _build(args.model, mk, device)
    if args.channels_last and args.model in ("resnet18", "resnet50"):
        net = net.to(memory_format=torch.channels_last)
        inner = loss_fn

        def loss_fn(model, batch):  # noqa: F811
            batch = dict(batch, x=batch["x"].contiguous(
                memory_format=torch.channels_last))
            return inner(model, batch)

    batches = make_batches(dataset, batch, device, seed=17 + comm.rank)
    dp_cfg = None
    if args.dp:
        from vantage6_amd.parallel.dp import DPConfig

        dp_cfg = DPConfig(max_grad_norm=1.0, noise_multiplier=1.0)
    cfg = FedAvgConfig(
        local_steps=args.local_steps,
        lr=0.1 if args.model in ("resnet18", "resnet50", "logreg") else 1e-4,
        optimizer="sgd" if args.model in ("resnet18", "resnet50", "logreg")
        else "adamw",  # vit trains with adamw like the transformers
        secure_agg=args.secure_agg,
        dp=dp_cfg,
        agg_algo=args.agg_algo,
        overlap=args.overlap,
    )
    trainer = FedAvgTrainer(net, loss_fn, batches, comm, cfg, n_samples=shard)
    if dp_cfg is not None:
        from vantage6_amd.models.factory import model_loss_per_sample

        trainer.per_sample_loss_fn = model_loss_per_sample(args.model)

    for _ in range(args.warmup):
        trainer.run_round()
    trainer.finalize()  # drain warmup's in-flight aggregation
    used_graph = trainer.enable_graph() if (on_gpu and args.graph) else False

    from vantage6_amd.utils.metrics import RoundMetrics

    metrics = RoundMetrics()
    comm.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    losses = []
    for _ in range(args.steps):
        stats = trainer.run_round()
        metrics.record(stats)
        losses.append(stats["loss"])
    trainer.finalize()  # join in-flight overlapped aggregation
    comm.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # MAX elapsed over ranks
    t = torch.tensor([elapsed], device=device if on_gpu else "cpu")
    if comm.is_distributed:
        import torch.distributed as dist

        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    n = comm.world_size
    rounds_per_sec = args.steps / elapsed
    if comm.rank == 0:
        out = {
            "metric": "FedAvg rounds/sec (ResNet-18 @ N stations)"
            if args.model == "resnet18" else f"FedAvg rounds/sec ({args.model})",
            "value": rounds_per_sec,
            "unit": "rounds/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": batch * n,
                "per_station_batch": batch,
                "local_steps": args.local_steps,
                "seq_len": args.seq_len if args.model in ("bert_base", "llama_1b") else None,
                "parallelism": f"federated-dp{n}",
                "samples_per_round": batch * args.local_steps * n,
                "samples_per_sec": rounds_per_sec * batch * args.local_steps * n,
                "secure_agg": args.secure_agg,
                "dp": args.dp,
                "hip_graph": used_graph,
                "overlap": bool(args.overlap and comm.is_distributed
                                and not args.secure_agg and on_gpu),
                "overlap_stats": trainer.overlap_stats or None,
                "dp_privacy": (trainer.dp_engine.epsilon_report(
                    steps=args.steps * args.local_steps,
                    sample_rate=min(1.0, batch / shard))
                    if args.dp else None),
                "loss_first": losses[0],
                "final_loss": losses[-1],
                "n_params": trainer.flat.numel,
                "hip_ops": __import__("vantage6_amd.ops", fromlist=["has_hip_ops"]).has_hip_ops(),
            },
        }
        if args.target_loss is not None:
            hit = next((i for i, l in enumerate(losses)
                        if l == l and l < args.target_loss), None)
            out["config"]["target_loss"] = args.target_loss
            out["config"]["rounds_to_target"] = None if hit is None else hit + 1
            out["config"]["secs_to_target"] = (
                None if hit is None else (hit + 1) * elapsed / args.steps)
        # per-round timing table on stderr; stdout stays
        # the single JSON contract line
        print(metrics.table(last=min(args.steps, 12)), file=sys.stderr)
        print(json.dumps(out))


if __name__ == "__main__":
    main()
