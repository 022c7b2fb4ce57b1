"""Checkpoint / resume for the FedAvg engine.

Reference parity note: the reference has no ML-sense
checkpointing (durable state = server SQL DB). Here: global master weights
+ optimizer state + round counter via safetensors every K rounds; rank 0
writes, resume loads + broadcasts so every station restarts identically.
"""
from __future__ import annotations

import json
from pathlib import Path

import torch
from safetensors.torch import load_file, save_file


def save_checkpoint(trainer, path: str | Path) -> None:
    path = Path(path)
    path.parent.mkdir(parents=True, exist_ok=True)
    if hasattr(trainer, "finalize"):
        trainer.finalize()  # drain any in-flight overlapped aggregation
    if trainer.comm.rank == 0:
        tensors = {
            "master": trainer.flat.master_flat.cpu().contiguous(),
            "momentum": trainer.flat.momentum_flat.cpu().contiguous(),
        }
        if trainer.flat.exp_avg_sq is not None:
            tensors["exp_avg_sq"] = trainer.flat.exp_avg_sq.cpu().contiguous()
        # server-optimizer state (FedAvgM momentum / FedAdam moments) —
        # without it a resumed run silently restarts the server recurrence
        if getattr(trainer, "_srv_v", None) is not None:
            tensors["srv_v"] = trainer._srv_v.cpu().contiguous()
        if getattr(trainer, "_srv_m", None) is not None:
            tensors["srv_m"] = trainer._srv_m.cpu().contiguous()
        save_file(tensors, str(path))
        if trainer.flat._step_dev is not None:  # device counter is authoritative
            trainer.flat._step = int(trainer.flat._step_dev.item())
        meta = {"round": trainer.round_idx, "numel": trainer.flat.numel,
                "optimizer": trainer.cfg.optimizer, "step": trainer.flat._step}
        Path(str(path) + ".json").write_text(json.dumps(meta))
    trainer.comm.barrier()


def load_checkpoint(trainer, path: str | Path) -> int:
    """Restore state on every rank; returns the round counter."""
    path = Path(path)
    meta = json.loads(Path(str(path) + ".json").read_text())
    if meta["numel"] != trainer.flat.numel:
        raise ValueError(
            f"checkpoint has {meta['numel']} params, model has {trainer.flat.numel}"
        )
    tensors = load_file(str(path))
    dev = trainer.flat.master_flat.device
    trainer.flat.master_flat.copy_(tensors["master"].to(dev))
    trainer.flat.momentum_flat.copy_(tensors["momentum"].to(dev))
    if "exp_avg_sq" in tensors:
        if trainer.flat.exp_avg_sq is None:
            trainer.flat.exp_avg_sq = torch.zeros_like(trainer.flat.master_flat)
        trainer.flat.exp_avg_sq.copy_(tensors["exp_avg_sq"].to(dev))
    if "srv_v" in tensors:
        if getattr(trainer, "_srv_v", None) is None:
            trainer._srv_v = torch.zeros_like(trainer.flat.master_flat)
            trainer._srv_old = torch.empty_like(trainer.flat.master_flat)
        trainer._srv_v.copy_(tensors["srv_v"].to(dev))
    if "srv_m" in tensors:
        if getattr(trainer, "_srv_m", None) is None:
            trainer._srv_m = torch.zeros_like(trainer.flat.master_flat)
        trainer._srv_m.copy_(tensors["srv_m"].to(dev))
    trainer.flat._step = meta.get("step", 0)
    # all ranks read the same file in tests; in production rank 0's state is
    # authoritative — broadcast ALL optimizer state (incl. second moment and
    # the bias-correction step counter) to make divergence impossible
    trainer.comm.broadcast(trainer.flat.master_flat, src=0)
    trainer.comm.broadcast(trainer.flat.momentum_flat, src=0)
    if trainer.flat.exp_avg_sq is not None:
        trainer.comm.broadcast(trainer.flat.exp_avg_sq, src=0)
    if getattr(trainer, "_srv_v", None) is not None:
        trainer.comm.broadcast(trainer._srv_v, src=0)
    if getattr(trainer, "_srv_m", None) is not None:
        trainer.comm.broadcast(trainer._srv_m, src=0)
    step_t = torch.tensor([trainer.flat._step], dtype=torch.long, device=dev)
    trainer.comm.broadcast(step_t, src=0)
    trainer.flat._step = int(step_t.item())
    if trainer.flat._step_dev is not None:
        trainer.flat._step_dev.fill_(trainer.flat._step)
    trainer.flat.sync_param_from_master()
    trainer.round_idx = meta["round"]
    return trainer.round_idx
