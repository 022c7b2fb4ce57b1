"""FedAvg engine — the per-round hot loop.

Reference parity: one vantage6 FedAvg round is
{central task → N partial subtasks over REST → Docker fit → poll results →
aggregate in Python}. Here a round is:

1. ``local_steps`` SGD/AdamW steps on the station's resident shard
   (bf16 compute, hand-written fused optimizer, no grad sync — stations
   intentionally diverge, that is the federated part),
2. weighted FedAvg over the flat fp32 master buffer: ``master *= w_i`` then
   bucketed all-reduce(SUM) over RCCL/xGMI, overlapped bucket-by-bucket
   with the master→bf16 param copy,
3. optional secure aggregation: fixed-point int64 quantization + pairwise
   Philox masks that cancel EXACTLY under modular int64 summation (fp32
   masking cannot cancel bit-exactly under reduction reordering — SURVEY.md
   §8.3 hard part #3).
"""
from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Callable, Iterator, Optional

import torch

from .comm import CommContext
from .dp import DPConfig, DPEngine
from .flat import FlatParamSpace

_FIXED_POINT_BITS = 32  # fraction bits for secure-agg quantization


@dataclass
class FedAvgConfig:
    local_steps: int = 4
    lr: float = 0.1
    momentum: float = 0.9
    weight_decay: float = 5e-4
    nesterov: bool = False
    optimizer: str = "sgd"  # "sgd" | "adamw"
    bucket_bytes: int = 64 << 20
    secure_agg: bool = False
    secagg_seed: int = 0x5ECA66
    lr_schedule: Optional[Callable[[int], float]] = None
    dp: Optional[DPConfig] = None  # per-sample clip + noise (ghost clipping)
    # "allreduce": one bucketed all-reduce (ring; per-link bound on xGMI).
    # "reduce_bcast": reduce-to-root + broadcast — on the fully-connected
    # 7-link xGMI topology the root's links aggregate ~1 TB/s in/out
    # (SURVEY.md §6.8); pick per measurement on the 8-GPU node.
    agg_algo: str = "allreduce"


class FedAvgTrainer:
    def __init__(
        self,
        model: torch.nn.Module,
        loss_fn: Callable,
        batches: Iterator,
        comm: CommContext,
        cfg: FedAvgConfig,
        n_samples: int = 1,
    ):
        self.model = model
        self.loss_fn = loss_fn
        self.batches = batches
        self.comm = comm
        self.cfg = cfg
        # scattered grads (no zero+accumulate passes) on GPU unless DP
        # needs the flat grad buffer for clip/noise
        scatter = comm.device.type == "cuda" and cfg.dp is None
        self.flat = FlatParamSpace(model, device=comm.device,
                                   scatter_grads=scatter)
        self.round_idx = 0
        # FedAvg weight w_i = n_i / sum_j n_j
        n = torch.tensor([float(n_samples)], device=comm.device)
        total = n.clone()
        self.comm.all_reduce(total)
        self.weight = float(n.item() / total.item())
        # make every rank start from identical params (rank-0's init)
        self.comm.broadcast(self.flat.master_flat, src=0)
        self.flat.sync_param_from_master()
        self.comm_stream = (
            torch.cuda.Stream() if comm.device.type == "cuda" else None
        )
        self.last_stats: dict = {}
        self.dp_engine = DPEngine(model, cfg.dp) if cfg.dp else None
        self.per_sample_loss_fn: Optional[Callable] = None  # set when dp on
        self.last_dp_stats: dict = {}

    # -------------------------------------------------- hipGraph local step
    def enable_graph(self) -> bool:
        """Capture {zero_grad → fwd → bwd → fused optimizer} into one
        hipGraph: the ResNet round is ~19% launch-gap idle eager (≈250
        launches/step). Flat grads make this safe — every gradient buffer
        address is static. Call AFTER warmup (MIOpen find must be done).
        Freezes lr/hyperparams at capture values; not for DP (dynamic
        clipping control flow). AdamW is capturable because its bias
        correction derives from an int32 DEVICE step counter that the
        captured tick kernel advances on every replay."""
        if self.comm.device.type != "cuda" or self.dp_engine is not None:
            return False
        try:
            batch = next(self.batches)
            self._static_batch = {k: v.clone() for k, v in batch.items()}
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):
                    self._graph_body()
            torch.cuda.current_stream().wait_stream(side)
            self._graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self._graph):
                self._static_loss = self._graph_body()
            if self.flat.scatter_grads:
                # replays reuse the capture pool's grad addresses: point the
                # persistent chunk table at them now (outside capture);
                # strict: any layout conversion would break replay
                self.flat.refresh_sgd_table(strict=True)
            return True
        except Exception:  # noqa: BLE001 — fall back to eager
            import logging

            logging.getLogger(__name__).exception("hipGraph capture failed")
            self._graph = None
            return False

    def _graph_body(self):
        self.flat.zero_grad()
        loss = self.loss_fn(self.model, self._static_batch)
        loss.backward()
        lr = self.cfg.lr_schedule(self.round_idx) if self.cfg.lr_schedule else self.cfg.lr
        if self.cfg.optimizer == "adamw":
            self.flat.adamw_step(lr=lr, weight_decay=self.cfg.weight_decay)
        else:
            self.flat.sgd_step(lr=lr, momentum=self.cfg.momentum,
                               weight_decay=self.cfg.weight_decay,
                               nesterov=self.cfg.nesterov)
        return loss

    # ------------------------------------------------------------ local fit
    def local_step(self) -> float:
        batch = next(self.batches)
        if getattr(self, "_graph", None) is not None:
            for k, v in batch.items():
                self._static_batch[k].copy_(v, non_blocking=True)
            self._graph.replay()
            return float(self._static_loss.detach())
        if self.dp_engine is not None:
            return self._dp_local_step(batch)
        self.flat.zero_grad()
        loss = self.loss_fn(self.model, batch)
        loss.backward()
        lr = self.cfg.lr_schedule(self.round_idx) if self.cfg.lr_schedule else self.cfg.lr
        if self.cfg.optimizer == "adamw":
            self.flat.adamw_step(lr=lr, weight_decay=self.cfg.weight_decay)
        else:
            self.flat.sgd_step(
                lr=lr, momentum=self.cfg.momentum,
                weight_decay=self.cfg.weight_decay, nesterov=self.cfg.nesterov,
            )
        return float(loss.detach())

    def _dp_local_step(self, batch) -> float:
        if self.per_sample_loss_fn is None:
            raise RuntimeError("DP-SGD needs trainer.per_sample_loss_fn set")
        losses = self.per_sample_loss_fn(self.model, batch)
        B = losses.shape[0]
        self.last_dp_stats = self.dp_engine.compute_clipped_grads(
            losses, self.flat.zero_grad, B
        )
        self.dp_engine.add_noise_and_scale(self.flat.grad_flat, B)
        lr = self.cfg.lr_schedule(self.round_idx) if self.cfg.lr_schedule else self.cfg.lr
        if self.cfg.optimizer == "adamw":
            self.flat.adamw_step(lr=lr, weight_decay=self.cfg.weight_decay)
        else:
            self.flat.sgd_step(lr=lr, momentum=self.cfg.momentum,
                               weight_decay=self.cfg.weight_decay)
        return float(losses.detach().mean())

    # ----------------------------------------------------------- aggregation
    def aggregate(self) -> None:
        if self.cfg.secure_agg:
            self._aggregate_secure()
        else:
            self._aggregate_plain()

    def _aggregate_plain(self) -> None:
        master = self.flat.master_flat
        if not self.comm.is_distributed:
            self.flat.sync_param_from_master()
            return
        master.mul_(self.weight)
        bucket = max(1, self.cfg.bucket_bytes // 4)
        works = []
        use_rb = self.cfg.agg_algo == "reduce_bcast"
        for off in range(0, master.numel(), bucket):
            chunk = master[off : off + bucket]
            if use_rb:
                w = self.comm.reduce(chunk, dst=0, async_op=True)
            else:
                w = self.comm.all_reduce(chunk, async_op=True)
            works.append((off, chunk, w))
        if use_rb:
            for off, chunk, work in works:
                if work is not None:
                    work.wait()
            bworks = [(off, chunk, self.comm.broadcast(chunk, src=0, async_op=True))
                      for off, chunk, _ in works]
            works = bworks
        # overlap: convert each bucket to bf16 params as soon as it lands
        for off, chunk, work in works:
            if work is not None:
                work.wait()
            self.flat.param_flat[off : off + chunk.numel()].copy_(
                chunk.to(self.flat.dtype)
            )

    def _aggregate_secure(self) -> None:
        """Fixed-point + pairwise-mask secure aggregation.

        q_i = round(w_i * master_i * 2^F) as int64, masked with
        sum-cancelling pairwise Philox masks (ops/secagg kernel on GPU;
        torch reference on CPU); all-reduce(SUM) wraps mod 2^64 so masks
        cancel exactly; dequantize. The server/other stations only ever see
        masked integers.

        Limitation (same trade-off the reference ecosystem's basic
        pairwise-mask schemes make): no dropout recovery — if a rank dies
        MID-collective, its pair masks do not cancel and the round result
        is garbage; the collective itself also fails, so in practice the
        job restarts from the last checkpoint. Bonawitz-style secret-shared
        mask recovery is a round-2+ item.
        """
        from ..ops import secagg as secagg_ops

        master = self.flat.master_flat
        if not self.comm.is_distributed:
            self.flat.sync_param_from_master()
            return
        q = torch.empty(master.numel(), dtype=torch.int64, device=master.device)
        scale = float(1 << _FIXED_POINT_BITS)
        secagg_ops.quantize_mask(
            q, master, self.weight, scale,
            seed=self.cfg.secagg_seed + self.round_idx,
            rank=self.comm.rank, world_size=self.comm.world_size,
        )
        self.comm.all_reduce(q)
        master.copy_(q.to(torch.float64).mul_(1.0 / scale).to(torch.float32))
        self.flat.sync_param_from_master()

    # ---------------------------------------------------------------- rounds
    def run_round(self) -> dict:
        t0 = time.perf_counter()
        losses = [self.local_step() for _ in range(self.cfg.local_steps)]
        if self.comm.device.type == "cuda":
            torch.cuda.synchronize()
        t1 = time.perf_counter()
        self.aggregate()
        if self.comm.device.type == "cuda":
            torch.cuda.synchronize()
        t2 = time.perf_counter()
        self.round_idx += 1
        self.last_stats = {
            "round": self.round_idx,
            "loss": sum(losses) / len(losses),
            "fit_ms": (t1 - t0) * 1e3,
            "agg_ms": (t2 - t1) * 1e3,
            "round_ms": (t2 - t0) * 1e3,
        }
        return self.last_stats
