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
    # fit/agg overlap: the bucketed collective of round k
    # runs on a dedicated HIP stream while round k+1's forward begins;
    # per-bucket events gate each module's forward on exactly the buckets
    # that hold its parameters, so the semantics are EXACTLY FedAvg — the
    # forward simply starts before the full collective has landed. GPU +
    # distributed + non-secure only; CPU/gloo and secure-agg stay
    # synchronous.
    overlap: bool = True
    # FedProx (Li et al. 2020): local objective gains mu/2*||w - w_round||^2,
    # i.e. grad += mu*(w - w_round) before every optimizer step. 0 = plain
    # FedAvg (default; the reference ecosystem's FedAvg algorithms).
    fedprox_mu: float = 0.0
    # FedAvgM (Hsu et al. 2019): server-side momentum over round deltas —
    # v = beta*v + (mean_w - w_old); w = w_old + server_lr*v. beta 0 = plain
    # averaging (default). Disables fit/agg overlap when on (the sync
    # aggregation path applies the transform per bucket).
    server_momentum: float = 0.0
    server_lr: float = 1.0
    # FedAdam (Reddi et al. 2021): adaptive server update over round
    # deltas — m = b1*m + (1-b1)*d; v = b2*v + (1-b2)*d^2;
    # w = w_old + server_lr * m / (sqrt(v) + tau). Mutually exclusive with
    # server_momentum; also sync-path only.
    server_adam: bool = False
    server_adam_betas: tuple = (0.9, 0.99)
    server_adam_tau: float = 1e-3


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
        # ---- fit/agg overlap state --------------------------------------
        self._agg_flat: Optional[torch.Tensor] = None  # collective workspace
        self._agg_events: list = []  # [(lo, hi, cuda.Event)] per bucket
        self._bucket_waited: list = []
        self._agg_pending = False
        self._stall_pairs: list = []  # (ev_a, ev_b) compute-stream stalls
        self._ev_agg_start: Optional[torch.cuda.Event] = None
        self._ev_agg_end: Optional[torch.cuda.Event] = None
        self.overlap_stats: dict = {}  # comm_ms / exposed_ms / overlap_pct
        if self._overlap_capable():
            self._install_overlap_hooks()
        # ---- FedProx / FedAvgM state ------------------------------------
        self._prox_ref: list = []  # per-param round-start snapshots
        if cfg.fedprox_mu > 0:
            if cfg.dp is not None:
                raise ValueError("fedprox_mu with DP-SGD is unsupported "
                                 "(per-sample clipping of the prox term is "
                                 "not defined here)")
            self._prox_params = [p for p in self.model.parameters()
                                 if p.requires_grad]
            self._prox_ref = [p.detach().clone() for p in self._prox_params]
        self._srv_v = None  # server momentum / adam second-moment buffer
        self._srv_m = None  # adam first moment
        self._srv_old = None
        if cfg.server_adam and cfg.server_momentum:
            raise ValueError("server_adam and server_momentum are exclusive")

    # ------------------------------------------------- fit/agg overlap plumbing
    def _overlap_capable(self) -> bool:
        return (self.cfg.overlap and self.comm.device.type == "cuda"
                and self.comm.is_distributed and not self.cfg.secure_agg
                and self.cfg.server_momentum == 0.0
                and not self.cfg.server_adam)

    def _install_overlap_hooks(self) -> None:
        """Per-module forward-pre-hooks: wait on exactly the collective
        buckets that hold this module's parameters (timed, to report the
        exposed fraction of the collective)."""
        by_param = {id(p): (off, n) for p, off, n, _ in self.flat._specs}
        for mod in self.model.modules():
            lo, hi = None, 0
            for p in mod.parameters(recurse=False):
                if id(p) in by_param:
                    off, n = by_param[id(p)]
                    lo = off if lo is None else min(lo, off)
                    hi = max(hi, off + n)
            if lo is None:
                continue
            mod.register_forward_pre_hook(self._make_overlap_hook(lo, hi))

    def _make_overlap_hook(self, mlo: int, mhi: int):
        def hook(mod, inputs):
            if not self._agg_pending:
                return
            cur = torch.cuda.current_stream()
            waited = False
            ev_a = ev_b = None
            for i, (lo, hi, ev) in enumerate(self._agg_events):
                if not self._bucket_waited[i] and lo < mhi and hi > mlo:
                    if not waited:
                        ev_a = torch.cuda.Event(enable_timing=True)
                        ev_a.record(cur)
                        waited = True
                    cur.wait_event(ev)
                    self._bucket_waited[i] = True
            if waited:
                ev_b = torch.cuda.Event(enable_timing=True)
                ev_b.record(cur)
                self._stall_pairs.append((ev_a, ev_b))
                if all(self._bucket_waited):
                    self._agg_pending = False

        return hook

    def _join_agg(self) -> None:
        """Make the compute stream wait for every outstanding collective
        bucket (called before anything writes master/params: the optimizer
        step, a graph replay, checkpointing)."""
        if not self._agg_pending:
            return
        cur = torch.cuda.current_stream()
        ev_a = torch.cuda.Event(enable_timing=True)
        ev_a.record(cur)
        for i, (lo, hi, ev) in enumerate(self._agg_events):
            if not self._bucket_waited[i]:
                cur.wait_event(ev)
                self._bucket_waited[i] = True
        ev_b = torch.cuda.Event(enable_timing=True)
        ev_b.record(cur)
        self._stall_pairs.append((ev_a, ev_b))
        self._agg_pending = False

    def finalize(self) -> None:
        """Join any in-flight aggregation and drain the device (call before
        reading final parameters or exiting the round loop)."""
        self._join_agg()
        if self.comm.device.type == "cuda":
            torch.cuda.synchronize()

    def _collect_overlap_stats(self) -> None:
        """Called at the next aggregate launch: by then last round's events
        are complete, so elapsed_time is free."""
        if self._ev_agg_start is None:
            return
        try:
            comm_ms = self._ev_agg_start.elapsed_time(self._ev_agg_end)
            exposed_ms = sum(a.elapsed_time(b) for a, b in self._stall_pairs)
        except RuntimeError:  # events not yet complete (shouldn't happen)
            return
        self.overlap_stats = {
            "agg_comm_ms": comm_ms,
            "agg_exposed_ms": exposed_ms,
            "overlap_pct": max(0.0, 100.0 * (1.0 - exposed_ms / comm_ms))
            if comm_ms > 0 else 0.0,
        }
        self._stall_pairs = []

    # -------------------------------------------------- hipGraph local step
    def enable_graph(self) -> bool:
        """Capture {zero_grad → fwd → bwd → fused optimizer} into one
        hipGraph: the ResNet round is ~19% launch-gap idle eager (≈250
        launches/step). Flat grads make this safe — every gradient buffer
        address is static. Call AFTER warmup (MIOpen find must be done).
        Freezes lr/hyperparams at capture values. AdamW is capturable
        because its bias correction derives from an int32 DEVICE step
        counter that the captured tick kernel advances on every replay.
        DP-SGD is capturable too: the clip factors are a device tensor
        multiplied into the per-sample losses (no host control flow), the
        noise seed advances via a device counter, and clip stats land in a
        persistent device buffer (dp_engine.stats())."""
        if self.comm.device.type != "cuda":
            return False
        try:
            self.finalize()  # no in-flight collective may touch params mid-capture
            # warmup + capture run ~4 real optimizer steps — snapshot the
            # training state so enabling the graph does not perturb the
            # trajectory (exact eager↔graph equivalence)
            snap = {k: (v.clone() if torch.is_tensor(v) else v)
                    for k, v in self.flat.state_dict().items()
                    if v is not None}
            dp_tick_snap = None
            if self.dp_engine is not None and self.dp_engine._step_dev is not None:
                dp_tick_snap = (self.dp_engine._step_dev.clone(),
                                self.dp_engine._step)
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
            self.flat.load_state_dict(snap)
            if dp_tick_snap is not None:
                self.dp_engine._step_dev.copy_(dp_tick_snap[0])
                self.dp_engine._step = dp_tick_snap[1]
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
        if self.dp_engine is not None:
            losses = self.per_sample_loss_fn(self.model, self._static_batch)
            B = losses.shape[0]
            self.dp_engine.compute_clipped_grads(
                losses, self.flat.zero_grad, B, sync_stats=False)
            self.dp_engine.add_noise_and_scale(self.flat.grad_flat, B)
            loss = losses.detach().mean()
        else:
            self.flat.zero_grad()
            loss = self.loss_fn(self.model, self._static_batch)
            loss.backward()
            if self.cfg.fedprox_mu > 0:
                self._apply_prox_grad()
        lr = self.cfg.lr_schedule(self.round_idx) if self.cfg.lr_schedule else self.cfg.lr
        if self.cfg.optimizer == "adamw":
            self.flat.adamw_step(lr=lr, weight_decay=self.cfg.weight_decay)
        else:
            self.flat.sgd_step(lr=lr, momentum=self.cfg.momentum,
                               weight_decay=self.cfg.weight_decay,
                               nesterov=self.cfg.nesterov)
        return loss

    def _apply_prox_grad(self) -> None:
        # grad += mu*(w - w_round); per-param so it works with both flat
        # and scattered grad layouts (p.grad views the flat buffer in the
        # former). Pure tensor ops -> hipGraph-capturable; the _prox_ref
        # buffers are persistent and refreshed by copy_ at round start.
        with torch.no_grad():
            for p, r in zip(self._prox_params, self._prox_ref):
                if p.grad is not None:
                    p.grad.add_(p.detach() - r, alpha=self.cfg.fedprox_mu)

    # ------------------------------------------------------------ local fit
    def local_step(self) -> float:
        batch = next(self.batches)
        if getattr(self, "_graph", None) is not None:
            for k, v in batch.items():
                self._static_batch[k].copy_(v, non_blocking=True)
            self._join_agg()  # replay reads AND writes params/master
            self._graph.replay()
            return float(self._static_loss.detach())
        if self.dp_engine is not None:
            return self._dp_local_step(batch)
        self.flat.zero_grad()
        loss = self.loss_fn(self.model, batch)
        loss.backward()
        if self.cfg.fedprox_mu > 0:
            self._apply_prox_grad()
        self._join_agg()  # optimizer writes master/params
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
        self._join_agg()  # optimizer writes master/params
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
        if self._overlap_capable():
            self._aggregate_overlapped()
            return
        beta = self.cfg.server_momentum  # _srv_old snapshot: run_round start
        adam = self.cfg.server_adam
        if (beta > 0 or adam) and self._srv_old is None:
            beta, adam = 0.0, False  # aggregate() outside run_round
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
            if beta > 0:
                # FedAvgM per bucket: v = beta*v + (mean - old);
                # w = old + server_lr*v  (chunk aliases master)
                hi = off + chunk.numel()
                v = self._srv_v[off:hi]
                old = self._srv_old[off:hi]
                v.mul_(beta).add_(chunk).sub_(old)
                chunk.copy_(old).add_(v, alpha=self.cfg.server_lr)
            elif adam:
                hi = off + chunk.numel()
                m = self._srv_m[off:hi]
                v = self._srv_v[off:hi]
                old = self._srv_old[off:hi]
                b1, b2 = self.cfg.server_adam_betas
                d = chunk - old
                m.mul_(b1).add_(d, alpha=1 - b1)
                v.mul_(b2).addcmul_(d, d, value=1 - b2)
                chunk.copy_(old).addcdiv_(
                    m, v.sqrt().add_(self.cfg.server_adam_tau),
                    value=self.cfg.server_lr)
            self.flat.param_flat[off : off + chunk.numel()].copy_(
                chunk.to(self.flat.dtype)
            )

    def _aggregate_overlapped(self) -> None:
        """Launch the weighted FedAvg collective on the comm stream and
        return immediately. Per-bucket, on the comm stream:

            agg = weight * master  →  all-reduce(agg[b])  →
            master[b] = agg[b]  →  param[b] = bf16(agg[b])  →  event[b]

        Round k+1's forward waits per-module on exactly the buckets holding
        its params (hooks), and the optimizer step joins the rest — so the
        result is bit-identical to the synchronous path while the collective
        hides behind the next forward. The separate ``agg`` workspace keeps
        the collective's reads/writes off master until each bucket commits,
        which is what makes the early forward safe."""
        master = self.flat.master_flat
        self._collect_overlap_stats()
        if self._agg_flat is None:
            self._agg_flat = torch.empty_like(master)
            self._ev_agg_start = torch.cuda.Event(enable_timing=True)
            self._ev_agg_end = torch.cuda.Event(enable_timing=True)
        agg = self._agg_flat
        bucket = max(1, self.cfg.bucket_bytes // 4)
        use_rb = self.cfg.agg_algo == "reduce_bcast"
        self.comm_stream.wait_stream(torch.cuda.current_stream())
        events = []
        with torch.cuda.stream(self.comm_stream):
            self._ev_agg_start.record(self.comm_stream)
            torch.mul(master, self.weight, out=agg)
            for off in range(0, master.numel(), bucket):
                chunk = agg[off : off + bucket]
                if use_rb:
                    w = self.comm.reduce(chunk, dst=0, async_op=True)
                    w.wait()
                    w = self.comm.broadcast(chunk, src=0, async_op=True)
                else:
                    w = self.comm.all_reduce(chunk, async_op=True)
                w.wait()  # comm stream (not host) waits for this bucket
                hi = off + chunk.numel()
                master[off:hi].copy_(chunk)
                self.flat.param_flat[off:hi].copy_(chunk.to(self.flat.dtype))
                ev = torch.cuda.Event()
                ev.record(self.comm_stream)
                events.append((off, hi, ev))
            self._ev_agg_end.record(self.comm_stream)
        self._agg_events = events
        self._bucket_waited = [False] * len(events)
        self._agg_pending = True

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
        beta = self.cfg.server_momentum  # _srv_old snapshot: run_round start
        adam = self.cfg.server_adam
        if (beta > 0 or adam) and self._srv_old is None:
            beta, adam = 0.0, False  # aggregate() outside run_round
        master.copy_(q.to(torch.float64).mul_(1.0 / scale).to(torch.float32))
        if beta > 0:
            self._srv_v.mul_(beta).add_(master).sub_(self._srv_old)
            master.copy_(self._srv_old).add_(self._srv_v,
                                             alpha=self.cfg.server_lr)
        elif adam:
            b1, b2 = self.cfg.server_adam_betas
            d = master - self._srv_old
            self._srv_m.mul_(b1).add_(d, alpha=1 - b1)
            self._srv_v.mul_(b2).addcmul_(d, d, value=1 - b2)
            master.copy_(self._srv_old).addcdiv_(
                self._srv_m, self._srv_v.sqrt().add_(self.cfg.server_adam_tau),
                value=self.cfg.server_lr)
        self.flat.sync_param_from_master()

    # ---------------------------------------------------------------- rounds
    def run_round(self) -> dict:
        overlapping = self._overlap_capable()
        if self.cfg.fedprox_mu > 0:
            # w_round := the post-aggregation weights (persistent buffers,
            # refreshed in place so a captured graph keeps reading them)
            with torch.no_grad():
                for p, r in zip(self._prox_params, self._prox_ref):
                    r.copy_(p.detach())
        if self.cfg.server_momentum > 0 or self.cfg.server_adam:
            # the server optimizer's w_old is the ROUND-START global state
            # — by aggregate time master has drifted through local steps
            if self._srv_v is None:
                self._srv_v = torch.zeros_like(self.flat.master_flat)
                self._srv_old = torch.empty_like(self.flat.master_flat)
                if self.cfg.server_adam:
                    self._srv_m = torch.zeros_like(self.flat.master_flat)
            self._srv_old.copy_(self.flat.master_flat)
        t0 = time.perf_counter()
        losses = [self.local_step() for _ in range(self.cfg.local_steps)]
        if self.comm.device.type == "cuda":
            # compute stream only: the in-flight collective (overlap mode)
            # must keep running on the comm stream
            torch.cuda.current_stream().synchronize()
        t1 = time.perf_counter()
        self.aggregate()
        if self.comm.device.type == "cuda" and not overlapping:
            torch.cuda.synchronize()
        t2 = time.perf_counter()
        self.round_idx += 1
        self.last_stats = {
            "round": self.round_idx,
            "loss": sum(losses) / len(losses),
            "fit_ms": (t1 - t0) * 1e3,
            # overlap mode: launch cost only — the collective itself runs
            # under the next round's forward; see agg_comm_ms/overlap_pct
            "agg_ms": (t2 - t1) * 1e3,
            "round_ms": (t2 - t0) * 1e3,
        }
        if overlapping and self.overlap_stats:
            self.last_stats.update(self.overlap_stats)
        return self.last_stats
