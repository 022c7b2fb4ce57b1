"""Flat parameter space: bf16 compute copy + fp32 master + flat grads.

MI355X-first memory layout (288 GB HBM per GPU: keep everything resident,
prefer few big buffers over many small tensors):

- every model parameter's storage is a slice of ONE flat bf16 buffer,
- autograd accumulates into views of ONE flat bf16 grad buffer,
- fp32 master weights + optimizer state are flat fp32 buffers,
- FedAvg aggregation all-reduces the fp32 master buffer (bucketed),
- the fused optimizer kernel updates the whole model in one launch.
"""
from __future__ import annotations

import torch

from ..ops import optim as fused_optim


class FlatParamSpace:
    def __init__(self, model: torch.nn.Module, device=None, dtype=torch.bfloat16,
                 scatter_grads: bool = False):
        """``scatter_grads=True`` (GPU fast path): gradients are NOT
        pre-assigned flat views — autograd "steals" each backward output
        (no grad-buffer zeroing, no AccumulateGrad add pass, ~20% of a
        ResNet round) and the multi-tensor SGD kernel walks a chunk table
        of {grad ptr, flat offset, n}. DP keeps the flat-grad mode (it
        norms/noises the flat buffer)."""
        self.model = model
        self.dtype = dtype
        self.scatter_grads = scatter_grads
        params = [p for p in model.parameters() if p.requires_grad]
        if not params:
            raise ValueError("model has no trainable parameters")
        self.device = device or params[0].device
        self.numel = sum(p.numel() for p in params)
        self.param_flat = torch.empty(self.numel, dtype=dtype, device=self.device)
        self.grad_flat = torch.zeros(self.numel, dtype=dtype, device=self.device)
        offset = 0
        self._specs: list[tuple[torch.nn.Parameter, int, int]] = []
        for p in params:
            n = p.numel()
            # preserve channels_last (NHWC) layout for conv weights: the flat
            # slice stores the tensor in its PHYSICAL order, the view+permute
            # reconstructs the same strides backed by the flat buffer
            nhwc = p.dim() == 4 and p.is_contiguous(memory_format=torch.channels_last) \
                and not p.is_contiguous()
            src = p.detach().to(device=self.device, dtype=dtype)
            if nhwc:
                phys = src.permute(0, 2, 3, 1).reshape(-1)  # physical NHWC order
                self.param_flat[offset : offset + n].copy_(phys)
                N, C, H, W = p.shape
                p.data = (self.param_flat[offset : offset + n]
                          .view(N, H, W, C).permute(0, 3, 1, 2))
                if not self.scatter_grads:
                    p.grad = (self.grad_flat[offset : offset + n]
                              .view(N, H, W, C).permute(0, 3, 1, 2))
            else:
                self.param_flat[offset : offset + n].copy_(src.reshape(-1))
                # re-point the parameter at the flat slice (no copy from here on)
                p.data = self.param_flat[offset : offset + n].view(p.shape)
                if not self.scatter_grads:
                    p.grad = self.grad_flat[offset : offset + n].view(p.shape)
            self._specs.append((p, offset, n, nhwc))
            offset += n
        self._sgd_table: torch.Tensor | None = None  # persistent device table
        self._table_chunk = 16384
        self.master_flat = self.param_flat.to(torch.float32)
        self.momentum_flat = torch.zeros_like(self.master_flat)
        self.exp_avg_sq: torch.Tensor | None = None  # lazily for AdamW
        self._step = 0
        # int32 device step counter: the AdamW kernels pre-increment it and
        # derive bias correction from it, so the optimizer step is
        # hipGraph-capturable (a host-side counter freezes at capture time)
        self._step_dev: torch.Tensor | None = None

    # ------------------------------------------------------------- optimizer
    def zero_grad(self) -> None:
        if self.scatter_grads:
            for p, *_ in self._specs:
                p.grad = None  # autograd steals the next backward output
        else:
            self.grad_flat.zero_()

    # -- scattered-grad fast path -----------------------------------------
    def _grad_phys(self, p, nhwc, strict: bool = False):
        g = p.grad
        if g is None:
            raise RuntimeError("scatter_grads: parameter has no grad")
        fmt = torch.channels_last if nhwc else torch.contiguous_format
        if not g.is_contiguous(memory_format=fmt):
            if strict:
                # post-hipGraph-capture: a layout conversion here would make
                # the chunk table point at a copy the replayed kernels never
                # write — the caller must fall back to eager instead
                raise RuntimeError(
                    f"scatter_grads: grad layout mismatch for shape "
                    f"{tuple(p.shape)} under graph capture"
                )
            g = g.contiguous(memory_format=fmt)
            p.grad = g
        return g

    def refresh_sgd_table(self, strict: bool = False) -> torch.Tensor:
        """(Re)build the {grad ptr, flat offset, n} chunk table from the
        current p.grad tensors. Called per eager step, and once after a
        hipGraph capture (replays reuse the same pool addresses)."""
        entries = []
        ch = self._table_chunk
        for p, offset, n, nhwc in self._specs:
            g = self._grad_phys(p, nhwc, strict=strict)
            ptr = g.data_ptr()
            for o in range(0, n, ch):
                entries.append((ptr + o * 2, offset + o, min(ch, n - o)))
        cpu = torch.tensor(entries, dtype=torch.int64)
        if self._sgd_table is None or self._sgd_table.shape != cpu.shape:
            self._sgd_table = torch.empty_like(cpu, device=self.device)
        self._sgd_table.copy_(cpu, non_blocking=True)
        return self._sgd_table

    def sgd_step(self, lr: float, momentum: float = 0.9, weight_decay: float = 0.0,
                 nesterov: bool = False) -> None:
        if self.scatter_grads:
            if self.param_flat.is_cuda:
                from ..ops import hip_ops

                capturing = torch.cuda.is_current_stream_capturing()
                if not capturing:
                    self.refresh_sgd_table()
                # during capture the table tensor must already exist (warmup
                # built it); the kernel reads its CONTENTS at replay time
                hip_ops().fused_sgd_mt(
                    self._sgd_table, self.param_flat, self.master_flat,
                    self.momentum_flat, lr, momentum, weight_decay, nesterov,
                )
            else:
                # CPU reference: per-param update into the flat buffers
                for p, offset, n, nhwc in self._specs:
                    g = self._grad_phys(p, nhwc).reshape(-1).float()
                    m = self.master_flat[offset : offset + n]
                    v = self.momentum_flat[offset : offset + n]
                    if weight_decay:
                        g = g.add(m, alpha=weight_decay)
                    v.mul_(momentum).add_(g)
                    step = g.add(v, alpha=momentum) if nesterov else v
                    m.add_(step, alpha=-lr)
                    self.param_flat[offset : offset + n].copy_(m.to(self.dtype))
            return
        fused_optim.fused_sgd(
            self.param_flat, self.master_flat, self.grad_flat, self.momentum_flat,
            lr=lr, momentum=momentum, weight_decay=weight_decay, nesterov=nesterov,
        )

    def adamw_step(self, lr: float, beta1=0.9, beta2=0.999, eps=1e-8,
                   weight_decay=0.01) -> None:
        if self.exp_avg_sq is None:
            self.exp_avg_sq = torch.zeros_like(self.master_flat)
        if self.param_flat.is_cuda and self._step_dev is None:
            self._step_dev = torch.full((1,), self._step, dtype=torch.int32,
                                        device=self.device)
        self._step += 1
        if self.scatter_grads:
            if self.param_flat.is_cuda:
                from ..ops import hip_ops

                if not torch.cuda.is_current_stream_capturing():
                    # during capture the table tensor must already exist
                    # (warmup built it); refreshing here would record an H2D
                    # copy from a python temp that is freed before replay
                    self.refresh_sgd_table()
                hip_ops().fused_adamw_mt(
                    self._sgd_table, self.param_flat, self.master_flat,
                    self.momentum_flat, self.exp_avg_sq, lr, beta1, beta2,
                    eps, weight_decay, self._step, self._step_dev,
                )
            else:
                bc1 = 1 - beta1 ** self._step
                bc2 = 1 - beta2 ** self._step
                for p, offset, n, nhwc in self._specs:
                    g = self._grad_phys(p, nhwc).reshape(-1).float()
                    m = self.master_flat[offset : offset + n]
                    a = self.momentum_flat[offset : offset + n]
                    v = self.exp_avg_sq[offset : offset + n]
                    m.mul_(1.0 - lr * weight_decay)
                    a.mul_(beta1).add_(g, alpha=1 - beta1)
                    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                    denom = (v / bc2).sqrt().add_(eps)
                    m.addcdiv_(a, denom, value=-lr / bc1)
                    self.param_flat[offset : offset + n].copy_(m.to(self.dtype))
            return
        fused_optim.fused_adamw(
            self.param_flat, self.master_flat, self.grad_flat,
            self.momentum_flat, self.exp_avg_sq,
            lr=lr, beta1=beta1, beta2=beta2, eps=eps, weight_decay=weight_decay,
            step=self._step, step_dev=self._step_dev,
        )

    # ------------------------------------------------------------ aggregation
    def sync_param_from_master(self) -> None:
        self.param_flat.copy_(self.master_flat.to(self.dtype))

    def load_master(self, flat: torch.Tensor) -> None:
        self.master_flat.copy_(flat)
        self.sync_param_from_master()

    def state_dict(self) -> dict:
        # under hipGraph replay the python counter goes stale — the device
        # counter is authoritative
        if self._step_dev is not None:
            self._step = int(self._step_dev.item())
        return {
            "master": self.master_flat,
            "momentum": self.momentum_flat,
            "exp_avg_sq": self.exp_avg_sq,
            "step": self._step,
        }

    def load_state_dict(self, sd: dict) -> None:
        self.master_flat.copy_(sd["master"])
        self.momentum_flat.copy_(sd["momentum"])
        if sd.get("exp_avg_sq") is not None:
            if self.exp_avg_sq is None:
                self.exp_avg_sq = torch.zeros_like(self.master_flat)
            self.exp_avg_sq.copy_(sd["exp_avg_sq"])
        self._step = sd.get("step", 0)
        if self._step_dev is not None:
            self._step_dev.fill_(self._step)
        self.sync_param_from_master()
