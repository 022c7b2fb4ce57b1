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
    def __init__(self, model: torch.nn.Module, device=None, dtype=torch.bfloat16):
        self.model = model
        self.dtype = dtype
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
                p.grad = (self.grad_flat[offset : offset + n]
                          .view(N, H, W, C).permute(0, 3, 1, 2))
            else:
                self.param_flat[offset : offset + n].copy_(src.reshape(-1))
                # re-point the parameter at the flat slice (no copy from here on)
                p.data = self.param_flat[offset : offset + n].view(p.shape)
                p.grad = self.grad_flat[offset : offset + n].view(p.shape)
            self._specs.append((p, offset, n))
            offset += n
        self.master_flat = self.param_flat.to(torch.float32)
        self.momentum_flat = torch.zeros_like(self.master_flat)
        self.exp_avg_sq: torch.Tensor | None = None  # lazily for AdamW
        self._step = 0

    # ------------------------------------------------------------- optimizer
    def zero_grad(self) -> None:
        self.grad_flat.zero_()

    def sgd_step(self, lr: float, momentum: float = 0.9, weight_decay: float = 0.0,
                 nesterov: bool = False) -> None:
        fused_optim.fused_sgd(
            self.param_flat, self.master_flat, self.grad_flat, self.momentum_flat,
            lr=lr, momentum=momentum, weight_decay=weight_decay, nesterov=nesterov,
        )

    def adamw_step(self, lr: float, beta1=0.9, beta2=0.999, eps=1e-8,
                   weight_decay=0.01) -> None:
        if self.exp_avg_sq is None:
            self.exp_avg_sq = torch.zeros_like(self.master_flat)
        self._step += 1
        fused_optim.fused_adamw(
            self.param_flat, self.master_flat, self.grad_flat,
            self.momentum_flat, self.exp_avg_sq,
            lr=lr, beta1=beta1, beta2=beta2, eps=eps, weight_decay=weight_decay,
            step=self._step,
        )

    # ------------------------------------------------------------ aggregation
    def sync_param_from_master(self) -> None:
        self.param_flat.copy_(self.master_flat.to(self.dtype))

    def load_master(self, flat: torch.Tensor) -> None:
        self.master_flat.copy_(flat)
        self.sync_param_from_master()

    def state_dict(self) -> dict:
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
        self.sync_param_from_master()
