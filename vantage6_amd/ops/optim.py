"""Fused optimizers on flat parameter buffers.

The local-fit hot path keeps three flat buffers per station (sized for
288 GB HBM — everything stays resident):

- ``param``    flat bf16 — the compute copy the model's tensors view into
- ``master``   flat fp32 — master weights (update precision)
- ``grad``     flat bf16 — autograd accumulates into views of this
- ``momentum`` flat fp32 (+ ``exp_avg_sq`` for AdamW)

One kernel launch updates the whole model (vs per-tensor eager ops):
HBM-bound elementwise, fused read-modify-write of all buffers in a single
pass. GPU path requires the hand-written HIP kernel (`csrc/optim.hip`);
the CPU path below is the numerics reference the GPU kernel is tested
against.
"""
from __future__ import annotations

import torch

from . import hip_ops


def fused_sgd(
    param_bf16: torch.Tensor,
    master_f32: torch.Tensor,
    grad_bf16: torch.Tensor,
    momentum_f32: torch.Tensor,
    *,
    lr: float,
    momentum: float = 0.9,
    weight_decay: float = 0.0,
    nesterov: bool = False,
) -> None:
    """master += SGD-momentum update; param_bf16 = bf16(master). In-place."""
    if param_bf16.is_cuda:
        hip_ops().fused_sgd(
            param_bf16, master_f32, grad_bf16, momentum_f32,
            lr, momentum, weight_decay, nesterov,
        )
        return
    g = grad_bf16.float()
    if weight_decay:
        g = g.add(master_f32, alpha=weight_decay)
    momentum_f32.mul_(momentum).add_(g)
    step = g.add(momentum_f32, alpha=momentum) if nesterov else momentum_f32
    master_f32.add_(step, alpha=-lr)
    param_bf16.copy_(master_f32.to(torch.bfloat16))


def fused_adamw(
    param_bf16: torch.Tensor,
    master_f32: torch.Tensor,
    grad_bf16: torch.Tensor,
    exp_avg: torch.Tensor,
    exp_avg_sq: torch.Tensor,
    *,
    lr: float,
    beta1: float = 0.9,
    beta2: float = 0.999,
    eps: float = 1e-8,
    weight_decay: float = 0.01,
    step: int = 1,
    step_dev: torch.Tensor | None = None,
) -> None:
    """AdamW (decoupled weight decay), fp32 state, bf16 compute copy.

    ``step_dev`` (int32 device scalar) makes bias correction device-side so
    the step is hipGraph-capturable; the kernel pre-increments it each call.
    """
    if param_bf16.is_cuda:
        hip_ops().fused_adamw(
            param_bf16, master_f32, grad_bf16, exp_avg, exp_avg_sq,
            lr, beta1, beta2, eps, weight_decay, step, step_dev,
        )
        return
    g = grad_bf16.float()
    master_f32.mul_(1.0 - lr * weight_decay)
    exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1**step
    bc2 = 1 - beta2**step
    denom = (exp_avg_sq / bc2).sqrt_().add_(eps)
    master_f32.addcdiv_(exp_avg, denom, value=-lr / bc1)
    param_bf16.copy_(master_f32.to(torch.bfloat16))
