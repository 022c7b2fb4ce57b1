"""Fused bf16 cross-entropy for LM heads (csrc/norms.hip ce_* kernels).

torch's path materialises fp32 logits (``logits.float()``) and runs a
two-kernel softmax fwd/bwd over the 128k vocab — on Llama-1B that chain
(cunn_SoftMax* + bfloat16_copy casts) is ~5 ms/round, ~2x the bf16 bytes
actually needed. The fused op reads bf16 logits directly: forward emits
per-row loss + logsumexp, backward emits bf16 ``softmax - onehot`` in one
pass. Returning the per-row loss vector from the autograd Function serves
both mean reduction and DP-SGD's per-sample path with exact torch
semantics (ignored rows contribute 0 loss and exact-zero grads).

The upstream framework has no loss kernels (compute lives in user images,
SURVEY.md §3.9); this backs models/factory.py's LM losses.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

from . import hip_ops


class _CEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets, ignore_index):
        loss, lse = hip_ops().ce_fwd(logits, targets, ignore_index)
        ctx.save_for_backward(logits, targets, lse)
        ctx.ignore_index = ignore_index
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, targets, lse = ctx.saved_tensors
        # dloss arrives as a stride-0 expand when the caller reduced with
        # sum/mean; the kernel wants a real [R] fp32 vector
        dx = hip_ops().ce_bwd(logits, targets, lse,
                              dloss.to(torch.float32).contiguous(),
                              ctx.ignore_index)
        return dx, None, None


def cross_entropy(logits: torch.Tensor, targets: torch.Tensor,
                  reduction: str = "mean", ignore_index: int = -100):
    """Drop-in for ``F.cross_entropy(logits.float(), targets, ...)``.

    HIP path for cuda bf16 logits with V % 8 == 0 and V >= 4096 (LM
    heads); everything else falls back to torch in fp32. ``reduction``
    supports "mean" and "none" (per-row)."""
    V = logits.shape[-1]
    l2 = logits.reshape(-1, V)
    t = targets.reshape(-1)
    if (logits.is_cuda and logits.dtype == torch.bfloat16 and V % 8 == 0
            and V >= 4096 and l2.is_contiguous()):
        loss = _CEFn.apply(l2, t, ignore_index)
        if reduction == "none":
            return loss
        valid = (t != ignore_index).sum().clamp_min(1).to(loss.dtype)
        return loss.sum() / valid
    return F.cross_entropy(l2.float(), t, reduction=reduction,
                           ignore_index=ignore_index)
