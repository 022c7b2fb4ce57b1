"""Graph-safe embedding.

torch's ``embedding_dense_backward`` on this ROCm build faults with an
HSA memory-aperture violation on the SECOND replay of a captured hipGraph
(bisected in round 1: forward replays clean, detached embeddings clean,
training tok embeddings faults — see NOTES.md). Its cub sort-based
accumulation relies on temp storage that is only valid for one replay.

This module keeps the forward as the plain gather (``F.embedding``) and
replaces the backward with hand-written kernels: a fp32 scatter workspace
zeroed BY A KERNEL, vectorized bf16 atomic accumulation, and a bf16
convert — every byte of state rewritten on each replay.

Reference parity: the reference trains its torch models with stock
``nn.Embedding``; this class subclasses it so ``isinstance`` checks
(e.g. the DP-SGD ghost-clipping hooks) keep working.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import hip_ops


class _EmbFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, ids, weight):
        ctx.save_for_backward(ids)
        ctx.vocab, ctx.hidden = weight.shape
        ctx.wdtype = weight.dtype
        return F.embedding(ids, weight)

    @staticmethod
    def backward(ctx, dy):
        (ids,) = ctx.saved_tensors
        ids_flat = ids.reshape(-1)
        dy2 = dy.reshape(-1, ctx.hidden).contiguous()
        if dy.is_cuda and dy.dtype == torch.bfloat16:
            if not ids_flat.is_contiguous():  # e.g. expanded position ids
                ids_flat = ids_flat.contiguous()
            dw = hip_ops().emb_grad_bf16(ids_flat, dy2, ctx.vocab)
        else:
            ws = torch.zeros(ctx.vocab, ctx.hidden, dtype=torch.float32,
                             device=dy.device)
            ws.index_add_(0, ids_flat, dy2.float())
            dw = ws.to(ctx.wdtype)
        return None, dw


class Embedding(nn.Embedding):
    """``nn.Embedding`` with a hipGraph-replay-safe backward.

    ``padding_idx``/``max_norm`` options are not supported (unused by the
    model zoo here).
    """

    def forward(self, ids):  # noqa: D102
        return _EmbFn.apply(ids, self.weight)
