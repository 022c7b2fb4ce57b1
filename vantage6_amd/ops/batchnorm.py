"""Fused BatchNorm2d (+ReLU, +residual) on the hand-written NHWC kernels.

Replaces per-block {BN, add, relu} eager chains (5 MIOpen kernels + casts)
with 3 fused launches forward / 3 backward (csrc/bn.hip). GPU path operates
on channels_last tensors viewed as [R, C]; CPU path is the fp32 torch
reference the kernels are tested against.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

from . import hip_ops


class _FusedBNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, training,
                momentum, eps, relu, residual, fwd_acc=None, bwd_acc=None,
                stats_precomputed=False):
        ctx.fwd_acc, ctx.bwd_acc = fwd_acc, bwd_acc
        if x.is_cuda:
            xc = x.contiguous(memory_format=torch.channels_last)
            N, C, H, W = xc.shape
            x2 = xc.permute(0, 2, 3, 1).reshape(-1, C)  # zero-copy view
            res2 = None
            if residual is not None:
                res2 = (residual.contiguous(memory_format=torch.channels_last)
                        .permute(0, 2, 3, 1).reshape(-1, C))
            y2, mean, rstd, msk = hip_ops().bn_fwd(
                x2, weight, bias, running_mean, running_var, training,
                momentum, eps, relu, res2, ctx.fwd_acc, stats_precomputed,
            )
            y = y2.view(N, H, W, C).permute(0, 3, 1, 2)
            ctx.save_for_backward(x2, msk, weight, mean, rstd)
            ctx.meta = (relu, residual is not None, (N, C, H, W), True)
            return y
        # ---- CPU reference (fp32 math)
        y = F.batch_norm(x.float(), running_mean, running_var, weight.float(),
                         bias.float(), training, momentum, eps)
        if residual is not None:
            y = y + residual.float()
        if relu:
            y = torch.relu(y)
        y = y.to(x.dtype)
        # recompute batch stats for backward
        if training:
            xf = x.float()
            mean = xf.mean(dim=(0, 2, 3))
            var = xf.var(dim=(0, 2, 3), unbiased=False)
        else:
            mean, var = running_mean.clone(), running_var.clone()
        rstd = (var + eps).rsqrt()
        ctx.save_for_backward(x, y, weight, mean, rstd)
        ctx.meta = (relu, residual is not None, tuple(x.shape), False)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y_or_mask, weight, mean, rstd = ctx.saved_tensors
        relu, has_res, shape, on_gpu = ctx.meta
        if on_gpu:
            N, C, H, W = shape
            dy2 = (dy.contiguous(memory_format=torch.channels_last)
                   .permute(0, 2, 3, 1).reshape(-1, C))
            outs = hip_ops().bn_bwd(dy2, y_or_mask, x, weight, mean, rstd, relu,
                                    has_res, ctx.bwd_acc)
            dx2, dw, db = outs[0], outs[1], outs[2]
            dx = dx2.view(N, H, W, C).permute(0, 3, 1, 2)
            dres = None
            if has_res:
                dres = outs[3].view(N, H, W, C).permute(0, 3, 1, 2)
            return (dx, dw, db, None, None, None, None, None, None, dres,
                    None, None, None)
        # ---- CPU reference (y_or_mask holds y on CPU)
        y = y_or_mask
        dyf = dy.float()
        if relu:
            dyf = dyf * (y.float() > 0)
        dres = dyf.to(dy.dtype) if has_res else None
        xf = x.float()
        xhat = (xf - mean[None, :, None, None]) * rstd[None, :, None, None]
        R = xf.numel() / xf.shape[1]
        db_ = dyf.sum(dim=(0, 2, 3))
        dw_ = (dyf * xhat).sum(dim=(0, 2, 3))
        dxf = (weight.float() * rstd)[None, :, None, None] * (
            dyf - db_[None, :, None, None] / R - xhat * dw_[None, :, None, None] / R
        )
        return (dxf.to(x.dtype), dw_.to(weight.dtype), db_.to(weight.dtype),
                None, None, None, None, None, None, dres, None, None, None)


class FusedBatchNorm2d(torch.nn.Module):
    def __init__(self, num_features: int, eps: float = 1e-5, momentum: float = 0.1):
        super().__init__()
        self.weight = torch.nn.Parameter(
            torch.ones(num_features, dtype=torch.bfloat16))
        self.bias = torch.nn.Parameter(
            torch.zeros(num_features, dtype=torch.bfloat16))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.eps, self.momentum = eps, momentum
        # persistent atomic-stat accumulators (zeroed once; the finalize
        # kernels re-zero after reading — saves a launch per call)
        self._fwd_acc = None
        self._bwd_acc = None

    # set True by a producing Conv2d whose epilogue already accumulated
    # this BN's batch sum/sumsq into _fwd_acc (ops/conv.py)
    _pending_stats = False

    def _ensure_fwd_acc(self, c, device):
        if self._fwd_acc is None:
            self._fwd_acc = torch.zeros(16, 2, c, device=device)
            self._bwd_acc = torch.zeros(16, 2, c, device=device)
        return self._fwd_acc

    def forward(self, x, residual=None, relu: bool = False):
        if x.is_cuda and self._fwd_acc is None:
            c = self.weight.numel()
            self._fwd_acc = torch.zeros(16, 2, c, device=x.device)
            self._bwd_acc = torch.zeros(16, 2, c, device=x.device)
        pre = self._pending_stats
        self._pending_stats = False
        return _FusedBNFn.apply(
            x, self.weight, self.bias, self.running_mean, self.running_var,
            self.training, self.momentum, self.eps, relu, residual,
            self._fwd_acc, self._bwd_acc, pre,
        )
