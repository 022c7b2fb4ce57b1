"""LayerNorm / RMSNorm with hand-written HIP forward+backward.

One workgroup per row, fp32 accumulation, bf16 I/O; forward saves
mean/rstd for backward (csrc/norms.hip). CPU reference below defines the
numerics contract the HIP kernels are tested against.
"""
from __future__ import annotations

import torch

from . import hip_ops


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        if x.is_cuda:
            y, mean, rstd = hip_ops().layernorm_fwd(x, weight, bias, eps)
        else:
            xf = x.float()
            mean = xf.mean(-1)
            var = xf.var(-1, unbiased=False)
            rstd = (var + eps).rsqrt()
            y = ((xf - mean[..., None]) * rstd[..., None]) * weight.float() + bias.float()
            y = y.to(x.dtype)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        if x.is_cuda:
            dx, dw, db = hip_ops().layernorm_bwd(dy.contiguous(), x, weight, mean, rstd)
        else:
            xf, dyf, wf = x.float(), dy.float(), weight.float()
            xhat = (xf - mean[..., None]) * rstd[..., None]
            dw = (dyf * xhat).reshape(-1, x.shape[-1]).sum(0)
            db = dyf.reshape(-1, x.shape[-1]).sum(0)
            wdy = dyf * wf
            c1 = wdy.mean(-1, keepdim=True)
            c2 = (wdy * xhat).mean(-1, keepdim=True)
            dx = ((wdy - c1 - xhat * c2) * rstd[..., None]).to(x.dtype)
            dw, db = dw.to(weight.dtype), db.to(weight.dtype)
        return dx, dw, db, None


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        if x.is_cuda:
            y, rstd = hip_ops().rmsnorm_fwd(x, weight, eps)
        else:
            xf = x.float()
            rstd = (xf.pow(2).mean(-1) + eps).rsqrt()
            y = (xf * rstd[..., None] * weight.float()).to(x.dtype)
        ctx.save_for_backward(x, weight, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, rstd = ctx.saved_tensors
        if x.is_cuda:
            dx, dw = hip_ops().rmsnorm_bwd(dy.contiguous(), x, weight, rstd)
        else:
            xf, dyf, wf = x.float(), dy.float(), weight.float()
            xhat = xf * rstd[..., None]
            dw = (dyf * xhat).reshape(-1, x.shape[-1]).sum(0).to(weight.dtype)
            wdy = dyf * wf
            c = (wdy * xhat).mean(-1, keepdim=True)
            dx = ((wdy - xhat * c) * rstd[..., None]).to(x.dtype)
        return dx, dw, None


def layer_norm(x, weight, bias, eps: float = 1e-5):
    return _LayerNormFn.apply(x, weight, bias, eps)


def rms_norm(x, weight, eps: float = 1e-6):
    return _RMSNormFn.apply(x, weight, eps)


class LayerNorm(torch.nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5, dtype=torch.bfloat16):
        super().__init__()
        self.weight = torch.nn.Parameter(torch.ones(dim, dtype=dtype))
        self.bias = torch.nn.Parameter(torch.zeros(dim, dtype=dtype))
        self.eps = eps

    def forward(self, x):
        return layer_norm(x, self.weight, self.bias, self.eps)


class RMSNorm(torch.nn.Module):
    def __init__(self, dim: int, eps: float = 1e-6, dtype=torch.bfloat16):
        super().__init__()
        self.weight = torch.nn.Parameter(torch.ones(dim, dtype=dtype))
        self.eps = eps

    def forward(self, x):
        return rms_norm(x, self.weight, self.eps)
