"""Fused activations: SwiGLU (silu(gate) * up) with HIP fwd/bwd."""
from __future__ import annotations

import torch

from . import hip_ops


class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, g, u):
        ctx.save_for_backward(g, u)
        if g.is_cuda:
            return hip_ops().swiglu_fwd(g.contiguous(), u.contiguous())
        return torch.nn.functional.silu(g.float()).mul(u.float()).to(g.dtype)

    @staticmethod
    def backward(ctx, dout):
        g, u = ctx.saved_tensors
        if g.is_cuda:
            dg, du = hip_ops().swiglu_bwd(dout.contiguous(), g.contiguous(),
                                          u.contiguous())
            return dg, du
        gf, uf, df = g.float(), u.float(), dout.float()
        sig = torch.sigmoid(gf)
        s = gf * sig
        ds = sig * (1 + gf * (1 - sig))
        return (df * uf * ds).to(g.dtype), (df * s).to(u.dtype)


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    return _SwiGLUFn.apply(gate, up)


class _RoPEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, coss, sins):
        ctx.save_for_backward(coss, sins)
        if x.is_cuda:
            return hip_ops().rope_apply(x.contiguous(), coss, sins, 1)
        return _rope_ref(x, coss, sins, 1)

    @staticmethod
    def backward(ctx, dout):
        coss, sins = ctx.saved_tensors
        if dout.is_cuda:
            return hip_ops().rope_apply(dout.contiguous(), coss, sins, -1), None, None
        return _rope_ref(dout, coss, sins, -1), None, None


def _rope_ref(x, coss, sins, direction):
    half = x.shape[-1] // 2
    xf = x.float()
    x1, x2 = xf[..., :half], xf[..., half:]
    c = coss.reshape(*([1] * (x.dim() - 2)), coss.shape[0], half)
    s = sins.reshape(*([1] * (x.dim() - 2)), sins.shape[0], half) * direction
    return torch.cat([x1 * c - x2 * s, x1 * s + x2 * c], -1).to(x.dtype)


def rope_tables(seq: int, dim: int, base: float, device):
    """cos/sin lookup tables [S, D/2] fp32 (computed once per shape)."""
    half = dim // 2
    inv = 1.0 / base ** (torch.arange(half, device=device,
                                      dtype=torch.float32) / half)
    t = torch.arange(seq, device=device, dtype=torch.float32)
    freqs = torch.outer(t, inv)
    return freqs.cos().contiguous(), freqs.sin().contiguous()


def rope(x: torch.Tensor, coss: torch.Tensor, sins: torch.Tensor) -> torch.Tensor:
    return _RoPEFn.apply(x, coss, sins)
