"""bf16 Linear layer on the hand-written MFMA GEMM (csrc/gemm.hip).

The three products of a Linear layer map onto one stride-general
MFMA kernel (fp32 accumulate, bf16 I/O):

- forward:  Y[M,N] = X[M,K] · W[N,K]^T      (both operands K-contiguous)
- dgrad:    dX[M,K] = dY[M,N] · W[N,K]      (A reduction-contiguous)
- wgrad:    dW[N,K] = dY[M,N]^T · X[M,K]    (A reduction-strided)

CPU fallback is plain fp32 torch matmul (numerics reference).
"""
from __future__ import annotations

import torch

from . import hip_ops


def matmul_bf16(a: torch.Tensor, b: torch.Tensor, trans_a: bool, trans_b: bool
                ) -> torch.Tensor:
    """C = op(A) · op(B), bf16 in/out, fp32 accumulate."""
    if a.is_cuda:
        return hip_ops().gemm_bf16(a.contiguous(), b.contiguous(), trans_a, trans_b)
    af = (a.float().t() if trans_a else a.float())
    bf = (b.float().t() if trans_b else b.float())
    return (af @ bf).to(a.dtype)  # bf16 for bf16 inputs (mirrors the kernel)


class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        shape = x.shape
        x2 = x.reshape(-1, shape[-1])
        y = matmul_bf16(x2, weight, False, True)
        if bias is not None:
            y = y + bias
        ctx.save_for_backward(x2, weight)
        ctx.has_bias = bias is not None
        ctx.in_shape = shape
        return y.reshape(*shape[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, dy):
        x2, weight = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx = matmul_bf16(dy2, weight, False, False).reshape(ctx.in_shape)
        dw = matmul_bf16(dy2, x2, True, False)
        if ctx.has_bias:
            if dy2.is_cuda and dy2.shape[-1] % 8 == 0:
                # hand-written colsum: coalesced bf16 reads, and (unlike
                # torch's two-pass reduce) hipGraph-replay-safe — its
                # workspace is zeroed by a kernel every call
                db = hip_ops().colsum_bf16(dy2)
            else:
                db = dy2.sum(0)
        else:
            db = None
        return dx, dw, db


def linear(x, weight, bias=None):
    return _LinearFn.apply(x, weight, bias)


class Linear(torch.nn.Module):
    """Drop-in bf16 Linear running on the hand-written MFMA GEMM."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True,
                 dtype=torch.bfloat16):
        super().__init__()
        self.in_features, self.out_features = in_features, out_features
        w = torch.empty(out_features, in_features, dtype=dtype)
        torch.nn.init.kaiming_uniform_(w, a=5**0.5)
        self.weight = torch.nn.Parameter(w)
        if bias:
            bound = 1.0 / in_features**0.5
            self.bias = torch.nn.Parameter(
                torch.empty(out_features, dtype=dtype).uniform_(-bound, bound)
            )
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        return linear(x, self.weight, self.bias)
