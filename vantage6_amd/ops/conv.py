"""Direct NHWC conv on the hand-written implicit-GEMM kernel (csrc/conv.hip).

Forward runs the CDNA4 kernel for channels-last bf16 convs with
Cin/Cout % 64 == 0 (every ResNet-18 conv except the 3-channel stem);
backward delegates to torch's convolution_backward (MIOpen igemm) —
replacing dgrad/wgrad with tr_b16-read kernels is a follow-up.

Reference: the upstream framework ships no conv code (compute lives in
user algorithm images); this is net-new MI355X work.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from . import hip_ops


class _ConvFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, stride, pad, bn_acc=None, stats_sink=None):
        if bn_acc is not None:
            y, done = hip_ops().conv_nhwc_fwd_stats(x, w, stride, pad, bn_acc)
            if stats_sink is not None and bool(done.item() if done.is_cpu
                                               else done.cpu().item()):
                stats_sink._pending_stats = True
        else:
            y = hip_ops().conv_nhwc_fwd(x, w, stride, pad)
        ctx.save_for_backward(x, w)
        ctx.stride, ctx.pad = stride, pad
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        s, p = ctx.stride, ctx.pad
        k = w.shape[2]
        cin = w.shape[1]
        # the hand-written wgrad is numerically verified but measured
        # slower than the benchmark-tuned igemm_wrw on every ResNet shape
        # in-graph (49-85 us vs 24-30 us) — per-shape fallback, overridable
        import os

        own_wgrad = (os.environ.get("V6AMD_CONV_WGRAD", "0") == "1"
                     and cin >= 64 and (cin & (cin - 1)) == 0)
        if own_wgrad:
            dw = hip_ops().conv_nhwc_wgrad(dy, x, k, k, s, p)
        if s == 1 and p == (k - 1) // 2 and w.shape[0] % 64 == 0:
            # stride-1 dgrad IS a forward conv of dy with the spatially
            # flipped, channel-transposed weights — reuse the hand-written
            # kernel instead of MIOpen igemm_bwd
            w_flip = hip_ops().conv_flip_weights(w)
            dx = hip_ops().conv_nhwc_fwd(dy, w_flip, 1, p)
            if not own_wgrad:
                _, dw, _ = torch.ops.aten.convolution_backward(
                    dy, x, w, None, [s, s], [p, p], [1, 1], False, [0, 0], 1,
                    [False, True, False])
            return dx, dw, None, None, None, None
        dx, dw_lib, _ = torch.ops.aten.convolution_backward(
            dy, x, w, None, [s, s], [p, p], [1, 1], False, [0, 0], 1,
            [True, not own_wgrad, False])
        return dx, (dw if own_wgrad else dw_lib), None, None, None, None


class Conv2d(nn.Conv2d):
    """Drop-in Conv2d: hand-written NHWC kernel on the eligible path,
    library conv otherwise (3-channel stem, CPU, non-NHWC).

    When ``stats_bn`` is set to the downstream :class:`FusedBatchNorm2d`,
    the conv epilogue accumulates that BN's batch sum/sumsq directly and
    the BN forward skips its whole partials read pass."""

    stats_bn = None  # wired by the model (e.g. ResNet BasicBlock)

    def forward(self, x):
        w = self.weight
        if (x.is_cuda and x.dtype == torch.bfloat16
                and self.in_channels % 64 == 0 and self.out_channels % 64 == 0
                and self.groups == 1 and self.dilation == (1, 1)
                and self.bias is None
                and self.stride[0] == self.stride[1]
                and self.padding[0] == self.padding[1]
                and x.is_contiguous(memory_format=torch.channels_last)
                and w.is_contiguous(memory_format=torch.channels_last)):
            bn = self.stats_bn
            acc = None
            if bn is not None and bn.training and self.training:
                acc = bn._ensure_fwd_acc(self.out_channels, x.device)
            return _ConvFn.apply(x, w, self.stride[0], self.padding[0],
                                 acc, bn)
        return super().forward(x)


class _MaxPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        y, idx = hip_ops().maxpool_nhwc_fwd(x)
        ctx.save_for_backward(idx)
        ctx.hw = (x.shape[2], x.shape[3])
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        h, w = ctx.hw
        return hip_ops().maxpool_nhwc_bwd(dy, idx, h, w)


class MaxPool2d(nn.Module):
    """3×3 stride-2 pad-1 max pool: hand-written NHWC kernels with a
    uint8 argmax (gather-based backward, no atomics); library fallback
    off the eligible path."""

    def forward(self, x):
        if (x.is_cuda and x.dtype == torch.bfloat16 and x.shape[1] % 8 == 0
                and x.is_contiguous(memory_format=torch.channels_last)):
            return _MaxPoolFn.apply(x)
        return torch.nn.functional.max_pool2d(x, 3, stride=2, padding=1)
