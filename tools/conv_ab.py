#!/usr/bin/env python3
"""Direct NHWC conv fwd vs MIOpen (F.conv2d) — numerics + interleaved A/B
on the ResNet-18 shapes (batch 32, channels_last bf16).
"""
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch
import torch.nn.functional as F


def main():
    from vantage6_amd.ops import hip_ops

    ops = hip_ops()
    dev = "cuda:0"
    torch.manual_seed(0)
    torch.backends.cudnn.benchmark = True
    B = 32
    shapes = [  # (Cin, H, W, Cout, k, stride, pad, name)
        (64, 56, 56, 64, 3, 1, 1, "l1_3x3"),
        (64, 56, 56, 128, 3, 2, 1, "l2_down3x3"),
        (128, 28, 28, 128, 3, 1, 1, "l2_3x3"),
        (128, 28, 28, 256, 3, 2, 1, "l3_down3x3"),
        (256, 14, 14, 256, 3, 1, 1, "l3_3x3"),
        (256, 14, 14, 512, 3, 2, 1, "l4_down3x3"),
        (512, 7, 7, 512, 3, 1, 1, "l4_3x3"),
        (64, 56, 56, 128, 1, 2, 0, "l2_down1x1"),
        (128, 28, 28, 256, 1, 2, 0, "l3_down1x1"),
        (256, 14, 14, 512, 1, 2, 0, "l4_down1x1"),
    ]
    iters = 30
    for cin, h, w, cout, k, s, p, name in shapes:
        x = torch.randn(B, cin, h, w, device=dev, dtype=torch.bfloat16).to(
            memory_format=torch.channels_last)
        wt = (torch.randn(cout, cin, k, k, device=dev, dtype=torch.bfloat16)
              * (2.0 / (cin * k * k)) ** 0.5).to(memory_format=torch.channels_last)
        # numerics vs fp32 torch
        ref = F.conv2d(x.float(), wt.float(), stride=s, padding=p)
        got = ops.conv_nhwc_fwd(x, wt, s, p).float()
        err = (got - ref).abs().max().item()
        scale = ref.abs().max().item()
        ok = err <= 0.03 * scale + 0.05
        # timing interleaved
        for _ in range(5):
            ops.conv_nhwc_fwd(x, wt, s, p)
            F.conv2d(x, wt, stride=s, padding=p)
        torch.cuda.synchronize()
        t_ours = t_lib = 0.0
        for _ in range(iters):
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            ops.conv_nhwc_fwd(x, wt, s, p)
            torch.cuda.synchronize()
            t_ours += time.perf_counter() - t0
            t0 = time.perf_counter()
            F.conv2d(x, wt, stride=s, padding=p)
            torch.cuda.synchronize()
            t_lib += time.perf_counter() - t0
        oh = (h + 2 * p - k) // s + 1
        fl = 2.0 * B * oh * oh * cout * cin * k * k
        print(json.dumps({
            "shape": name, "ok": bool(ok), "max_err": round(err, 4),
            "ours_us": round(t_ours / iters * 1e6, 1),
            "miopen_us": round(t_lib / iters * 1e6, 1),
            "ours_tf": round(fl / (t_ours / iters) / 1e12, 1),
            "miopen_tf": round(fl / (t_lib / iters) / 1e12, 1),
        }))
        assert ok, f"{name} WRONG (err {err} scale {scale})"


if __name__ == "__main__":
    main()


def wgrad_ab():
    from vantage6_amd.ops import hip_ops

    ops = hip_ops()
    dev = "cuda:0"
    torch.manual_seed(0)
    B = 32
    shapes = [
        (64, 56, 56, 64, 3, 1, 1, "l1_3x3"),
        (128, 28, 28, 128, 3, 1, 1, "l2_3x3"),
        (256, 14, 14, 256, 3, 1, 1, "l3_3x3"),
        (512, 7, 7, 512, 3, 1, 1, "l4_3x3"),
        (64, 56, 56, 128, 3, 2, 1, "l2_down3x3"),
        (64, 56, 56, 128, 1, 2, 0, "l2_down1x1"),
    ]
    iters = 30
    for cin, h, w, cout, k, s, p, name in shapes:
        x = torch.randn(B, cin, h, w, device=dev, dtype=torch.bfloat16).to(
            memory_format=torch.channels_last)
        oh = (h + 2 * p - k) // s + 1
        dy = torch.randn(B, cout, oh, oh, device=dev, dtype=torch.bfloat16).to(
            memory_format=torch.channels_last)
        wt = torch.randn(cout, cin, k, k, device=dev, dtype=torch.bfloat16).to(
            memory_format=torch.channels_last)
        import time as _t

        def lib():
            return torch.ops.aten.convolution_backward(
                dy, x, wt, None, [s, s], [p, p], [1, 1], False, [0, 0], 1,
                [False, True, False])[1]

        for _ in range(5):
            ops.conv_nhwc_wgrad(dy, x, k, k, s, p)
            lib()
        torch.cuda.synchronize()
        t_ours = t_lib = 0.0
        for _ in range(iters):
            torch.cuda.synchronize()
            t0 = _t.perf_counter()
            ops.conv_nhwc_wgrad(dy, x, k, k, s, p)
            torch.cuda.synchronize()
            t_ours += _t.perf_counter() - t0
            t0 = _t.perf_counter()
            lib()
            torch.cuda.synchronize()
            t_lib += _t.perf_counter() - t0
        print(json.dumps({"wgrad": name,
                          "ours_us": round(t_ours / iters * 1e6, 1),
                          "miopen_us": round(t_lib / iters * 1e6, 1)}))
