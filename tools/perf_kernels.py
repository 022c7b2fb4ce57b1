#!/usr/bin/env python3
"""Kernel microbenchmarks on MI355X: TFLOP/s and GB/s vs library baselines.

Run on a GPU box: ``python tools/perf_kernels.py [--quick]``.
Prints one JSON line per measurement (kernel, shape, ms, tflops/gbps).
"""
import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch


def timeit(fn, warmup=3, iters=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--quick", action="store_true")
    args = ap.parse_args()
    assert torch.cuda.is_available()
    from vantage6_amd.ops import hip_ops

    ops = hip_ops()
    dev = "cuda:0"
    out = []

    # ---------------- GEMM: ours vs torch(hipBLASLt) ----------------
    sizes = [(2048, 2048, 2048), (4096, 4096, 4096)]
    if not args.quick:
        sizes += [(8192, 8192, 8192), (4096, 1000, 512), (16384, 3072, 768)]
    for m, n, k in sizes:
        a = torch.randn(m, k, dtype=torch.bfloat16, device=dev)
        b = torch.randn(n, k, dtype=torch.bfloat16, device=dev)
        t_ours = timeit(lambda: ops.gemm_bf16(a, b, False, True))
        bt = b.t().contiguous()
        t_lib = timeit(lambda: a @ b.t())
        fl = 2.0 * m * n * k
        out.append({"kernel": "gemm_nt_bf16", "shape": [m, n, k],
                    "ms": t_ours * 1e3, "tflops": fl / t_ours / 1e12,
                    "lib_ms": t_lib * 1e3, "lib_tflops": fl / t_lib / 1e12})

    # correctness spot-check at 4096³ on-device
    m, n, k = 2048, 2048, 2048
    a = torch.randn(m, k, dtype=torch.bfloat16, device=dev)
    b = torch.randn(n, k, dtype=torch.bfloat16, device=dev)
    c = ops.gemm_bf16(a, b, False, True).float()
    ref = (a.float() @ b.float().t())
    rel = (c - ref).abs().max().item() / ref.abs().max().item()
    out.append({"kernel": "gemm_check", "rel_err": rel})

    # ---------------- transpose ----------------
    x = torch.randn(8192, 8192, dtype=torch.bfloat16, device=dev)
    t = timeit(lambda: ops.transpose_bf16(x))
    out.append({"kernel": "transpose_bf16", "shape": [8192, 8192],
                "ms": t * 1e3, "gbps": 2 * x.numel() * 2 / t / 1e9})

    # ---------------- norms ----------------
    for rows, h in [(16384, 2048), (8192, 8192)]:
        x = torch.randn(rows, h, dtype=torch.bfloat16, device=dev)
        w = torch.randn(h, dtype=torch.bfloat16, device=dev)
        bb = torch.randn(h, dtype=torch.bfloat16, device=dev)
        t_f = timeit(lambda: ops.layernorm_fwd(x, w, bb, 1e-5))
        tl = timeit(lambda: torch.nn.functional.layer_norm(x, (h,), w, bb))
        bytes_moved = 2 * x.numel() * 2  # read + write bf16
        out.append({"kernel": "layernorm_fwd", "shape": [rows, h],
                    "ms": t_f * 1e3, "gbps": bytes_moved / t_f / 1e9,
                    "lib_ms": tl * 1e3, "lib_gbps": bytes_moved / tl / 1e9})
        t_r = timeit(lambda: ops.rmsnorm_fwd(x, w, 1e-6))
        out.append({"kernel": "rmsnorm_fwd", "shape": [rows, h],
                    "ms": t_r * 1e3, "gbps": bytes_moved / t_r / 1e9})

    # ---------------- fused sgd ----------------
    n = 50_000_000 if not args.quick else 10_000_000
    master = torch.randn(n, device=dev)
    param = master.to(torch.bfloat16)
    grad = torch.randn(n, device=dev).to(torch.bfloat16)
    mom = torch.zeros(n, device=dev)
    t = timeit(lambda: ops.fused_sgd(param, master, grad, mom, 0.1, 0.9, 1e-4, False))
    # traffic: read grad(2)+master(4)+mom(4), write master(4)+mom(4)+param(2) = 20 B/elem
    out.append({"kernel": "fused_sgd", "n": n, "ms": t * 1e3,
                "gbps": 20.0 * n / t / 1e9})

    # torch eager equivalent (5 separate kernels) for comparison
    def eager_sgd():
        g = grad.float().add_(0)  # materialize
        mom.mul_(0.9).add_(g)
        master.add_(mom, alpha=-0.1)
        param.copy_(master.to(torch.bfloat16))

    t = timeit(eager_sgd)
    out.append({"kernel": "eager_sgd_reference", "n": n, "ms": t * 1e3})

    # ---------------- secagg ----------------
    master = torch.randn(12_000_000, device=dev)
    q = torch.empty(12_000_000, dtype=torch.int64, device=dev)
    t = timeit(lambda: ops.secagg_quantize_mask(q, master, 0.125, float(1 << 32),
                                                1234, 3, 8))
    out.append({"kernel": "secagg_quantize_mask(world=8)", "n": 12_000_000,
                "ms": t * 1e3, "gbps": 12.0 * 12_000_000 / t / 1e9})

    for row in out:
        print(json.dumps(row))


if __name__ == "__main__":
    main()
