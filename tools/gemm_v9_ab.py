#!/usr/bin/env python3
"""Within-probe interleaved A/B of NT GEMM variants (guide §5.4 rule:
cross-run noise ±30 TF, so variants are timed round-robin in ONE process)
plus bitwise race screens and an fp32-torch refcheck.

Run on a GPU box:  python tools/gemm_v9_ab.py [--iters 30]
"""
import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=30)
    ap.add_argument("--race-runs", type=int, default=10)
    args = ap.parse_args()
    from vantage6_amd.ops import hip_ops

    ops = hip_ops()
    dev = "cuda:0"
    torch.manual_seed(0)

    # ---- refcheck at 512³ vs fp32 torch --------------------------------
    a = (torch.rand(512, 512, device=dev, dtype=torch.bfloat16) - 0.5)
    b = (torch.rand(512, 512, device=dev, dtype=torch.bfloat16) - 0.5)
    ref = (a.float() @ b.float().t()).to(torch.bfloat16).float()
    for variant, nsplit in [(7, 1), (8, 1), (9, 1), (10, 1), (9, 2), (10, 2),
                            (9, 4), (10, 4)]:
        c = ops.gemm_nt_probe(a, b, variant, nsplit).float()
        err = (c - ref).abs().max().item()
        scale = ref.abs().max().item()
        ok = err <= 0.05 * scale + 0.05
        print(json.dumps({"refcheck": variant, "nsplit": nsplit,
                          "max_err": err, "ok": bool(ok)}))
        assert ok, f"variant {variant}/ns{nsplit} WRONG (err {err})"

    # ---- race screen: bitwise stability over N runs --------------------
    for m, n, k in [(4096, 4096, 4096), (4096, 4096, 8192)]:
        a = torch.randn(m, k, device=dev, dtype=torch.bfloat16)
        b = torch.randn(n, k, device=dev, dtype=torch.bfloat16)
        for variant, nsplit in [(9, 1), (10, 1), (9, 4), (10, 4)]:
            first = ops.gemm_nt_probe(a, b, variant, nsplit)
            stable = True
            for _ in range(args.race_runs - 1):
                again = ops.gemm_nt_probe(a, b, variant, nsplit)
                if not torch.equal(first, again):
                    stable = False
                    break
            print(json.dumps({"race_screen": variant, "nsplit": nsplit,
                              "shape": [m, n, k], "bitwise_stable": stable}))
            assert stable, f"variant {variant} RACES at {(m,n,k)}"

    # ---- interleaved A/B timing ----------------------------------------
    shapes = [
        ("4k3", 4096, 4096, 4096, [(7, 1), (8, 1), (9, 1), (10, 1)]),
        ("8k3", 8192, 8192, 8192, [(7, 1), (8, 1), (9, 1), (10, 1)]),
        # llama starved shapes: few 256² tiles — split-K candidates vs v5
        ("llama_down", 1024, 2048, 8192,
         [(5, 1), (7, 1), (9, 4), (9, 8), (10, 4), (10, 8)]),
        ("llama_qo", 1024, 2048, 2048, [(5, 1), (9, 4), (9, 8), (10, 4)]),
        ("llama_gate", 1024, 8192, 2048, [(5, 1), (7, 1), (9, 2), (9, 4), (10, 4)]),
        ("bert_head", 4096, 30720, 768, [(5, 1), (7, 1), (9, 2), (10, 2)]),
        ("lm_head", 1024, 128256, 2048, [(5, 1), (7, 1), (9, 1), (9, 2), (10, 1)]),
        # bert fwd shapes (M=4096): nwg6 between the 128/384 thresholds
        ("bert_qkv", 4096, 2304, 768, [(5, 1), (7, 1), (10, 1)]),
        ("bert_fc1", 4096, 3072, 768, [(5, 1), (7, 1), (10, 1)]),
        ("bert_fc2", 4096, 768, 3072, [(5, 1), (7, 1), (9, 2), (10, 2)]),
        ("bert_proj", 4096, 768, 768, [(5, 1), (7, 1), (10, 1)]),
    ]
    for name, m, n, k, variants in shapes:
        a = torch.randn(m, k, device=dev, dtype=torch.bfloat16)
        b = torch.randn(n, k, device=dev, dtype=torch.bfloat16)
        fl = 2.0 * m * n * k
        # warmup all
        for v, ns in variants:
            for _ in range(3):
                ops.gemm_nt_probe(a, b, v, ns)
        torch.cuda.synchronize()
        acc = {f"{v}/{ns}": 0.0 for v, ns in variants}
        for _ in range(args.iters):
            for v, ns in variants:
                torch.cuda.synchronize()
                t0 = time.perf_counter()
                ops.gemm_nt_probe(a, b, v, ns)
                torch.cuda.synchronize()
                acc[f"{v}/{ns}"] += time.perf_counter() - t0
        res = {kk: round(fl / (vv / args.iters) / 1e12, 1)
               for kk, vv in acc.items()}
        # library reference
        bt = b  # torch NT via .t()
        for _ in range(3):
            a @ bt.t()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            a @ bt.t()
        torch.cuda.synchronize()
        res["hipblaslt"] = round(fl / ((time.perf_counter() - t0) / args.iters) / 1e12, 1)
        print(json.dumps({"shape": name, "mnk": [m, n, k], "tflops": res}))


if __name__ == "__main__":
    main()
