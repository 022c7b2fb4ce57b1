#!/usr/bin/env python3
"""wgrad 128² (v1) vs 256² (v2) A/B on the training shapes + refcheck."""
import json
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
import torch


def run(variant_env):
    os.environ["V6AMD_WGRAD2"] = variant_env
    # env is read once per process via static — so fork per variant
    import subprocess
    out = subprocess.run(
        [sys.executable, __file__, "child"], capture_output=True, text=True,
        env=dict(os.environ))
    print(out.stdout, end="")
    if out.returncode:
        print(out.stderr[-2000:])


def child():
    from vantage6_amd.ops import hip_ops

    ops = hip_ops()
    dev = "cuda:0"
    torch.manual_seed(0)
    tag = os.environ.get("V6AMD_WGRAD2", "1")
    # refcheck 512³
    a = torch.randn(512, 512, device=dev, dtype=torch.bfloat16)
    b = torch.randn(512, 512, device=dev, dtype=torch.bfloat16)
    ref = (a.float().t() @ b.float()).to(torch.bfloat16).float()
    got = ops.gemm_wgrad(a, b).float()
    err = (got - ref).abs().max().item()
    ok = err <= 0.05 * ref.abs().max().item() + 0.05
    print(json.dumps({"v2": tag, "refcheck_err": round(err, 4), "ok": bool(ok)}))
    assert ok
    shapes = [  # (M, Na, Nb)
        (1024, 2048, 2048), (1024, 8192, 2048), (1024, 2048, 8192),
        (1024, 512, 2048), (4096, 768, 768), (4096, 3072, 768),
        (4096, 768, 3072), (8192, 2048, 2048),
    ]
    for m, na, nb in shapes:
        a = torch.randn(m, na, device=dev, dtype=torch.bfloat16)
        b = torch.randn(m, nb, device=dev, dtype=torch.bfloat16)
        for _ in range(5):
            ops.gemm_wgrad(a, b)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(30):
            ops.gemm_wgrad(a, b)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 30
        fl = 2.0 * m * na * nb
        print(json.dumps({"v2": tag, "shape": [m, na, nb],
                          "us": round(dt * 1e6, 1),
                          "tf": round(fl / dt / 1e12, 1)}))


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "child":
        child()
    else:
        run("0")
        run("1")
