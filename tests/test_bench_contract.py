"""Driver-contract test for bench.py: the round-end driver launches
``python -m torch.distributed.run --nnodes=1 --nproc-per-node N
--master-addr 127.0.0.1 bench.py --gpus N ...`` — this runs that exact
shape on CPU (gloo, world 2) and validates the single JSON line."""
import json
import os
import subprocess
import sys
from pathlib import Path

import pytest

ROOT = Path(__file__).resolve().parents[1]


def _run(cmd, timeout=300):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    return subprocess.run(cmd, cwd=ROOT, capture_output=True, text=True,
                          timeout=timeout, env=env)


def _last_json(stdout: str) -> dict:
    lines = [ln for ln in stdout.strip().splitlines() if ln.startswith("{")]
    assert lines, f"no JSON line in output:\n{stdout[-2000:]}"
    return json.loads(lines[-1])


def test_bench_single_process_cpu():
    r = _run([sys.executable, "bench.py", "--device", "cpu", "--model",
              "logreg", "--steps", "2", "--warmup", "1"])
    assert r.returncode == 0, r.stderr[-2000:]
    out = _last_json(r.stdout)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in out, f"missing {key}"
    assert out["n_gpus"] == 1 and out["steps"] == 2
    assert out["data"] == "synthetic"


def test_bench_two_rank_gloo():
    r = _run([sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
              "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
              "--master-port", "29712", "bench.py", "--gpus", "2",
              "--device", "cpu", "--model", "logreg",
              "--steps", "2", "--warmup", "1"], timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    out = _last_json(r.stdout)
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"].startswith("federated")
    # exactly ONE rank prints the JSON line
    njson = sum(1 for ln in r.stdout.splitlines() if ln.startswith("{"))
    assert njson == 1, r.stdout[-1500:]
