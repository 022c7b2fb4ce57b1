"""Hand-written CDNA4 compute kernels + dispatch layer.

This package supplies the per-station compute that the reference delegates
to user Docker images (the reference has ZERO native/GPU
code — these kernels are the MI355X-native replacement mandated by the
north star):

- ``gemm``      — bf16 MFMA GEMM (LDS-tiled, gfx950) for Linear fwd/bwd
- ``norms``     — LayerNorm / RMSNorm fwd+bwd
- ``optim``     — fused SGD-momentum / AdamW on flat parameter buffers
- ``dp``        — DP-SGD per-sample clip + Philox gaussian noise
- ``secagg``    — pairwise Philox secure-aggregation masks

Dispatch policy: on CUDA(HIP) tensors the hand-written extension is
REQUIRED — a missing extension raises instead of silently falling back to
eager PyTorch; on CPU tensors a plain PyTorch reference implementation runs
(used by the numerics tests and the cpu-sim stations).
"""
from __future__ import annotations

_EXT = None
_EXT_ERR: str | None = None


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        import importlib

        import torch  # noqa: F401 — the extension links against torch libs

        _EXT = importlib.import_module("vantage6_amd.ops._hip_ops")
    except ImportError as exc:
        _EXT_ERR = str(exc)
        _EXT = None
    return _EXT


def hip_ops():
    """The compiled extension module, or raise (GPU path must be native)."""
    ext = _load_extension()
    if ext is None:
        raise RuntimeError(
            "vantage6_amd HIP extension (_hip_ops) is not built — run "
            "`python setup.py build_ext --inplace` (or __graft_entry__.build()); "
            f"import error: {_EXT_ERR}"
        )
    return ext


def has_hip_ops() -> bool:
    return _load_extension() is not None
