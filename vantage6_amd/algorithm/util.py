"""Algorithm-side logging + env helpers.

Reference parity: ``vantage6-algorithm-tools/vantage6/algorithm/tools/
util.py`` — ``info()/warn()/error()`` write structured
lines to stdout, which the station captures into ``Run.log`` (the
reference captures container stdout the same way), and ``get_env_var``
reads algorithm environment with a default.
"""
from __future__ import annotations

import os
import sys


def info(msg: str) -> None:
    print(f"info > {msg}", flush=True)


def warn(msg: str) -> None:
    print(f"warn > {msg}", flush=True)


def error(msg: str) -> None:
    print(f"error > {msg}", file=sys.stderr, flush=True)
    # mirrored on stdout so it lands in the captured Run.log as well
    print(f"error > {msg}", flush=True)


def get_env_var(name: str, default: str | None = None) -> str | None:
    """Read an algorithm environment variable (the station injects the
    A.3 contract vars plus any ``algorithm_env`` config entries)."""
    return os.environ.get(name, default)
