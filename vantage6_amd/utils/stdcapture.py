"""Thread-local stdout/stderr capture.

The station captures each run's prints as its log (reference: container
stdout → Run.log), but runs execute on worker THREADS of one process —
``contextlib.redirect_stdout`` swaps the process-global ``sys.stdout`` and
concurrent runs corrupt each other (and the main thread loses its stdout).
This router is installed once per stream; each thread can point its writes
at a private buffer; every other thread keeps the original stream.
"""
from __future__ import annotations

import io
import sys
import threading
from contextlib import contextmanager


class _Router(io.TextIOBase):
    def __init__(self, default):
        self._default = default
        self._local = threading.local()

    def _target(self):
        return getattr(self._local, "target", None) or self._default

    def write(self, s):  # noqa: D102
        return self._target().write(s)

    def flush(self):  # noqa: D102
        try:
            self._target().flush()
        except Exception:  # noqa: BLE001
            pass

    def isatty(self):  # noqa: D102
        try:
            return self._default.isatty()
        except Exception:  # noqa: BLE001
            return False

    @property
    def encoding(self):
        return getattr(self._default, "encoding", "utf-8")


_lock = threading.Lock()
_routers: dict[str, _Router] = {}


def _install(stream_name: str) -> _Router:
    with _lock:
        current = getattr(sys, stream_name)
        if isinstance(current, _Router):
            return current
        router = _Router(current)
        setattr(sys, stream_name, router)
        _routers[stream_name] = router
        return router


@contextmanager
def capture_thread_output():
    """Capture this thread's stdout+stderr into a StringIO; yields it."""
    out_router = _install("stdout")
    err_router = _install("stderr")
    buf = io.StringIO()
    out_router._local.target = buf
    err_router._local.target = buf
    try:
        yield buf
    finally:
        out_router._local.target = None
        err_router._local.target = None
