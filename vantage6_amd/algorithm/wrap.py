"""Algorithm entrypoint glue — method dispatch over the I/O contract.

Reference parity: ``vantage6-algorithm-tools/vantage6/algorithm/tools/
wrap.py`` — read ``INPUT_FILE`` JSON
``{"method", "args", "kwargs"}``, resolve the function in the algorithm
package, perform decorator injection, call it, serialize the return value to
``OUTPUT_FILE``.

The station executor calls :func:`dispatch` directly (in-process, env as a
dict); :func:`wrap_algorithm` is the file/env entrypoint for running an
algorithm standalone exactly like a reference container would.
"""
from __future__ import annotations

import importlib
import os
from typing import Any

from ..common.serialization import deserialize, serialize
from .exceptions import InputError


def resolve_module(image: str):
    """Resolve an algorithm "image" reference to a Python module.

    Accepted forms (the MI355X analog of a Docker image name):
    ``pkg.module`` import path, or ``local:///path/to/pkg.module``.
    """
    name = image
    if name.startswith("local://"):
        name = name[len("local://") :]
    # tolerate docker-style references for schema compatibility:
    # "registry/algorithms.column_mean:latest" -> "algorithms.column_mean"
    if "/" in name:
        name = name.rsplit("/", 1)[-1]
    if ":" in name:
        name = name.split(":", 1)[0]
    try:
        return importlib.import_module(name)
    except ImportError as exc:
        raise InputError(f"cannot import algorithm package {name!r}: {exc}") from exc


def dispatch(
    module,
    payload: dict[str, Any],
    env: dict[str, str],
    client=None,
) -> Any:
    """Execute one algorithm method with decorator injection."""
    method_name = payload.get("method")
    if not method_name:
        raise InputError("input payload has no 'method'")
    # in-process stations have no container boundary, so the dispatch gate
    # is tighter than the reference's: only PUBLIC functions DEFINED IN the
    # algorithm module are callable — never imported names (a module-level
    # `from os import system` must not become a remote method) and never
    # underscore-prefixed helpers
    func = getattr(module, method_name, None) \
        if not method_name.startswith("_") else None
    if func is None or not callable(func):
        raise InputError(f"method {method_name!r} not found in {module.__name__}")
    is_marked = (getattr(func, "_v6_wants_env", False)
                 or getattr(func, "_v6_wants_client", False))
    if not is_marked and getattr(func, "__module__", None) != module.__name__:
        # @data/@algorithm_client-decorated functions are explicitly
        # exported task methods wherever they were defined (re-export
        # from a submodule is a normal algorithm-package layout)
        raise InputError(
            f"method {method_name!r} is not defined in {module.__name__} "
            f"(imported names are not callable as task methods)")
    args = payload.get("args", []) or []
    kwargs = dict(payload.get("kwargs", {}) or {})
    if getattr(func, "_v6_wants_env", False):
        kwargs["_v6_env"] = env
    if getattr(func, "_v6_wants_client", False):
        kwargs["_v6_client"] = client
    return func(*args, **kwargs)


def wrap_algorithm(module_name: str | None = None, environ: dict | None = None) -> None:
    """Standalone entrypoint: file/env contract exactly as the reference.

    Reads ``INPUT_FILE``, writes ``OUTPUT_FILE``; the module comes from the
    ``PKG_NAME`` env entry when not given.
    """
    env = dict(environ if environ is not None else os.environ)
    module_name = module_name or env.get("PKG_NAME")
    if not module_name:
        raise InputError("no algorithm package: pass module_name or set PKG_NAME")
    module = importlib.import_module(module_name)
    with open(env["INPUT_FILE"], "rb") as fh:
        payload = deserialize(fh.read())
    client = None
    token_file = env.get("TOKEN_FILE")
    if token_file and os.path.exists(token_file):
        from .client import AlgorithmClient

        with open(token_file) as fh:
            token = fh.read().strip()
        host, port = env.get("HOST", "http://127.0.0.1"), env.get("PORT", "80")
        api_path = env.get("API_PATH", "/api")
        client = AlgorithmClient.from_http(token=token, host=host, port=port, api_path=api_path)
    result = dispatch(module, payload, env, client)
    with open(env["OUTPUT_FILE"], "wb") as fh:
        fh.write(serialize(result))
