"""Injection decorators for algorithm functions.

Reference parity: ``vantage6-algorithm-tools/vantage6/algorithm/tools/
decorators.py`` — ``@data(n)`` loads the station's
mounted database(s) into pandas DataFrames (from ``DATABASE_URI`` /
``<LABEL>_DATABASE_URI`` env entries), ``@algorithm_client`` injects the
:class:`AlgorithmClient`, ``@metadata`` injects run metadata.

Supported database types: ``csv``, ``parquet``, ``excel``, ``tensor``
(a safetensors/torch ``.pt`` file loaded as tensors — net-new for the GPU
stations) and ``synthetic`` (shape-spec URI, generates a deterministic
random shard — used by the benchmarks, there is no dataset egress).
"""
from __future__ import annotations

import functools
from dataclasses import dataclass
from typing import Callable

from .exceptions import DataError, EnvironmentVariableError


@dataclass
class RunMetadata:
    """Injected by ``@metadata`` (task/node identifiers, temp folder)."""

    task_id: int | None
    node_id: int | None
    organization_id: int | None
    collaboration_id: int | None
    temporary_directory: str | None


def _load_database(uri: str, dbtype: str, env: dict[str, str]):
    import pandas as pd

    dbtype = (dbtype or "csv").lower()
    try:
        if dbtype == "csv":
            return pd.read_csv(uri)
        if dbtype == "parquet":
            return pd.read_parquet(uri)
        if dbtype == "excel":
            return pd.read_excel(uri)
        if dbtype == "tensor":
            import torch

            return torch.load(uri, weights_only=True)
        if dbtype == "synthetic":
            return _synthetic_shard(uri, env)
        if dbtype == "sql":
            # reference: sql-type node databases; the shard is a table or
            # query against a sqlalchemy URI. Select via URI fragment:
            #   sqlite:///data.db#table=patients
            #   postgresql://...#query=SELECT%20...
            import urllib.parse

            from sqlalchemy import create_engine

            base, _, frag = uri.partition("#")
            q = dict(urllib.parse.parse_qsl(frag))
            engine = create_engine(base)
            try:
                if "query" in q:
                    return pd.read_sql(urllib.parse.unquote(q["query"]),
                                       engine)
                if "table" in q:
                    return pd.read_sql_table(q["table"], engine)
                raise DataError(
                    "sql database URI needs a #table=<name> or "
                    "#query=<urlencoded sql> fragment")
            finally:
                engine.dispose()
        if dbtype in ("sparql", "omop"):
            raise DataError(
                f"{dbtype} databases need client wheels not present in this "
                f"offline image (SPARQLWrapper / OMOP CDM drivers)")
    except DataError:
        raise
    except Exception as exc:  # noqa: BLE001
        raise DataError(f"failed to load {dbtype} database from {uri!r}: {exc}") from exc
    raise DataError(f"unsupported database type {dbtype!r}")


def _synthetic_shard(uri: str, env: dict[str, str]):
    """``synthetic://<kind>?n=..&d=..&seed=..`` — deterministic random data.

    Kinds: ``tabular`` (features+labels tensors), ``images`` (NCHW bf16),
    ``tokens`` (int64 token ids). Seed is combined with the station's
    organization id so shards differ per station, deterministically.
    """
    import urllib.parse

    import torch

    parsed = urllib.parse.urlparse(uri)
    kind = parsed.netloc or parsed.path.strip("/")
    q = dict(urllib.parse.parse_qsl(parsed.query))
    n = int(q.get("n", 1024))
    seed = int(q.get("seed", 0)) * 1000 + int(env.get("ORGANIZATION_ID", 0) or 0)
    g = torch.Generator().manual_seed(seed)
    if kind == "tabular":
        d = int(q.get("d", 32))
        classes = int(q.get("classes", 2))
        x = torch.randn(n, d, generator=g)
        w = torch.randn(d, classes, generator=g)
        y = (x @ w + 0.1 * torch.randn(n, classes, generator=g)).argmax(dim=1)
        return {"x": x, "y": y}
    if kind == "images":
        c, h, wdt = int(q.get("c", 3)), int(q.get("h", 224)), int(q.get("w", 224))
        classes = int(q.get("classes", 1000))
        x = torch.randn(n, c, h, wdt, generator=g)
        y = torch.randint(0, classes, (n,), generator=g)
        return {"x": x, "y": y}
    if kind == "tokens":
        s = int(q.get("seq", 512))
        vocab = int(q.get("vocab", 32000))
        return {"input_ids": torch.randint(0, vocab, (n, s), generator=g)}
    raise DataError(f"unknown synthetic kind {kind!r}")


def data(number_of_databases: int = 1):
    """Inject the first N station databases as leading positional args."""

    def decorator(func: Callable) -> Callable:
        @functools.wraps(func)
        def wrapper(*args, **kwargs):
            env = kwargs.pop("_v6_env", None) or {}
            if getattr(func, "_v6_wants_env", False):
                kwargs["_v6_env"] = env  # an inner decorator also wants the env
            frames = []
            labels = [s for s in env.get("USER_REQUESTED_DATABASE_LABELS", "").split(",") if s]
            if not labels:
                labels = ["default"] * number_of_databases
            for i in range(number_of_databases):
                label = labels[i] if i < len(labels) else labels[-1]
                key = f"{label.upper()}_DATABASE_URI"
                uri = env.get(key) or env.get("DATABASE_URI")
                if not uri:
                    raise EnvironmentVariableError(
                        f"no database uri for label {label!r} (looked for {key})"
                    )
                dbtype = env.get(f"{label.upper()}_DATABASE_TYPE", env.get("DATABASE_TYPE", "csv"))
                frame = _load_database(uri, dbtype, env)
                prep = env.get(f"{label.upper()}_PREPROCESSING")
                if prep:
                    import json as _json

                    from .preprocessing import apply_preprocessing

                    frame = apply_preprocessing(frame, _json.loads(prep))
                frames.append(frame)
            return func(*frames, *args, **kwargs)

        wrapper._v6_wants_env = True
        wrapper._v6_databases = number_of_databases
        return wrapper

    return decorator


def algorithm_client(func: Callable) -> Callable:
    """Inject the AlgorithmClient as the first positional argument."""

    @functools.wraps(func)
    def wrapper(*args, **kwargs):
        client = kwargs.pop("_v6_client", None)
        return func(client, *args, **kwargs)

    wrapper._v6_wants_client = True
    return wrapper


def metadata(func: Callable) -> Callable:
    """Inject a :class:`RunMetadata` as the first positional argument."""

    @functools.wraps(func)
    def wrapper(*args, **kwargs):
        env = kwargs.pop("_v6_env", None) or {}
        if getattr(func, "_v6_wants_env", False):
            kwargs["_v6_env"] = env

        def _maybe_int(v):
            return int(v) if v not in (None, "") else None

        md = RunMetadata(
            task_id=_maybe_int(env.get("TASK_ID")),
            node_id=_maybe_int(env.get("NODE_ID")),
            organization_id=_maybe_int(env.get("ORGANIZATION_ID")),
            collaboration_id=_maybe_int(env.get("COLLABORATION_ID")),
            temporary_directory=env.get("TEMPORARY_FOLDER"),
        )
        return func(md, *args, **kwargs)

    wrapper._v6_wants_env = True
    return wrapper
