"""Federated quantiles — bisection over global rank counts, so stations
only ever reveal "how many of my rows are below t", never values.

Mirrors the reference ecosystem's federated-average/quantile community
algorithms: the central part first pools [min, max] and the global row
count, then binary-searches each requested quantile — every iteration
fans out one ``partial_count_below(t)`` task and narrows the bracket
until the value interval is tighter than ``tol`` (relative to the data
range). Station-side disclosure guards: a ``min_rows`` privacy threshold,
and counts are only returned for the whole shard (never per-row).
"""
from vantage6_amd.algorithm import algorithm_client, data
from vantage6_amd.algorithm.exceptions import (InputError,
                                               PrivacyThresholdViolation)


@algorithm_client
def central_quantiles(client, column: str, q: list[float],
                      min_rows: int = 1, tol: float = 1e-6,
                      max_iters: int = 60):
    if not q or any(not (0.0 <= x <= 1.0) for x in q):
        raise InputError("q must be fractions in [0, 1]")
    org_ids = [org["id"] for org in client.organization.list()]

    def fan_out(method, kwargs):
        task = client.task.create(
            input_={"method": method, "kwargs": kwargs},
            organizations=org_ids, name=f"quantile {method}",
        )
        return client.wait_for_results(task["id"])

    ranges = fan_out("partial_range",
                     {"column": column, "min_rows": min_rows})
    n = sum(r["n"] for r in ranges)
    if n == 0:
        return {"n": 0, "column": column, "quantiles": {x: None for x in q}}
    lo0 = min(r["min"] for r in ranges if r["min"] is not None)
    hi0 = max(r["max"] for r in ranges if r["max"] is not None)
    span = max(hi0 - lo0, 1.0e-300)
    out = {}
    for frac in q:
        # rank on the 0..n-1 order statistics; bisection converges to the
        # LOWER order statistic (numpy's method="lower"), the natural
        # result when only counts — never values — cross station borders
        target = frac * (n - 1)
        lo, hi = lo0, hi0
        for _ in range(max_iters):
            if (hi - lo) <= tol * span:
                break
            mid = 0.5 * (lo + hi)
            below = sum(r["count"] for r in fan_out(
                "partial_count_below",
                {"column": column, "t": mid, "min_rows": min_rows}))
            # rank of mid = number of values strictly below it
            if below <= target:
                lo = mid
            else:
                hi = mid
        out[frac] = 0.5 * (lo + hi)
    return {"n": n, "column": column, "quantiles": out}


@data(1)
def partial_range(df, column: str, min_rows: int = 1):
    if len(df) < min_rows:
        raise PrivacyThresholdViolation(
            f"shard has {len(df)} rows < privacy threshold {min_rows}")
    col = df[column].dropna().astype(float)
    return {"n": int(len(col)),
            "min": float(col.min()) if len(col) else None,
            "max": float(col.max()) if len(col) else None}


@data(1)
def partial_count_below(df, column: str, t: float, min_rows: int = 1):
    if len(df) < min_rows:
        raise PrivacyThresholdViolation(
            f"shard has {len(df)} rows < privacy threshold {min_rows}")
    col = df[column].dropna().astype(float)
    return {"count": int((col < t).sum())}
