"""Federated summary statistics — count / mean / std / min / max / NA count
per column, without any station revealing row-level data.

Mirrors the reference ecosystem's ``v6-summary-statistics`` community
algorithm (the standard first analysis in vantage6 studies): a central
function fans out partials and combines moment sums; variance is computed
from pooled sum / sum-of-squares, so only aggregates ever leave a station.
A ``min_rows`` threshold (privacy guard, as in the reference algorithm's
disclosure checks) makes stations refuse to answer on tiny shards.
"""
from vantage6_amd.algorithm import algorithm_client, data
from vantage6_amd.algorithm.exceptions import PrivacyThresholdViolation


@algorithm_client
def central_summary(client, columns: list[str], min_rows: int = 1):
    org_ids = [org["id"] for org in client.organization.list()]
    task = client.task.create(
        input_={"method": "partial_summary",
                "kwargs": {"columns": columns, "min_rows": min_rows}},
        organizations=org_ids,
        name="partial summary statistics",
    )
    results = client.wait_for_results(task["id"])
    out = {}
    for c in columns:
        n = sum(r[c]["n"] for r in results)
        na = sum(r[c]["na"] for r in results)
        if n == 0:
            out[c] = {"n": 0, "na": na, "mean": None, "std": None,
                      "min": None, "max": None}
            continue
        s1 = sum(r[c]["sum"] for r in results)
        s2 = sum(r[c]["sum_sq"] for r in results)
        mean = s1 / n
        var = max(s2 / n - mean * mean, 0.0)
        out[c] = {
            "n": n,
            "na": na,
            "mean": mean,
            "std": (var * n / (n - 1)) ** 0.5 if n > 1 else 0.0,
            "min": min(r[c]["min"] for r in results),
            "max": max(r[c]["max"] for r in results),
        }
    return out


@data(1)
def partial_summary(df, columns: list[str], min_rows: int = 1):
    if len(df) < min_rows:
        raise PrivacyThresholdViolation(
            f"shard has {len(df)} rows < privacy threshold {min_rows}")
    out = {}
    for c in columns:
        col = df[c].dropna()
        out[c] = {
            "n": int(col.count()),
            "na": int(df[c].isna().sum()),
            "sum": float(col.sum()),
            "sum_sq": float((col * col).sum()),
            "min": float(col.min()) if len(col) else None,
            "max": float(col.max()) if len(col) else None,
        }
    return out
