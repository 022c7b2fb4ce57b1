"""Federated two-sample t-test — exact from per-station moment sums.

Group means and variances pool exactly from (n, Σx, Σx²) per station and
group, so the federated statistic is identical to running scipy on the
pooled rows. Supports Welch (default, unequal variances) and Student
(pooled variance) variants; p-value from the t survival function.

The grouping column must be binary-valued across stations (the two
groups to compare); rows with NA in value or group are dropped
station-side. A ``min_rows`` privacy threshold guards tiny shards.
"""
import math

from scipy import stats as _st

from vantage6_amd.algorithm import algorithm_client, data
from vantage6_amd.algorithm.exceptions import (InputError,
                                               PrivacyThresholdViolation)


@algorithm_client
def central_ttest(client, value: str, group: str, equal_var: bool = False,
                  min_rows: int = 1):
    org_ids = [org["id"] for org in client.organization.list()]
    task = client.task.create(
        input_={"method": "partial_ttest_moments",
                "kwargs": {"value": value, "group": group,
                           "min_rows": min_rows}},
        organizations=org_ids,
        name="partial t-test moments",
    )
    results = client.wait_for_results(task["id"])
    levels = sorted({lv for r in results for lv in r})
    if len(levels) != 2:
        raise InputError(f"group column {group!r} must have exactly two "
                         f"levels across stations, got {levels}")
    pooled = {}
    for lv in levels:
        n = sum(r.get(lv, {"n": 0})["n"] for r in results)
        s1 = sum(r.get(lv, {"sum": 0.0})["sum"] for r in results)
        s2 = sum(r.get(lv, {"sum_sq": 0.0})["sum_sq"] for r in results)
        if n < 2:
            raise InputError(f"group {lv!r} has {n} rows; need >= 2")
        mean = s1 / n
        var = max(s2 - n * mean * mean, 0.0) / (n - 1)  # sample variance
        pooled[lv] = {"n": n, "mean": mean, "var": var}
    a, b = (pooled[lv] for lv in levels)
    if equal_var:
        df = a["n"] + b["n"] - 2
        sp2 = ((a["n"] - 1) * a["var"] + (b["n"] - 1) * b["var"]) / df
        se = math.sqrt(sp2 * (1.0 / a["n"] + 1.0 / b["n"]))
    else:  # Welch
        va, vb = a["var"] / a["n"], b["var"] / b["n"]
        se = math.sqrt(va + vb)
        df = (va + vb) ** 2 / (va ** 2 / (a["n"] - 1) + vb ** 2 / (b["n"] - 1))
    t = (a["mean"] - b["mean"]) / se
    p = 2.0 * float(_st.t.sf(abs(t), df))
    return {
        "groups": {str(lv): pooled[lv] for lv in levels},
        "t": t,
        "df": df,
        "p_value": p,
        "equal_var": equal_var,
    }


@data(1)
def partial_ttest_moments(df, value: str, group: str, min_rows: int = 1):
    if len(df) < min_rows:
        raise PrivacyThresholdViolation(
            f"shard has {len(df)} rows < privacy threshold {min_rows}")
    sub = df[[value, group]].dropna()
    out = {}
    for lv, g in sub.groupby(group):
        col = g[value].astype(float)
        out[str(lv)] = {
            "n": int(len(col)),
            "sum": float(col.sum()),
            "sum_sq": float((col * col).sum()),
        }
    return out
