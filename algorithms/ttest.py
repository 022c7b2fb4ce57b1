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


@algorithm_client
def central_anova(client, value: str, group: str, min_rows: int = 1):
    """Federated one-way ANOVA — exact from the same per-group moment
    sums (matches scipy.stats.f_oneway on the pooled rows)."""
    org_ids = [org["id"] for org in client.organization.list()]
    task = client.task.create(
        input_={"method": "partial_ttest_moments",
                "kwargs": {"value": value, "group": group,
                           "min_rows": min_rows}},
        organizations=org_ids,
        name="partial anova moments",
    )
    results = client.wait_for_results(task["id"])
    levels = sorted({lv for r in results for lv in r})
    if len(levels) < 2:
        raise InputError(f"group column {group!r} needs >= 2 levels")
    n_tot, s_tot, ss_tot = 0, 0.0, 0.0
    groups = {}
    for lv in levels:
        n = sum(r.get(lv, {"n": 0})["n"] for r in results)
        s1 = sum(r.get(lv, {"sum": 0.0})["sum"] for r in results)
        s2 = sum(r.get(lv, {"sum_sq": 0.0})["sum_sq"] for r in results)
        if n < 1:
            continue
        groups[lv] = (n, s1, s2)
        n_tot += n
        s_tot += s1
        ss_tot += s2
    k = len(groups)
    grand_mean = s_tot / n_tot
    ss_between = sum(n * (s1 / n - grand_mean) ** 2
                     for n, s1, s2 in groups.values())
    ss_within = sum(s2 - s1 * s1 / n for n, s1, s2 in groups.values())
    df_b, df_w = k - 1, n_tot - k
    if df_w <= 0 or ss_within <= 0:
        raise InputError("not enough data for ANOVA")
    f = (ss_between / df_b) / (ss_within / df_w)
    p = float(_st.f.sf(f, df_b, df_w))
    return {"F": f, "p_value": p, "df_between": df_b, "df_within": df_w,
            "groups": {str(lv): {"n": n, "mean": s1 / n}
                       for lv, (n, s1, s2) in groups.items()}}


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
