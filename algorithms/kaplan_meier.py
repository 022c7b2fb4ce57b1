"""Federated Kaplan-Meier survival analysis.

The signature analysis for the reference's cancer-registry setting
(mirrors the ``v6-kaplan-meier`` community algorithm): stations share only
per-timepoint event/censoring COUNTS — never patient rows — and the
central function merges them into the product-limit estimator
``S(t) = prod_{t_i <= t} (1 - d_i / n_i)``.

Columns: ``time_col`` (follow-up time) and ``event_col`` (1 = event,
0 = censored). ``min_rows`` guards small shards (disclosure check).
"""
import math

from vantage6_amd.algorithm import algorithm_client, data
from vantage6_amd.algorithm.exceptions import PrivacyThresholdViolation


def _merge_counts(parts):
    """Sum per-station {n, events{t:d}, censored{t:c}} tables."""
    events: dict[float, int] = {}
    censored: dict[float, int] = {}
    total = 0
    for res in parts:
        total += res["n"]
        for t, d in res["events"].items():
            events[float(t)] = events.get(float(t), 0) + d
        for t, c in res["censored"].items():
            censored[float(t)] = censored.get(float(t), 0) + c
    return total, events, censored


@algorithm_client
def central_kaplan_meier(client, time_col: str, event_col: str,
                         group_col: str | None = None, min_rows: int = 1):
    """One pooled curve, or (with ``group_col``) one curve per group —
    the survival plot that pairs with ``central_logrank``."""
    org_ids = [org["id"] for org in client.organization.list()]
    if group_col is not None:
        task = client.task.create(
            input_={"method": "partial_event_counts_grouped",
                    "kwargs": {"time_col": time_col, "event_col": event_col,
                               "group_col": group_col,
                               "min_rows": min_rows}},
            organizations=org_ids, name="partial KM counts by group")
        results = client.wait_for_results(task["id"])
        levels = sorted({lv for r in results for lv in r["groups"]})
        out = {}
        for lv in levels:
            total, ev, ce = _merge_counts(
                [r["groups"][lv] for r in results if lv in r["groups"]])
            out[lv] = {"n": total, "curve": _build_curve(total, ev, ce)}
        return {"groups": out}
    task = client.task.create(
        input_={"method": "partial_event_counts",
                "kwargs": {"time_col": time_col, "event_col": event_col,
                           "min_rows": min_rows}},
        organizations=org_ids,
        name="partial KM event counts",
    )
    results = client.wait_for_results(task["id"])
    total, events, censored = _merge_counts(results)
    return {"n": total, "curve": _build_curve(total, events, censored)}


def _build_curve(total, events, censored):
    times = sorted(set(events) | set(censored))
    at_risk = total
    surv = 1.0
    gw = 0.0  # Greenwood accumulator: sum d/(n(n-d))
    curve = []
    for t in times:
        d = events.get(t, 0)
        if d and at_risk:
            surv *= 1.0 - d / at_risk
            if at_risk > d:
                gw += d / (at_risk * (at_risk - d))
        # Greenwood variance + log-transformed 95% CI (the plain
        # S ± 1.96·se clips at [0,1]; exp-log keeps it inside)
        se = surv * gw ** 0.5
        if 0.0 < surv < 1.0 and gw > 0:
            z = 1.959963984540054
            theta = z * (gw ** 0.5) / abs(math.log(surv))
            lo, hi = surv ** math.exp(theta), surv ** math.exp(-theta)
        else:
            lo = hi = surv
        curve.append({"time": t, "at_risk": at_risk, "events": d,
                      "censored": censored.get(t, 0), "survival": surv,
                      "std_err": se, "ci95": [lo, hi]})
        at_risk -= d + censored.get(t, 0)
    return curve


@data(1)
def partial_event_counts(df, time_col: str, event_col: str, min_rows: int = 1):
    if len(df) < min_rows:
        raise PrivacyThresholdViolation(
            f"shard has {len(df)} rows < privacy threshold {min_rows}")
    sub = df[[time_col, event_col]].dropna()
    ev = sub[sub[event_col] == 1].groupby(time_col).size()
    ce = sub[sub[event_col] == 0].groupby(time_col).size()
    return {
        "n": int(len(sub)),
        "events": {float(t): int(v) for t, v in ev.items()},
        "censored": {float(t): int(v) for t, v in ce.items()},
    }


@algorithm_client
def central_logrank(client, time_col: str, event_col: str, group_col: str,
                    min_rows: int = 1):
    """Federated two-group log-rank test (matches scipy.stats.logrank).

    Stations share the same per-timepoint event/censor counts as the KM
    curve, split by group. At each pooled event time: O-E for group 1 is
    ``d1 - d*n1/n`` with hypergeometric variance
    ``d*(n1/n)*(n2/n)*(n-d)/(n-1)``; the statistic is
    ``(sum(O-E))^2 / sum(V)`` ~ chi2(1)."""
    from scipy import stats as _st

    org_ids = [org["id"] for org in client.organization.list()]
    task = client.task.create(
        input_={"method": "partial_event_counts_grouped",
                "kwargs": {"time_col": time_col, "event_col": event_col,
                           "group_col": group_col, "min_rows": min_rows}},
        organizations=org_ids,
        name="partial log-rank counts",
    )
    results = client.wait_for_results(task["id"])
    levels = sorted({lv for r in results for lv in r["groups"]})
    if len(levels) != 2:
        raise ValueError(f"log-rank here is two-group; got {levels}")
    ev = {lv: {} for lv in levels}
    ce = {lv: {} for lv in levels}
    n_at_risk = {lv: 0 for lv in levels}
    for r in results:
        for lv, g in r["groups"].items():
            n_at_risk[lv] += g["n"]
            for t, d in g["events"].items():
                ev[lv][float(t)] = ev[lv].get(float(t), 0) + d
            for t, c in g["censored"].items():
                ce[lv][float(t)] = ce[lv].get(float(t), 0) + c
    times = sorted(set().union(*(set(ev[lv]) | set(ce[lv]) for lv in levels)))
    a, b = levels
    o_minus_e = 0.0
    var = 0.0
    for t in times:
        d1, d2 = ev[a].get(t, 0), ev[b].get(t, 0)
        n1, n2 = n_at_risk[a], n_at_risk[b]
        d, n = d1 + d2, n1 + n2
        if d and n > 1:
            o_minus_e += d1 - d * n1 / n
            var += d * (n1 / n) * (n2 / n) * (n - d) / (n - 1)
        n_at_risk[a] -= d1 + ce[a].get(t, 0)
        n_at_risk[b] -= d2 + ce[b].get(t, 0)
    if var <= 0:
        raise ValueError("no events in common risk sets")
    chi2 = o_minus_e * o_minus_e / var
    return {"chi2": float(chi2),
            "p_value": float(_st.chi2.sf(chi2, 1)),
            "groups": levels}


@data(1)
def partial_event_counts_grouped(df, time_col: str, event_col: str,
                                 group_col: str, min_rows: int = 1):
    if len(df) < min_rows:
        raise PrivacyThresholdViolation(
            f"shard has {len(df)} rows < privacy threshold {min_rows}")
    sub = df[[time_col, event_col, group_col]].dropna()
    out = {}
    for lv, g in sub.groupby(group_col):
        ev = g[g[event_col] == 1].groupby(time_col).size()
        ce = g[g[event_col] == 0].groupby(time_col).size()
        out[str(lv)] = {
            "n": int(len(g)),
            "events": {float(t): int(v) for t, v in ev.items()},
            "censored": {float(t): int(v) for t, v in ce.items()},
        }
    return {"groups": out}
