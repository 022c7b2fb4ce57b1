"""Federated Kaplan-Meier survival analysis.

The signature analysis for the reference's cancer-registry setting
(mirrors the ``v6-kaplan-meier`` community algorithm): stations share only
per-timepoint event/censoring COUNTS — never patient rows — and the
central function merges them into the product-limit estimator
``S(t) = prod_{t_i <= t} (1 - d_i / n_i)``.

Columns: ``time_col`` (follow-up time) and ``event_col`` (1 = event,
0 = censored). ``min_rows`` guards small shards (disclosure check).
"""
from vantage6_amd.algorithm import algorithm_client, data
from vantage6_amd.algorithm.exceptions import PrivacyThresholdViolation


@algorithm_client
def central_kaplan_meier(client, time_col: str, event_col: str,
                         min_rows: int = 1):
    org_ids = [org["id"] for org in client.organization.list()]
    task = client.task.create(
        input_={"method": "partial_event_counts",
                "kwargs": {"time_col": time_col, "event_col": event_col,
                           "min_rows": min_rows}},
        organizations=org_ids,
        name="partial KM event counts",
    )
    results = client.wait_for_results(task["id"])
    events: dict[float, int] = {}
    censored: dict[float, int] = {}
    total = 0
    for res in results:
        total += res["n"]
        for t, d in res["events"].items():
            events[float(t)] = events.get(float(t), 0) + d
        for t, c in res["censored"].items():
            censored[float(t)] = censored.get(float(t), 0) + c
    times = sorted(set(events) | set(censored))
    at_risk = total
    surv = 1.0
    curve = []
    for t in times:
        d = events.get(t, 0)
        if d and at_risk:
            surv *= 1.0 - d / at_risk
        curve.append({"time": t, "at_risk": at_risk, "events": d,
                      "censored": censored.get(t, 0), "survival": surv})
        at_risk -= d + censored.get(t, 0)
    return {"n": total, "curve": curve}


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
