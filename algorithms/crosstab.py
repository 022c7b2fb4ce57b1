"""Federated contingency table (crosstab) with disclosure control.

Mirrors the reference ecosystem's ``v6-crosstab``: per-station value
counts over one or two categorical columns are summed centrally. Cells
below ``suppress_below`` are OMITTED at the station (never shared — the
classic statistical-disclosure-control rule for frequency tables); the
central result reports how many cells each station withheld, and applies
the same threshold once more to the aggregate.
"""
from vantage6_amd.algorithm import algorithm_client, data
from vantage6_amd.algorithm.exceptions import PrivacyThresholdViolation


@algorithm_client
def central_crosstab(client, row_col: str, col_col: str | None = None,
                     suppress_below: int = 0, min_rows: int = 1):
    org_ids = [org["id"] for org in client.organization.list()]
    task = client.task.create(
        input_={"method": "partial_counts",
                "kwargs": {"row_col": row_col, "col_col": col_col,
                           "suppress_below": suppress_below,
                           "min_rows": min_rows}},
        organizations=org_ids, name="partial crosstab")
    results = client.wait_for_results(task["id"])
    totals: dict = {}
    withheld = 0
    for res in results:
        withheld += res["suppressed_cells"]
        for key, n in res["counts"].items():
            totals[key] = totals.get(key, 0) + n
    if suppress_below > 1:
        totals = {k: (v if v >= suppress_below else f"<{suppress_below}")
                  for k, v in totals.items()}
    return {"counts": totals, "n_stations": len(results),
            "suppressed_cells": withheld}


@data(1)
def partial_counts(df, row_col: str, col_col: str | None = None,
                   suppress_below: int = 0, min_rows: int = 1):
    if len(df) < min_rows:
        raise PrivacyThresholdViolation(
            f"shard has {len(df)} rows < privacy threshold {min_rows}")
    cols = [row_col] if col_col is None else [row_col, col_col]
    sub = df[cols].dropna()
    grouped = sub.groupby(cols).size()
    counts = {}
    suppressed = 0
    for key, n in grouped.items():
        if suppress_below > 1 and int(n) < suppress_below:
            suppressed += 1  # small cell: never leaves the station
            continue
        name = str(key) if col_col is None else f"{key[0]}|{key[1]}"
        counts[name] = int(n)
    return {"counts": counts, "suppressed_cells": suppressed}
