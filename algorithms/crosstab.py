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


@algorithm_client
def central_chi2(client, row_col: str, col_col: str, min_rows: int = 1):
    """Chi-square test of independence on the pooled two-way table
    (matches scipy.stats.chi2_contingency without continuity correction).

    Runs with station-side suppression DISABLED (suppress_below=0): a
    suppressed table would bias the statistic. The cells that reach the
    central part are the same aggregates a crosstab with no suppression
    shares, so the disclosure surface is identical to
    ``central_crosstab(suppress_below=0)``."""
    from scipy import stats as _st

    org_ids = [org["id"] for org in client.organization.list()]
    task = client.task.create(
        input_={"method": "partial_counts",
                "kwargs": {"row_col": row_col, "col_col": col_col,
                           "suppress_below": 0, "min_rows": min_rows}},
        organizations=org_ids, name="partial chi2 counts")
    results = client.wait_for_results(task["id"])
    totals: dict = {}
    for res in results:
        for key, n in res["counts"].items():
            totals[key] = totals.get(key, 0) + n
    rows = sorted({k.split("|")[0] for k in totals})
    cols = sorted({k.split("|")[1] for k in totals})
    if len(rows) < 2 or len(cols) < 2:
        raise ValueError("chi2 needs a table with >= 2 rows and columns")
    table = [[totals.get(f"{r}|{c}", 0) for c in cols] for r in rows]
    out = _st.chi2_contingency(table, correction=False)
    return {"chi2": float(out.statistic), "p_value": float(out.pvalue),
            "dof": int(out.dof), "rows": rows, "cols": cols,
            "table": table}
