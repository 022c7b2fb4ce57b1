"""Federated Pearson correlation matrix — pooled moment sums, no
row-level data leaves a station.

Mirrors the reference ecosystem's ``v6-correlation-matrix`` community
algorithm: each station reports n, per-column sums, and the cross-product
matrix over complete rows; the central part pools them into

    corr(a, b) = (n·Sab − Sa·Sb) / sqrt((n·Saa − Sa²)(n·Sbb − Sb²))

Complete-case analysis (rows with any NA in the requested columns are
dropped station-side, as the reference algorithm does). A ``min_rows``
privacy threshold makes stations refuse tiny shards.
"""
from vantage6_amd.algorithm import algorithm_client, data
from vantage6_amd.algorithm.exceptions import PrivacyThresholdViolation


@algorithm_client
def central_correlation(client, columns: list[str], min_rows: int = 1):
    org_ids = [org["id"] for org in client.organization.list()]
    task = client.task.create(
        input_={"method": "partial_correlation",
                "kwargs": {"columns": columns, "min_rows": min_rows}},
        organizations=org_ids,
        name="partial correlation moments",
    )
    results = client.wait_for_results(task["id"])
    k = len(columns)
    n = sum(r["n"] for r in results)
    if n < 2:
        return {"n": n, "columns": columns,
                "matrix": [[None] * k for _ in range(k)]}
    s = [sum(r["sums"][i] for r in results) for i in range(k)]
    xp = [[sum(r["cross"][i][j] for r in results) for j in range(k)]
          for i in range(k)]
    matrix = [[None] * k for _ in range(k)]
    for i in range(k):
        for j in range(k):
            num = n * xp[i][j] - s[i] * s[j]
            di = n * xp[i][i] - s[i] * s[i]
            dj = n * xp[j][j] - s[j] * s[j]
            if di <= 0 or dj <= 0:  # constant column
                matrix[i][j] = None if i != j else 1.0
            else:
                matrix[i][j] = num / (di * dj) ** 0.5
    return {"n": n, "columns": columns, "matrix": matrix}


@data(1)
def partial_correlation(df, columns: list[str], min_rows: int = 1):
    if len(df) < min_rows:
        raise PrivacyThresholdViolation(
            f"shard has {len(df)} rows < privacy threshold {min_rows}")
    sub = df[columns].dropna()
    k = len(columns)
    vals = [sub[c].astype(float) for c in columns]
    return {
        "n": int(len(sub)),
        "sums": [float(v.sum()) for v in vals],
        "cross": [[float((vals[i] * vals[j]).sum()) for j in range(k)]
                  for i in range(k)],
    }
