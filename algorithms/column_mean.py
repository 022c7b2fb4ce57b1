"""Federated column mean — the plumbing-check algorithm (BASELINE config[0]).

Mirrors the canonical vantage6 "v6-average" example algorithm: a *central*
function fans out *partial* computations to every organization and combines
the partial sums/counts into the global mean; the server (and every other
station) only ever sees aggregates.
"""
from vantage6_amd.algorithm import algorithm_client, data
from vantage6_amd.algorithm.exceptions import PrivacyThresholdViolation


@algorithm_client
def central_mean(client, columns: list[str], min_rows: int = 1):
    """Runs on one station: fan out partials, weighted-combine."""
    org_ids = [org["id"] for org in client.organization.list()]
    task = client.task.create(
        input_={"method": "partial_mean",
                "kwargs": {"columns": columns, "min_rows": min_rows}},
        organizations=org_ids,
        name="partial column mean",
    )
    results = client.wait_for_results(task["id"])
    sums = {c: 0.0 for c in columns}
    counts = {c: 0 for c in columns}
    for res in results:
        for c in columns:
            sums[c] += res["sums"][c]
            counts[c] += res["counts"][c]
    return {c: (sums[c] / counts[c] if counts[c] else None) for c in columns}


@data(1)
def partial_mean(df, columns: list[str], min_rows: int = 1):
    """Runs on every station against its local shard."""
    if len(df) < min_rows:
        raise PrivacyThresholdViolation(
            f"shard has {len(df)} rows < privacy threshold {min_rows}")
    return {
        "sums": {c: float(df[c].sum()) for c in columns},
        "counts": {c: int(df[c].count()) for c in columns},
    }
