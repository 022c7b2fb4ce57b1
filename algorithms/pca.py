"""Federated PCA — pooled covariance from per-station moment sums,
central eigendecomposition. No row-level data leaves a station.

Mirrors the reference ecosystem's federated-PCA community algorithms:
each station reports n, per-column sums and the raw cross-product matrix
over complete rows; the central part assembles the pooled covariance

    cov = (XᵀX − n·x̄x̄ᵀ) / (n − 1)

and eigendecomposes it. Returns components (rows = PCs, sign-normalised
so each component's largest-magnitude loading is positive, like
scikit-learn's svd_flip), explained variance and explained-variance
ratios. A ``min_rows`` privacy threshold guards tiny shards.
"""
import numpy as np

from vantage6_amd.algorithm import algorithm_client, data
from vantage6_amd.algorithm.exceptions import PrivacyThresholdViolation


@algorithm_client
def central_pca(client, columns: list[str], n_components: int | None = None,
                min_rows: int = 1):
    org_ids = [org["id"] for org in client.organization.list()]
    task = client.task.create(
        input_={"method": "partial_pca_moments",
                "kwargs": {"columns": columns, "min_rows": min_rows}},
        organizations=org_ids,
        name="partial PCA moments",
    )
    results = client.wait_for_results(task["id"])
    k = len(columns)
    n = sum(r["n"] for r in results)
    if n < 2:
        return {"n": n, "columns": columns, "components": None,
                "explained_variance": None, "explained_variance_ratio": None}
    s = np.array([sum(r["sums"][i] for r in results) for i in range(k)])
    xtx = np.array([[sum(r["cross"][i][j] for r in results)
                     for j in range(k)] for i in range(k)])
    mean = s / n
    cov = (xtx - n * np.outer(mean, mean)) / (n - 1)
    evals, evecs = np.linalg.eigh(cov)          # ascending
    order = np.argsort(evals)[::-1]
    evals = np.maximum(evals[order], 0.0)
    comps = evecs[:, order].T                   # rows = components
    # sign convention: largest-|loading| positive (sklearn svd_flip)
    for row in comps:
        piv = np.argmax(np.abs(row))
        if row[piv] < 0:
            row *= -1.0
    if n_components is not None:
        comps, evals = comps[:n_components], evals[:n_components]
    total = float(np.trace(cov))
    return {
        "n": n,
        "columns": columns,
        "mean": mean.tolist(),
        "components": comps.tolist(),
        "explained_variance": evals.tolist(),
        "explained_variance_ratio":
            (evals / total).tolist() if total > 0 else [0.0] * len(evals),
    }


@data(1)
def partial_pca_moments(df, columns: list[str], min_rows: int = 1):
    if len(df) < min_rows:
        raise PrivacyThresholdViolation(
            f"shard has {len(df)} rows < privacy threshold {min_rows}")
    sub = df[columns].dropna()
    x = sub.to_numpy(dtype=float)
    return {
        "n": int(len(sub)),
        "sums": x.sum(axis=0).tolist(),
        "cross": (x.T @ x).tolist(),
    }
