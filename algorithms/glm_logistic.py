"""Federated logistic regression via IRLS (Newton-Raphson).

Mirrors the reference ecosystem's ``v6-logistic-regression`` community
algorithm: unlike the SGD/FedAvg engine (``algorithms/fedavg.py``), this is
the classic *statistical* federation — each IRLS iteration the stations
share only their Gram-matrix contributions

    H_i = X_iᵀ W_i X_i          (p×p),
    g_i = X_iᵀ (y_i − μ_i)      (p,)

and the central part solves ``β ← β + (Σ H_i)⁻¹ Σ g_i``. Row-level data
never leaves a station; the shared aggregates are O(p²) regardless of n.
Converges to the exact pooled maximum-likelihood estimate (tested against
scikit-learn on the pooled data).
"""
import numpy as np

from vantage6_amd.algorithm import algorithm_client, data
from vantage6_amd.algorithm.exceptions import PrivacyThresholdViolation


@algorithm_client
def central_logistic_regression(
    client,
    feature_cols: list[str],
    target_col: str,
    max_iter: int = 25,
    tol: float = 1e-8,
    min_rows: int = 1,
):
    org_ids = [org["id"] for org in client.organization.list()]
    p = len(feature_cols) + 1  # + intercept
    beta = [0.0] * p
    history = []
    for it in range(max_iter):
        task = client.task.create(
            input_={"method": "partial_irls",
                    "kwargs": {"feature_cols": feature_cols,
                               "target_col": target_col, "beta": beta,
                               "min_rows": min_rows}},
            organizations=org_ids,
            name=f"IRLS iteration {it}",
        )
        results = client.wait_for_results(task["id"])
        H = np.zeros((p, p))
        g = np.zeros(p)
        n_total, loglik = 0, 0.0
        for res in results:
            H += np.asarray(res["hessian"])
            g += np.asarray(res["gradient"])
            n_total += res["n"]
            loglik += res["loglik"]
        # ridge jitter for numerical safety on collinear shards
        step = np.linalg.solve(H + 1e-9 * np.eye(p), g)
        beta = (np.asarray(beta) + step).tolist()
        history.append({"iteration": it, "loglik": loglik,
                        "max_step": float(np.abs(step).max())})
        if float(np.abs(step).max()) < tol:
            break
    from scipy import stats as _st

    cov = np.linalg.inv(H + 1e-9 * np.eye(p))
    se = np.sqrt(np.diag(cov))
    zv = np.asarray(beta) / se
    names = ["(intercept)"] + list(feature_cols)
    return {
        "coefficients": dict(zip(feature_cols, beta[1:])),
        "intercept": beta[0],
        "std_errors": dict(zip(names, se.tolist())),
        "p_values": dict(zip(names,
                             (2.0 * _st.norm.sf(np.abs(zv))).tolist())),
        "n": n_total,
        "iterations": len(history),
        "converged": history[-1]["max_step"] < tol,
        "history": history,
    }


@data(1)
def partial_irls(df, feature_cols: list[str], target_col: str,
                 beta: list[float], min_rows: int = 1):
    if len(df) < min_rows:
        raise PrivacyThresholdViolation(
            f"shard has {len(df)} rows < privacy threshold {min_rows}")
    sub = df[feature_cols + [target_col]].dropna()
    X = np.column_stack([np.ones(len(sub)), sub[feature_cols].to_numpy(float)])
    y = sub[target_col].to_numpy(float)
    eta = X @ np.asarray(beta)
    mu = 1.0 / (1.0 + np.exp(-np.clip(eta, -35, 35)))
    w = np.clip(mu * (1.0 - mu), 1e-12, None)
    return {
        "hessian": (X.T * w) @ X,
        "gradient": X.T @ (y - mu),
        "n": int(len(sub)),
        "loglik": float(np.sum(y * np.log(np.clip(mu, 1e-12, 1))
                               + (1 - y) * np.log(np.clip(1 - mu, 1e-12, 1)))),
    }


# ------------------------------------------------------------- general GLM
# Families as in the ecosystem's v6-glm: gaussian (identity), poisson
# (log), binomial (logit). Same federation: per-iteration Gram/score
# aggregates only.
_FAMILIES = ("gaussian", "poisson", "binomial")


@algorithm_client
def central_glm(
    client,
    feature_cols: list[str],
    target_col: str,
    family: str = "gaussian",
    max_iter: int = 25,
    tol: float = 1e-8,
    min_rows: int = 1,
):
    """Federated GLM via IRLS for gaussian/poisson/binomial families —
    converges to the pooled maximum-likelihood fit (gaussian closes in
    one iteration: IRLS with identity link IS the normal equation)."""
    if family not in _FAMILIES:
        raise ValueError(f"family must be one of {_FAMILIES}")
    org_ids = [org["id"] for org in client.organization.list()]
    p = len(feature_cols) + 1
    beta = [0.0] * p
    history = []
    n_total = 0
    for it in range(max_iter):
        task = client.task.create(
            input_={"method": "partial_glm_irls",
                    "kwargs": {"feature_cols": feature_cols,
                               "target_col": target_col, "beta": beta,
                               "family": family, "min_rows": min_rows}},
            organizations=org_ids,
            name=f"GLM {family} IRLS {it}",
        )
        results = client.wait_for_results(task["id"])
        H = np.zeros((p, p))
        g = np.zeros(p)
        n_total = sum(r["n"] for r in results)
        for res in results:
            H += np.asarray(res["hessian"])
            g += np.asarray(res["gradient"])
        step = np.linalg.solve(H + 1e-9 * np.eye(p), g)
        beta = (np.asarray(beta) + step).tolist()
        history.append({"iteration": it,
                        "max_step": float(np.abs(step).max())})
        if float(np.abs(step).max()) < tol:
            break
    # Wald inference from the converged information matrix: cov = phi*H^-1
    # (phi = 1 for poisson/binomial; gaussian dispersion = RSS/(n-p) from
    # the stations' pooled residual sum of squares)
    from scipy import stats as _st

    cov = np.linalg.inv(H + 1e-9 * np.eye(p))
    if family == "gaussian":
        rss = sum(r.get("rss", 0.0) for r in results)
        dof = max(n_total - p, 1)
        cov = cov * (rss / dof)
        se = np.sqrt(np.diag(cov))
        tv = np.asarray(beta) / se
        pv = 2.0 * _st.t.sf(np.abs(tv), dof)
    else:
        se = np.sqrt(np.diag(cov))
        tv = np.asarray(beta) / se
        pv = 2.0 * _st.norm.sf(np.abs(tv))
    names = ["(intercept)"] + list(feature_cols)
    return {
        "family": family,
        "coefficients": dict(zip(feature_cols, beta[1:])),
        "intercept": beta[0],
        "std_errors": dict(zip(names, se.tolist())),
        "p_values": dict(zip(names, pv.tolist())),
        "n": n_total,
        "iterations": len(history),
        "converged": history[-1]["max_step"] < tol,
    }


@data(1)
def partial_glm_irls(df, feature_cols: list[str], target_col: str,
                     beta: list[float], family: str = "gaussian",
                     min_rows: int = 1):
    if len(df) < min_rows:
        raise PrivacyThresholdViolation(
            f"shard has {len(df)} rows < privacy threshold {min_rows}")
    sub = df[feature_cols + [target_col]].dropna()
    X = np.column_stack([np.ones(len(sub)), sub[feature_cols].to_numpy(float)])
    y = sub[target_col].to_numpy(float)
    eta = X @ np.asarray(beta)
    if family == "gaussian":
        mu, w = eta, np.ones_like(eta)
    elif family == "poisson":
        mu = np.exp(np.clip(eta, -30, 30))
        w = np.clip(mu, 1e-12, None)
    else:  # binomial
        mu = 1.0 / (1.0 + np.exp(-np.clip(eta, -35, 35)))
        w = np.clip(mu * (1.0 - mu), 1e-12, None)
    return {
        "hessian": (X.T * w) @ X,
        "gradient": X.T @ (y - mu),
        "n": int(len(sub)),
        "rss": float(((y - mu) ** 2).sum()),
    }
