"""Federated Cox proportional-hazards regression (WebDISCO-style).

The hallmark analysis of the reference's ecosystem (``v6-cox`` /
WebDISCO, Lu et al. 2015): maximize the Breslow partial likelihood with
Newton-Raphson where stations share only per-event-time RISK-SET
aggregates for the current β —

    S0_j = Σ_{k: t_k ≥ τ_j} exp(x_k β)                (scalar)
    S1_j = Σ_{k: t_k ≥ τ_j} x_k exp(x_k β)            (p,)
    S2_j = Σ_{k: t_k ≥ τ_j} x_k x_kᵀ exp(x_k β)       (p×p)

plus per-time event counts d_j and event-covariate sums s_j, all additive
across stations. Row-level data never leaves a station; the global event
TIME GRID is shared (same disclosure as Kaplan-Meier). Converges to the
pooled maximum-partial-likelihood estimate exactly.
"""
import numpy as np

from vantage6_amd.algorithm import algorithm_client, data
from vantage6_amd.algorithm.exceptions import PrivacyThresholdViolation


@algorithm_client
def central_coxph(
    client,
    feature_cols: list[str],
    time_col: str,
    event_col: str,
    max_iter: int = 30,
    tol: float = 1e-9,
    min_rows: int = 1,
):
    org_ids = [org["id"] for org in client.organization.list()]
    p = len(feature_cols)

    # round 0: global distinct event-time grid (union of station grids)
    t0 = client.task.create(
        input_={"method": "partial_event_times",
                "kwargs": {"time_col": time_col, "event_col": event_col,
                           "min_rows": min_rows}},
        organizations=org_ids, name="cox event times")
    grids = client.wait_for_results(t0["id"])
    times = sorted({t for g in grids for t in g["event_times"]})
    if not times:
        raise ValueError("no events in any shard")

    beta = np.zeros(p)
    history = []
    for it in range(max_iter):
        task = client.task.create(
            input_={"method": "partial_risk_sets",
                    "kwargs": {"feature_cols": feature_cols,
                               "time_col": time_col, "event_col": event_col,
                               "beta": beta.tolist(), "times": times,
                               "min_rows": min_rows}},
            organizations=org_ids, name=f"cox NR iteration {it}")
        results = client.wait_for_results(task["id"])
        T = len(times)
        S0 = np.zeros(T)
        S1 = np.zeros((T, p))
        S2 = np.zeros((T, p, p))
        d = np.zeros(T)
        s = np.zeros((T, p))
        for res in results:
            S0 += np.asarray(res["s0"])
            S1 += np.asarray(res["s1"])
            S2 += np.asarray(res["s2"])
            d += np.asarray(res["d"])
            s += np.asarray(res["s"])
        keep = d > 0
        grad = np.zeros(p)
        hess = np.zeros((p, p))
        loglik = 0.0
        for j in np.nonzero(keep)[0]:
            m1 = S1[j] / S0[j]
            grad += s[j] - d[j] * m1
            hess += d[j] * (S2[j] / S0[j] - np.outer(m1, m1))
            loglik += float(s[j] @ beta - d[j] * np.log(S0[j]))
        step = np.linalg.solve(hess + 1e-10 * np.eye(p), grad)
        beta = beta + step
        history.append({"iteration": it, "loglik": loglik,
                        "max_step": float(np.abs(step).max())})
        if float(np.abs(step).max()) < tol:
            break
    # Wald inference from the converged observed information (the same
    # Hessian the Newton step used): cov = H^-1; HR CI = exp(b ± 1.96 se)
    from scipy import stats as _st

    cov = np.linalg.inv(hess + 1e-10 * np.eye(p))
    se = np.sqrt(np.diag(cov))
    zv = beta / se
    pv = 2.0 * _st.norm.sf(np.abs(zv))
    ci_lo = np.exp(beta - 1.959963984540054 * se)
    ci_hi = np.exp(beta + 1.959963984540054 * se)
    return {
        "coefficients": dict(zip(feature_cols, beta.tolist())),
        "hazard_ratios": dict(zip(feature_cols, np.exp(beta).tolist())),
        "std_errors": dict(zip(feature_cols, se.tolist())),
        "p_values": dict(zip(feature_cols, pv.tolist())),
        "hr_ci95": {c: [lo, hi] for c, lo, hi
                    in zip(feature_cols, ci_lo.tolist(), ci_hi.tolist())},
        "iterations": len(history),
        "converged": history[-1]["max_step"] < tol,
        "n_event_times": len(times),
        "history": history,
    }


@data(1)
def partial_event_times(df, time_col: str, event_col: str, min_rows: int = 1):
    if len(df) < min_rows:
        raise PrivacyThresholdViolation(
            f"shard has {len(df)} rows < privacy threshold {min_rows}")
    sub = df[[time_col, event_col]].dropna()
    return {"event_times":
            sorted(float(t) for t in sub[sub[event_col] == 1][time_col].unique())}


@data(1)
def partial_risk_sets(df, feature_cols: list[str], time_col: str,
                      event_col: str, beta: list[float], times: list[float],
                      min_rows: int = 1):
    if len(df) < min_rows:
        raise PrivacyThresholdViolation(
            f"shard has {len(df)} rows < privacy threshold {min_rows}")
    sub = df[feature_cols + [time_col, event_col]].dropna()
    X = sub[feature_cols].to_numpy(float)
    t = sub[time_col].to_numpy(float)
    e = sub[event_col].to_numpy(float)
    r = np.exp(np.clip(X @ np.asarray(beta), -500, 500))
    T, p = len(times), X.shape[1]
    s0 = np.zeros(T)
    s1 = np.zeros((T, p))
    s2 = np.zeros((T, p, p))
    d = np.zeros(T)
    s = np.zeros((T, p))
    # risk-set sums via suffix accumulation over the global grid
    order = np.argsort(t)
    Xs, ts, es, rs = X[order], t[order], e[order], r[order]
    # for each grid time τ_j: rows with ts >= τ_j → suffix starting at
    # searchsorted(ts, τ_j, 'left')
    xr = Xs * rs[:, None]
    xxr = Xs[:, :, None] * Xs[:, None, :] * rs[:, None, None]
    suf0 = np.concatenate([np.cumsum(rs[::-1])[::-1], [0.0]])
    suf1 = np.concatenate([np.cumsum(xr[::-1], axis=0)[::-1],
                           np.zeros((1, p))])
    suf2 = np.concatenate([np.cumsum(xxr[::-1], axis=0)[::-1],
                           np.zeros((1, p, p))])
    for j, tau in enumerate(times):
        i = int(np.searchsorted(ts, tau, side="left"))
        s0[j] = suf0[i]
        s1[j] = suf1[i]
        s2[j] = suf2[i]
        at = (ts == tau) & (es == 1)
        d[j] = float(at.sum())
        if d[j]:
            s[j] = Xs[at].sum(axis=0)
    return {"s0": s0, "s1": s1, "s2": s2, "d": d, "s": s}
