"""End-to-end plumbing check — BASELINE config[0]:

federated column-mean on synthetic CSV shards, 2 CPU stations, driven
through the client API exactly as a researcher would (SURVEY.md §4.1 call
stack: client → server → station → algorithm → subtasks → results).
"""
import numpy as np
import pandas as pd
import pytest

from vantage6_amd.cli.dev import DevNetwork
from vantage6_amd.common.config import DatabaseConfig


@pytest.fixture
def csv_shards(tmp_path):
    rng = np.random.default_rng(0)
    paths, frames = [], []
    for i in range(2):
        df = pd.DataFrame(
            {"age": rng.integers(20, 90, 50 + 10 * i), "bmi": rng.normal(25, 4, 50 + 10 * i)}
        )
        p = tmp_path / f"shard{i}.csv"
        df.to_csv(p, index=False)
        paths.append(p)
        frames.append(df)
    return paths, frames


def expected_means(frames):
    all_df = pd.concat(frames)
    return {c: float(all_df[c].mean()) for c in ("age", "bmi")}


def run_column_mean(net, frames):
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],  # central runs on station 0
            name="column-mean",
            image="algorithms.column_mean",
            input_={"method": "central_mean", "kwargs": {"columns": ["age", "bmi"]}},
            databases=[{"label": "default"}],
        )
        results = net.client.wait_for_results(task["id"], interval=0.05, timeout=60)
    assert len(results) == 1
    got = results[0]
    want = expected_means(frames)
    for c in ("age", "bmi"):
        assert got[c] == pytest.approx(want[c], rel=1e-9)


def test_column_mean_unencrypted(csv_shards):
    paths, frames = csv_shards
    net = DevNetwork(
        n_stations=2,
        databases=[[DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths],
    )
    run_column_mean(net, frames)


def test_column_mean_encrypted(csv_shards):
    """Same flow with end-to-end hybrid RSA encryption enabled."""
    paths, frames = csv_shards
    net = DevNetwork(
        n_stations=2,
        encrypted=True,
        key_bits=1024,  # fast keygen for tests; production default is 2048
        databases=[[DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths],
    )
    run_column_mean(net, frames)


def test_partial_direct_with_mock_client(csv_shards):
    """Algorithm unit-testing via MockAlgorithmClient (reference test style)."""
    paths, frames = csv_shards
    from vantage6_amd.algorithm import MockAlgorithmClient

    mock = MockAlgorithmClient(
        datasets=[[{"label": "default", "database": str(p), "type": "csv"}] for p in paths],
        module="algorithms.column_mean",
    )
    task = mock.task.create(
        input_={"method": "central_mean", "kwargs": {"columns": ["age", "bmi"]}},
        organizations=[1],
    )
    got = mock.result.from_task(task["id"])[0]
    want = expected_means(frames)
    for c in ("age", "bmi"):
        assert got[c] == pytest.approx(want[c], rel=1e-9)


def test_preprocessing_applied(csv_shards):
    """Task database specs can carry declarative preprocessing steps."""
    paths, frames = csv_shards
    net = DevNetwork(
        n_stations=2,
        databases=[[DatabaseConfig(label="default", uri=str(p), type="csv")]
                   for p in paths],
    )
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="prep mean", image="algorithms.column_mean",
            input_={"method": "central_mean", "kwargs": {"columns": ["age"]}},
            databases=[{"label": "default", "preprocessing": [
                {"function": "filter_range",
                 "parameters": {"column": "age", "min": 50}},
            ]}],
        )
        # NOTE: subtasks created by the central don't inherit database specs
        # (same as the reference), so test the partial directly
        task2 = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=net.org_ids,
            name="prep partial", image="algorithms.column_mean",
            input_={"method": "partial_mean", "kwargs": {"columns": ["age"]}},
            databases=[{"label": "default", "preprocessing": [
                {"function": "filter_range",
                 "parameters": {"column": "age", "min": 50}},
            ]}],
        )
        results = net.client.wait_for_results(task2["id"], timeout=60)
    import pandas as pd

    for res, df in zip(results, frames):
        expect = df[df["age"] >= 50]
        assert res["counts"]["age"] == len(expect)
        assert res["sums"]["age"] == pytest.approx(float(expect["age"].sum()))


def test_preprocessing_mock_client(csv_shards):
    paths, frames = csv_shards
    from vantage6_amd.algorithm import MockAlgorithmClient

    mock = MockAlgorithmClient(
        datasets=[[{"label": "default", "database": str(p), "type": "csv",
                    "preprocessing": [{"function": "select_columns",
                                       "parameters": {"columns": ["bmi"]}}]}]
                  for p in paths],
        module="algorithms.column_mean",
    )
    task = mock.task.create(
        input_={"method": "partial_mean", "kwargs": {"columns": ["bmi"]}},
        organizations=[1],
    )
    out = mock.result.from_task(task["id"])[0]
    assert out["counts"]["bmi"] == len(frames[0])


def test_summary_statistics(tmp_path):
    """Federated summary stats match pandas on the pooled data; a privacy
    threshold above the shard size makes the task fail cleanly."""
    import numpy as np
    import pandas as pd

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(11)
    frames, paths = [], []
    for i in range(2):
        df = pd.DataFrame({"v": rng.normal(5, 3, 30 + 10 * i)})
        df.loc[df.index[:3], "v"] = np.nan
        p = tmp_path / f"sum{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)
    pooled = pd.concat(frames)["v"]
    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="summary", image="algorithms.summary_stats",
            input_={"method": "central_summary",
                    "kwargs": {"columns": ["v"]}})
        (res,) = net.client.wait_for_results(task["id"])
        st = res["v"]
        assert st["n"] == int(pooled.count())
        assert st["na"] == int(pooled.isna().sum())
        assert st["mean"] == pytest.approx(pooled.mean(), rel=1e-9)
        assert st["std"] == pytest.approx(pooled.std(), rel=1e-9)
        assert st["min"] == pytest.approx(pooled.min())
        assert st["max"] == pytest.approx(pooled.max())
        # privacy threshold: shards are 30/40 rows — 1000 must fail
        t2 = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="summary-blocked", image="algorithms.summary_stats",
            input_={"method": "central_summary",
                    "kwargs": {"columns": ["v"], "min_rows": 1000}})
        with pytest.raises(RuntimeError, match="PrivacyThresholdViolation"):
            net.client.wait_for_results(t2["id"])


def test_kaplan_meier(tmp_path):
    """Federated KM equals the pooled product-limit estimator."""
    import numpy as np
    import pandas as pd

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(21)
    frames, paths = [], []
    for i in range(2):
        df = pd.DataFrame({
            "t": rng.integers(1, 15, 50).astype(float),
            "e": rng.integers(0, 2, 50),
        })
        p = tmp_path / f"km{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)
    pooled = pd.concat(frames, ignore_index=True)

    # plain pooled product-limit reference
    at_risk = len(pooled)
    surv_ref = {}
    s = 1.0
    for t in sorted(pooled["t"].unique()):
        d = int(((pooled["t"] == t) & (pooled["e"] == 1)).sum())
        c = int(((pooled["t"] == t) & (pooled["e"] == 0)).sum())
        if d:
            s *= 1.0 - d / at_risk
        surv_ref[t] = s
        at_risk -= d + c

    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="km", image="algorithms.kaplan_meier",
            input_={"method": "central_kaplan_meier",
                    "kwargs": {"time_col": "t", "event_col": "e"}})
        (res,) = net.client.wait_for_results(task["id"])
    assert res["n"] == len(pooled)
    for row in res["curve"]:
        assert row["survival"] == pytest.approx(surv_ref[row["time"]], rel=1e-12)


def test_federated_logistic_regression_matches_sklearn(tmp_path):
    """Federated IRLS equals the pooled sklearn (unpenalized) fit."""
    import numpy as np
    import pandas as pd
    from sklearn.linear_model import LogisticRegression

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(31)
    frames, paths = [], []
    true_beta = np.array([0.8, -1.2])
    for i in range(2):
        X = rng.normal(0, 1, (120, 2))
        logits = 0.3 + X @ true_beta
        y = (rng.random(120) < 1 / (1 + np.exp(-logits))).astype(int)
        df = pd.DataFrame({"a": X[:, 0], "b": X[:, 1], "y": y})
        p = tmp_path / f"glm{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)
    pooled = pd.concat(frames, ignore_index=True)
    sk = LogisticRegression(penalty=None, tol=1e-10, max_iter=500)
    sk.fit(pooled[["a", "b"]], pooled["y"])

    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="glm", image="algorithms.glm_logistic",
            input_={"method": "central_logistic_regression",
                    "kwargs": {"feature_cols": ["a", "b"],
                               "target_col": "y"}})
        (res,) = net.client.wait_for_results(task["id"], timeout=60)
    assert res["converged"] and res["n"] == 240
    assert res["intercept"] == pytest.approx(sk.intercept_[0], abs=1e-5)
    assert res["coefficients"]["a"] == pytest.approx(sk.coef_[0][0], abs=1e-5)
    assert res["coefficients"]["b"] == pytest.approx(sk.coef_[0][1], abs=1e-5)


def test_federated_coxph_matches_pooled_newton(tmp_path):
    """Federated Cox (WebDISCO) equals an independent pooled Breslow
    Newton fit to 1e-6, and recovers the simulated hazard directionally."""
    import numpy as np
    import pandas as pd

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(41)
    true_beta = np.array([0.7, -0.5])
    frames, paths = [], []
    for i in range(2):
        X = rng.normal(0, 1, (150, 2))
        # exponential survival with rate exp(X beta); admin censoring
        t = rng.exponential(1.0 / np.exp(X @ true_beta))
        c = rng.exponential(2.0, 150)
        obs = np.minimum(t, c)
        df = pd.DataFrame({"a": X[:, 0], "b": X[:, 1],
                           "t": np.round(obs, 3),
                           "e": (t <= c).astype(int)})
        p = tmp_path / f"cox{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)
    pooled = pd.concat(frames, ignore_index=True)

    # independent pooled Breslow Newton (plain loops, no suffix tricks)
    def pooled_cox(df, iters=40):
        X = df[["a", "b"]].to_numpy(float)
        t = df["t"].to_numpy(float)
        e = df["e"].to_numpy(int)
        times = np.unique(t[e == 1])
        beta = np.zeros(2)
        for _ in range(iters):
            r = np.exp(X @ beta)
            g = np.zeros(2)
            H = np.zeros((2, 2))
            for tau in times:
                risk = t >= tau
                ev = (t == tau) & (e == 1)
                s0 = r[risk].sum()
                s1 = (X[risk] * r[risk, None]).sum(0)
                s2 = (X[risk, :, None] * X[risk, None, :]
                      * r[risk, None, None]).sum(0)
                dj = ev.sum()
                g += X[ev].sum(0) - dj * s1 / s0
                H += dj * (s2 / s0 - np.outer(s1 / s0, s1 / s0))
            step = np.linalg.solve(H, g)
            beta = beta + step
            if np.abs(step).max() < 1e-12:
                break
        return beta

    ref_beta = pooled_cox(pooled)
    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="cox", image="algorithms.coxph",
            input_={"method": "central_coxph",
                    "kwargs": {"feature_cols": ["a", "b"], "time_col": "t",
                               "event_col": "e"}})
        (res,) = net.client.wait_for_results(task["id"], timeout=120)
    assert res["converged"]
    assert res["coefficients"]["a"] == pytest.approx(ref_beta[0], abs=1e-6)
    assert res["coefficients"]["b"] == pytest.approx(ref_beta[1], abs=1e-6)
    # directional recovery of the simulated effects
    assert res["coefficients"]["a"] > 0.3 and res["coefficients"]["b"] < -0.2
    # Wald inference sanity: positive SEs, CI brackets the HR, strong
    # true effects are significant
    for c in ("a", "b"):
        se = res["std_errors"][c]
        lo, hi = res["hr_ci95"][c]
        assert se > 0 and lo < res["hazard_ratios"][c] < hi
        assert res["p_values"][c] < 0.05


def test_federated_crosstab(tmp_path):
    import pandas as pd

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    frames, paths = [], []
    for i, rows in enumerate([[("m", "yes")] * 6 + [("f", "no")] * 3,
                              [("m", "yes")] * 2 + [("f", "yes")] * 1]):
        df = pd.DataFrame(rows, columns=["sex", "treated"])
        p = tmp_path / f"ct{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)
    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="ct", image="algorithms.crosstab",
            input_={"method": "central_crosstab",
                    "kwargs": {"row_col": "sex", "col_col": "treated",
                               "suppress_below": 2}})
        (res,) = net.client.wait_for_results(task["id"], timeout=30)
    assert res["counts"]["m|yes"] == 8
    assert res["counts"]["f|no"] == 3
    # the 1-count f|yes cell never left station 1
    assert "f|yes" not in res["counts"]
    assert res["suppressed_cells"] == 1


def test_multi_database_task(tmp_path):
    """A task naming two database labels gets BOTH shards injected, in the
    requested order (@data(2) + USER_REQUESTED_DATABASE_LABELS contract)."""
    import pandas as pd

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    a = tmp_path / "a.csv"
    b = tmp_path / "b.csv"
    pd.DataFrame({"v": [1.0, 2.0]}).to_csv(a, index=False)
    pd.DataFrame({"v": [10.0, 20.0, 30.0]}).to_csv(b, index=False)
    net = DevNetwork(n_stations=1, databases=[[
        DatabaseConfig(label="alpha", uri=str(a), type="csv"),
        DatabaseConfig(label="beta", uri=str(b), type="csv"),
    ]])
    # ad-hoc two-db algorithm: register a module object on the fly
    import sys
    import types

    from vantage6_amd.algorithm import data

    mod = types.ModuleType("algos_twodb")

    @data(2)
    def row_counts(df1, df2):
        return {"n1": int(len(df1)), "n2": int(len(df2)),
                "sum2": float(df2["v"].sum())}

    mod.row_counts = row_counts
    sys.modules["algos_twodb"] = mod
    try:
        with net:
            task = net.client.task.create(
                collaboration=net.collaboration["id"],
                organizations=[net.org_ids[0]],
                name="twodb", image="algos_twodb",
                input_={"method": "row_counts"},
                databases=[{"label": "alpha"}, {"label": "beta"}])
            (res,) = net.client.wait_for_results(task["id"], timeout=30)
        assert res == {"n1": 2, "n2": 3, "sum2": 60.0}
    finally:
        del sys.modules["algos_twodb"]


def test_federated_correlation_matrix(tmp_path):
    """Federated Pearson correlation equals pandas .corr() on the pooled
    complete-case data."""
    import numpy as np
    import pandas as pd

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(23)
    frames, paths = [], []
    for i in range(3):
        n = 40 + 15 * i
        a = rng.normal(0, 1, n)
        df = pd.DataFrame({
            "a": a,
            "b": 0.6 * a + rng.normal(0, 0.8, n),
            "c": rng.normal(4, 2, n),
        })
        df.loc[df.index[:2], "b"] = np.nan  # complete-case handling
        p = tmp_path / f"corr{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)
    pooled = pd.concat(frames)[["a", "b", "c"]].dropna()
    ref = pooled.corr().to_numpy()
    net = DevNetwork(n_stations=3, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="corr", image="algorithms.correlation",
            input_={"method": "central_correlation",
                    "kwargs": {"columns": ["a", "b", "c"]}})
        (res,) = net.client.wait_for_results(task["id"])
    assert res["n"] == len(pooled)
    got = np.array(res["matrix"], dtype=float)
    assert np.allclose(got, ref, atol=1e-9)


def test_federated_quantiles(tmp_path):
    """Bisection quantiles equal numpy's method='lower' order statistics
    on the pooled data; only counts ever leave a station."""
    import numpy as np
    import pandas as pd

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(31)
    vals, paths = [], []
    for i in range(2):
        v = rng.normal(10, 4, 60 + 20 * i)
        p = tmp_path / f"q{i}.csv"
        pd.DataFrame({"v": v}).to_csv(p, index=False)
        vals.append(v)
        paths.append(p)
    pooled = np.concatenate(vals)
    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    qs = [0.1, 0.5, 0.9]
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="quantiles", image="algorithms.quantiles",
            input_={"method": "central_quantiles",
                    "kwargs": {"column": "v", "q": qs}})
        (res,) = net.client.wait_for_results(task["id"])
    assert res["n"] == len(pooled)
    span = pooled.max() - pooled.min()
    for frac in qs:
        ref = np.quantile(pooled, frac, method="lower")
        assert abs(res["quantiles"][str(frac)] - ref) <= 2e-6 * span + 1e-12, frac


def test_federated_pca_matches_sklearn(tmp_path):
    """Federated PCA (pooled covariance + central eigh) equals sklearn
    PCA fit on the pooled rows."""
    import numpy as np
    import pandas as pd
    from sklearn.decomposition import PCA

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(47)
    frames, paths = [], []
    base = rng.normal(0, 1, (0, 4))
    for i in range(3):
        n = 50 + 10 * i
        z = rng.normal(0, 1, (n, 2))
        x = np.column_stack([
            z[:, 0] * 3.0,
            z[:, 0] * 1.5 + z[:, 1],
            rng.normal(0, 0.5, n),
            z[:, 1] * 2.0 + rng.normal(0, 0.2, n),
        ])
        df = pd.DataFrame(x, columns=list("abcd"))
        p = tmp_path / f"pca{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)
    pooled = pd.concat(frames).to_numpy()
    ref = PCA(n_components=3).fit(pooled)
    net = DevNetwork(n_stations=3, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="pca", image="algorithms.pca",
            input_={"method": "central_pca",
                    "kwargs": {"columns": list("abcd"), "n_components": 3}})
        (res,) = net.client.wait_for_results(task["id"])
    assert res["n"] == len(pooled)
    assert np.allclose(res["explained_variance"],
                       ref.explained_variance_, rtol=1e-8)
    assert np.allclose(res["explained_variance_ratio"],
                       ref.explained_variance_ratio_, rtol=1e-8)
    got = np.array(res["components"])
    # same subspace and same svd_flip sign convention
    assert np.allclose(np.abs(got), np.abs(ref.components_), atol=1e-8)
    for i in range(3):
        assert np.allclose(got[i], ref.components_[i], atol=1e-8) or \
            np.allclose(got[i], -ref.components_[i], atol=1e-8)


def test_federated_ttest_matches_scipy(tmp_path):
    """Federated Welch and Student t-tests equal scipy on pooled rows."""
    import numpy as np
    import pandas as pd
    from scipy import stats

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(59)
    frames, paths = [], []
    for i in range(2):
        n = 45 + 20 * i
        df = pd.DataFrame({
            "y": np.concatenate([rng.normal(5.0, 2.0, n),
                                 rng.normal(5.8, 3.0, n)]),
            "arm": ["ctl"] * n + ["trt"] * n,
        })
        p = tmp_path / f"tt{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)
    pooled = pd.concat(frames)
    ctl = pooled[pooled.arm == "ctl"].y.to_numpy()
    trt = pooled[pooled.arm == "trt"].y.to_numpy()
    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        for equal_var in (False, True):
            task = net.client.task.create(
                collaboration=net.collaboration["id"],
                organizations=[net.org_ids[0]],
                name="ttest", image="algorithms.ttest",
                input_={"method": "central_ttest",
                        "kwargs": {"value": "y", "group": "arm",
                                   "equal_var": equal_var}})
            (res,) = net.client.wait_for_results(task["id"])
            ref = stats.ttest_ind(ctl, trt, equal_var=equal_var)
            assert res["t"] == pytest.approx(ref.statistic, rel=1e-10)
            assert res["p_value"] == pytest.approx(ref.pvalue, rel=1e-9)
            assert res["groups"]["ctl"]["n"] == len(ctl)


def test_federated_anova_matches_scipy(tmp_path):
    """Federated one-way ANOVA equals scipy.stats.f_oneway on pooled rows."""
    import numpy as np
    import pandas as pd
    from scipy import stats

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(67)
    frames, paths = [], []
    for i in range(2):
        n = 30 + 10 * i
        df = pd.DataFrame({
            "y": np.concatenate([rng.normal(5, 2, n), rng.normal(6, 2, n),
                                 rng.normal(5.5, 2.5, n)]),
            "arm": ["a"] * n + ["b"] * n + ["c"] * n,
        })
        p = tmp_path / f"an{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)
    pooled = pd.concat(frames)
    samples = [pooled[pooled.arm == g].y.to_numpy() for g in ("a", "b", "c")]
    ref = stats.f_oneway(*samples)
    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="anova", image="algorithms.ttest",
            input_={"method": "central_anova",
                    "kwargs": {"value": "y", "group": "arm"}})
        (res,) = net.client.wait_for_results(task["id"])
    assert res["F"] == pytest.approx(ref.statistic, rel=1e-9)
    assert res["p_value"] == pytest.approx(ref.pvalue, rel=1e-8)
    assert res["df_between"] == 2


def test_federated_chi2_matches_scipy(tmp_path):
    """Chi-square independence on the pooled contingency table equals
    scipy.chi2_contingency."""
    import numpy as np
    import pandas as pd
    from scipy import stats

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(71)
    frames, paths = [], []
    for i in range(2):
        n = 120 + 40 * i
        g = rng.choice(["x", "y"], n)
        o = np.where((g == "x") & (rng.random(n) < 0.6), "hi", "lo")
        df = pd.DataFrame({"g": g, "o": o})
        p = tmp_path / f"chi{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)
    pooled = pd.concat(frames)
    ct = pd.crosstab(pooled.g, pooled.o)
    ref = stats.chi2_contingency(ct.to_numpy(), correction=False)
    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="chi2", image="algorithms.crosstab",
            input_={"method": "central_chi2",
                    "kwargs": {"row_col": "g", "col_col": "o"}})
        (res,) = net.client.wait_for_results(task["id"])
    assert res["chi2"] == pytest.approx(ref.statistic, rel=1e-10)
    assert res["p_value"] == pytest.approx(ref.pvalue, rel=1e-9)
    assert np.array(res["table"]).sum() == len(pooled)


def test_federated_logrank_matches_scipy(tmp_path):
    """Two-group log-rank from pooled per-time counts equals
    scipy.stats.logrank on the pooled rows."""
    import numpy as np
    import pandas as pd
    from scipy import stats

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(83)
    frames, paths = [], []
    for i in range(2):
        n = 60 + 20 * i
        t_a = rng.exponential(10, n).round(1) + 0.1
        t_b = rng.exponential(14, n).round(1) + 0.1
        cens = rng.random(2 * n) < 0.25
        df = pd.DataFrame({
            "t": np.concatenate([t_a, t_b]),
            "e": (~cens).astype(int),
            "arm": ["a"] * n + ["b"] * n,
        })
        p = tmp_path / f"lr{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)
    pooled = pd.concat(frames)

    def surv(arm):
        sub = pooled[pooled.arm == arm]
        return stats.CensoredData.right_censored(
            sub.t.to_numpy(), (sub.e == 0).to_numpy())

    ref = stats.logrank(surv("a"), surv("b"))
    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="logrank", image="algorithms.kaplan_meier",
            input_={"method": "central_logrank",
                    "kwargs": {"time_col": "t", "event_col": "e",
                               "group_col": "arm"}})
        (res,) = net.client.wait_for_results(task["id"])
    # scipy's statistic is the signed sqrt(chi2) (normal form)
    assert res["chi2"] == pytest.approx(ref.statistic ** 2, rel=1e-9)
    assert res["p_value"] == pytest.approx(ref.pvalue, rel=1e-7)


def test_federated_glm_families(tmp_path):
    """Gaussian GLM equals the pooled OLS normal equation exactly;
    Poisson GLM matches sklearn's PoissonRegressor (alpha=0) MLE."""
    import numpy as np
    import pandas as pd
    from sklearn.linear_model import PoissonRegressor

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(91)
    frames, paths = [], []
    for i in range(2):
        n = 80 + 20 * i
        x1 = rng.normal(0, 1, n)
        x2 = rng.normal(0, 1, n)
        ygauss = 2.0 + 1.5 * x1 - 0.7 * x2 + rng.normal(0, 0.3, n)
        ypois = rng.poisson(np.exp(0.4 + 0.5 * x1 - 0.3 * x2))
        df = pd.DataFrame({"x1": x1, "x2": x2, "yg": ygauss, "yp": ypois})
        p = tmp_path / f"glm{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)
    pooled = pd.concat(frames)
    X = np.column_stack([np.ones(len(pooled)), pooled.x1, pooled.x2])
    ols = np.linalg.solve(X.T @ X, X.T @ pooled.yg.to_numpy())
    sk = PoissonRegressor(alpha=0.0, max_iter=2000, tol=1e-12).fit(
        pooled[["x1", "x2"]], pooled.yp)
    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        def fit(family, target):
            task = net.client.task.create(
                collaboration=net.collaboration["id"],
                organizations=[net.org_ids[0]],
                name=f"glm-{family}", image="algorithms.glm_logistic",
                input_={"method": "central_glm",
                        "kwargs": {"feature_cols": ["x1", "x2"],
                                   "target_col": target, "family": family}})
            (res,) = net.client.wait_for_results(task["id"])
            return res

        g = fit("gaussian", "yg")
        assert g["converged"] and g["n"] == len(pooled)
        assert g["intercept"] == pytest.approx(ols[0], abs=1e-8)
        assert g["coefficients"]["x1"] == pytest.approx(ols[1], abs=1e-8)
        assert g["coefficients"]["x2"] == pytest.approx(ols[2], abs=1e-8)
        p_ = fit("poisson", "yp")
        assert p_["converged"]
        assert p_["intercept"] == pytest.approx(sk.intercept_, abs=2e-5)
        assert p_["coefficients"]["x1"] == pytest.approx(sk.coef_[0], abs=2e-5)
        assert p_["coefficients"]["x2"] == pytest.approx(sk.coef_[1], abs=2e-5)


def test_federated_glm_standard_errors(tmp_path):
    """Gaussian GLM Wald SEs equal the OLS closed form
    sqrt(diag((X'X)^-1) * RSS/(n-p))."""
    import numpy as np
    import pandas as pd

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(97)
    frames, paths = [], []
    for i in range(2):
        n = 70
        x1 = rng.normal(0, 1, n)
        y = 1.0 + 0.8 * x1 + rng.normal(0, 0.5, n)
        df = pd.DataFrame({"x1": x1, "y": y})
        p = tmp_path / f"se{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)
    pooled = pd.concat(frames)
    X = np.column_stack([np.ones(len(pooled)), pooled.x1])
    bhat = np.linalg.solve(X.T @ X, X.T @ pooled.y.to_numpy())
    rss = float(((pooled.y.to_numpy() - X @ bhat) ** 2).sum())
    cov = np.linalg.inv(X.T @ X) * rss / (len(pooled) - 2)
    want_se = np.sqrt(np.diag(cov))
    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="glm-se", image="algorithms.glm_logistic",
            input_={"method": "central_glm",
                    "kwargs": {"feature_cols": ["x1"], "target_col": "y",
                               "family": "gaussian"}})
        (res,) = net.client.wait_for_results(task["id"])
    assert res["std_errors"]["(intercept)"] == pytest.approx(want_se[0], rel=1e-6)
    assert res["std_errors"]["x1"] == pytest.approx(want_se[1], rel=1e-6)
    assert res["p_values"]["x1"] < 1e-10  # strong true effect


def test_km_greenwood_matches_manual(tmp_path):
    """KM std_err equals the Greenwood formula computed by hand on the
    pooled event table; CI brackets S(t)."""
    import numpy as np
    import pandas as pd

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(101)
    frames, paths = [], []
    for i in range(2):
        n = 50
        df = pd.DataFrame({
            "t": rng.exponential(8, n).round(1) + 0.1,
            "e": (rng.random(n) < 0.8).astype(int),
        })
        p = tmp_path / f"gw{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)
    # pool from the CSVs the stations actually read (text round-trip
    # changes float reprs vs the in-memory frames)
    pooled = pd.concat([pd.read_csv(p) for p in paths])
    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="km-gw", image="algorithms.kaplan_meier",
            input_={"method": "central_kaplan_meier",
                    "kwargs": {"time_col": "t", "event_col": "e"}})
        (res,) = net.client.wait_for_results(task["id"])
    # manual pooled Greenwood
    tab = pooled.sort_values("t")
    at_risk = len(tab)
    surv, gw = 1.0, 0.0
    manual = {}
    for t, grp in tab.groupby("t"):
        d = int((grp.e == 1).sum())
        c = int((grp.e == 0).sum())
        if d and at_risk:
            surv *= 1 - d / at_risk
            if at_risk > d:
                gw += d / (at_risk * (at_risk - d))
        manual[float(t)] = surv * gw ** 0.5
        at_risk -= d + c
    checked = 0
    for row in res["curve"]:
        assert row["std_err"] == pytest.approx(manual[row["time"]], abs=1e-12)
        lo, hi = row["ci95"]
        assert lo <= row["survival"] <= hi
        checked += 1
    assert checked > 20


def test_km_by_group(tmp_path):
    """group_col returns one KM curve per arm; each equals the pooled
    single-arm computation."""
    import numpy as np
    import pandas as pd

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(103)
    paths = []
    for i in range(2):
        n = 40
        df = pd.DataFrame({
            "t": np.concatenate([rng.exponential(6, n),
                                 rng.exponential(12, n)]).round(1) + 0.1,
            "e": (rng.random(2 * n) < 0.8).astype(int),
            "arm": ["a"] * n + ["b"] * n,
        })
        p = tmp_path / f"kmg{i}.csv"
        df.to_csv(p, index=False)
        paths.append(p)
    pooled = pd.concat([pd.read_csv(p) for p in paths])
    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="km-groups", image="algorithms.kaplan_meier",
            input_={"method": "central_kaplan_meier",
                    "kwargs": {"time_col": "t", "event_col": "e",
                               "group_col": "arm"}})
        (res,) = net.client.wait_for_results(task["id"])
    assert set(res["groups"]) == {"a", "b"}
    for arm in ("a", "b"):
        sub = pooled[pooled.arm == arm].sort_values("t")
        at_risk, surv = len(sub), 1.0
        manual_last = None
        for t, grp in sub.groupby("t"):
            d = int((grp.e == 1).sum())
            if d and at_risk:
                surv *= 1 - d / at_risk
            at_risk -= len(grp)
            manual_last = surv
        curve = res["groups"][arm]["curve"]
        assert res["groups"][arm]["n"] == len(sub)
        assert curve[-1]["survival"] == pytest.approx(manual_last, abs=1e-12)


def test_sql_database_type(tmp_path):
    """sql-type station databases (reference parity): table and query
    fragments against a sqlalchemy URI."""
    import numpy as np
    import pandas as pd
    from sqlalchemy import create_engine

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(107)
    uris = []
    vals = []
    for i in range(2):
        db = tmp_path / f"s{i}.db"
        eng = create_engine(f"sqlite:///{db}")
        v = rng.normal(5, 2, 30)
        pd.DataFrame({"v": v}).to_sql("measurements", eng, index=False)
        eng.dispose()
        vals.append(v)
        uris.append(f"sqlite:///{db}#table=measurements")
    pooled_mean = float(np.concatenate(vals).mean())
    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=u, type="sql")] for u in uris])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="sql-mean", image="algorithms.column_mean",
            input_={"method": "central_mean", "kwargs": {"columns": ["v"]}})
        (res,) = net.client.wait_for_results(task["id"])
        # column discovery works for sql-type databases too
        root = net.root_client
        nodes = root.node.list(per_page=10)["data"]
        cols = root.node.columns(nodes[0]["id"])  # list for label "default"
        assert cols == ["v"]
    assert res["v"] == pytest.approx(pooled_mean, rel=1e-12)


def test_mock_client_agrees_with_real_network(tmp_path):
    """MockAlgorithmClient (algorithm-dev harness) produces the SAME
    result as the full client->orchestrator->station path for the same
    shards — the mock is a faithful stand-in."""
    import numpy as np
    import pandas as pd

    from vantage6_amd.algorithm.mock_client import MockAlgorithmClient
    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig

    rng = np.random.default_rng(113)
    frames, paths = [], []
    for i in range(2):
        df = pd.DataFrame({"v": rng.normal(3, 1, 40 + 10 * i)})
        p = tmp_path / f"mk{i}.csv"
        df.to_csv(p, index=False)
        frames.append(df)
        paths.append(p)

    mock = MockAlgorithmClient(
        datasets=[[{"database": str(p), "type": "csv"}] for p in paths],
        module="algorithms.summary_stats")
    from algorithms.summary_stats import central_summary

    mock_res = central_summary.__wrapped__(mock, columns=["v"])

    net = DevNetwork(n_stations=2, databases=[
        [DatabaseConfig(label="default", uri=str(p), type="csv")] for p in paths])
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="real", image="algorithms.summary_stats",
            input_={"method": "central_summary",
                    "kwargs": {"columns": ["v"]}})
        (real_res,) = net.client.wait_for_results(task["id"])
    assert mock_res == real_res
