#!/usr/bin/env python3
"""Federated GLM with Wald inference across two stations (CPU, seconds).

python examples/federated_glm.py
"""
import sys
import tempfile
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import numpy as np
import pandas as pd

from vantage6_amd.cli.dev import DevNetwork
from vantage6_amd.common.config import DatabaseConfig


def main() -> None:
    tmp = Path(tempfile.mkdtemp())
    rng = np.random.default_rng(1)
    paths = []
    for i in range(2):
        n = 300
        age = rng.normal(60, 10, n)
        dose = rng.normal(50, 15, n)
        logit = -8.0 + 0.1 * age + 0.02 * dose
        outcome = (rng.random(n) < 1 / (1 + np.exp(-logit))).astype(int)
        df = pd.DataFrame({"age": age, "dose": dose, "outcome": outcome})
        p = tmp / f"site{i}.csv"
        df.to_csv(p, index=False)
        paths.append(p)

    with DevNetwork(
        n_stations=2,
        databases=[[DatabaseConfig(label="default", uri=str(p), type="csv")]
                   for p in paths],
        name="glm-demo",
    ) as net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="glm", image="algorithms.glm_logistic",
            input_={"method": "central_glm",
                    "kwargs": {"feature_cols": ["age", "dose"],
                               "target_col": "outcome",
                               "family": "binomial"}},
        )
        (res,) = net.client.wait_for_results(task["id"])
        print(f"converged in {res['iterations']} IRLS rounds "
              f"(only XᵀWX / score aggregates crossed station borders)")
        print(f"{'term':>12} {'coef':>9} {'se':>8} {'p':>10}")
        rows = [("(intercept)", res["intercept"])] + \
            list(res["coefficients"].items())
        for name, b in rows:
            print(f"{name:>12} {b:>9.4f} {res['std_errors'][name]:>8.4f} "
                  f"{res['p_values'][name]:>10.2e}")


if __name__ == "__main__":
    main()
