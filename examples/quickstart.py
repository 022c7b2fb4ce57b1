#!/usr/bin/env python3
"""Quickstart: boot an in-process demo network, run a federated analysis.

Runs on CPU in a few seconds:  python examples/quickstart.py
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
    # two "hospitals" with private CSV shards
    tmp = Path(tempfile.mkdtemp())
    rng = np.random.default_rng(0)
    paths = []
    for i in range(2):
        df = pd.DataFrame({
            "age": rng.normal(62, 12, 200).round(0),
            "biomarker": rng.normal(4.2, 1.1, 200),
        })
        p = tmp / f"hospital{i}.csv"
        df.to_csv(p, index=False)
        paths.append(p)

    with DevNetwork(
        n_stations=2,
        databases=[[DatabaseConfig(label="default", uri=str(p), type="csv")]
                   for p in paths],
        name="quickstart",
    ) as net:
        client = net.client  # authenticated researcher

        # 1. federated summary statistics — only aggregates leave a station
        task = client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="summary", image="algorithms.summary_stats",
            input_={"method": "central_summary",
                    "kwargs": {"columns": ["age", "biomarker"]}},
        )
        (summary,) = client.wait_for_results(task["id"])
        for col, st in summary.items():
            print(f"{col:>10}: n={st['n']}  mean={st['mean']:.2f}  "
                  f"std={st['std']:.2f}")

        # 2. federated correlation
        task = client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="corr", image="algorithms.correlation",
            input_={"method": "central_correlation",
                    "kwargs": {"columns": ["age", "biomarker"]}},
        )
        (corr,) = client.wait_for_results(task["id"])
        print(f"corr(age, biomarker) = {corr['matrix'][0][1]:+.3f} "
              f"over {corr['n']} pooled rows (no row ever left a station)")


if __name__ == "__main__":
    main()
