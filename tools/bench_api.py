#!/usr/bin/env python3
"""Control-plane microbenchmark — BASELINE config[0] evidence.

Measures federated task round-trip latency + throughput through the full
client→orchestrator→station→algorithm→result path on CPU stations (the
reference's analog is seconds/round: Docker start + 1 s polling).
"""
import json
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from vantage6_amd.cli.dev import DevNetwork
from vantage6_amd.common.config import DatabaseConfig


def main(n_tasks: int = 30) -> None:
    dbs = [[DatabaseConfig(label="default",
                           uri=f"synthetic://tabular?n=512&d=8&seed={i}",
                           type="synthetic")] for i in range(2)]
    with DevNetwork(n_stations=2, databases=dbs, name="apibench") as net:
        lat = []
        t0 = time.perf_counter()
        for i in range(n_tasks):
            t1 = time.perf_counter()
            task = net.client.task.create(
                collaboration=net.collaboration["id"],
                organizations=net.org_ids,
                name=f"bench-{i}", image="algorithms.fedavg",
                input_={"method": "partial_fit", "kwargs": {
                    "model": "logreg",
                    "model_kwargs": {"in_features": 8, "num_classes": 2},
                    "local_steps": 1, "batch_size": 32}},
            )
            net.client.wait_for_results(task["id"], interval=0.002, timeout=60)
            lat.append(time.perf_counter() - t1)
        wall = time.perf_counter() - t0
    print(json.dumps({
        "metric": "federated task round-trip (2 CPU stations, both fit)",
        "tasks": n_tasks,
        "median_latency_ms": statistics.median(lat) * 1e3,
        "p90_latency_ms": sorted(lat)[int(0.9 * len(lat))] * 1e3,
        "tasks_per_sec": n_tasks / wall,
        "note": "reference: Docker container start + ~1s polling per task",
    }))


if __name__ == "__main__":
    main()
