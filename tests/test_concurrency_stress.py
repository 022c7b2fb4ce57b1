"""Concurrency stress: many clients hammering the orchestrator while
stations execute — races here surfaced as intermittent PendingRollbackError
before the locked-session hardening."""
import threading

import pytest

from vantage6_amd.cli.dev import DevNetwork
from vantage6_amd.client import UserClient
from vantage6_amd.common.config import DatabaseConfig


@pytest.mark.timeout(180)
def test_many_concurrent_tasks_and_queries():
    dbs = [[DatabaseConfig(label="default",
                           uri=f"synthetic://tabular?n=64&d=4&seed={i}",
                           type="synthetic")] for i in range(3)]
    with DevNetwork(n_stations=3, databases=dbs, name="stress") as net:
        errors = []

        def submitter(idx):
            try:
                client = UserClient(core=net.core)
                client.authenticate("stress-researcher", "password")
                for t in range(6):
                    task = client.task.create(
                        collaboration=net.collaboration["id"],
                        organizations=net.org_ids,
                        name=f"s{idx}-{t}", image="algorithms.fedavg",
                        input_={"method": "partial_fit", "kwargs": {
                            "model": "logreg",
                            "model_kwargs": {"in_features": 4, "num_classes": 2},
                            "local_steps": 1, "batch_size": 16}},
                    )
                    res = client.wait_for_results(task["id"], interval=0.01,
                                                  timeout=120)
                    assert len(res) == 3
            except Exception as exc:  # noqa: BLE001
                errors.append((idx, repr(exc)))

        def lister():
            try:
                client = UserClient(core=net.core)
                client.authenticate("stress-researcher", "password")
                for _ in range(60):
                    client.task.list(per_page=100)
                    client.node.list()
            except Exception as exc:  # noqa: BLE001
                errors.append(("lister", repr(exc)))

        threads = [threading.Thread(target=submitter, args=(i,)) for i in range(4)]
        threads.append(threading.Thread(target=lister))
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=150)
        assert not errors, errors
        # every run reached a terminal state
        root = UserClient(core=net.core)
        root.authenticate("root", "root")
        tasks = root.task.list(per_page=200)["data"]
        assert len(tasks) == 24
        assert all(t["status"] == "completed" for t in tasks)
