"""Kill preemption: a RUNNING algorithm is interrupted and the station
accepts the next task within seconds (VERDICT round-1 item 6; reference:
DockerManager.kill_tasks kills the container).
"""
import time

import pytest

from vantage6_amd.cli.dev import DevNetwork
from vantage6_amd.common.task_status import RunStatus


@pytest.mark.timeout(120)
def test_kill_preempts_running_algorithm(tmp_path):
    marker = tmp_path / "spinning.txt"
    net = DevNetwork(n_stations=1)
    with net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="spin", image="algorithms.diagnostics",
            input_={"method": "partial_spin",
                    "kwargs": {"seconds": 300.0, "marker_file": str(marker)}},
        )
        # wait until the algorithm loop is definitely executing
        t0 = time.monotonic()
        while not marker.exists():
            assert time.monotonic() - t0 < 30, "spin task never started"
            time.sleep(0.02)
        run = net.client.run.from_task(task["id"])[0]
        assert run["status"] == RunStatus.ACTIVE.value

        t_kill = time.monotonic()
        net.client.task.delete(task["id"])
        # the station must be free for new work within seconds: the next
        # task completes even though the spin loop claimed 300 s
        task2 = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="after-kill", image="algorithms.diagnostics",
            input_={"method": "partial_diagnostics"},
        )
        results = net.client.wait_for_results(task2["id"], interval=0.05,
                                              timeout=30)
        assert len(results) == 1
        assert time.monotonic() - t_kill < 30
        # the killed run stays terminally KILLED
        run = net.client.run.from_task(task["id"])[0]
        assert run["status"] == RunStatus.KILLED.value
        # and its executor thread actually stopped (no lingering active run)
        st = net.stations[0]
        t0 = time.monotonic()
        while st._active and time.monotonic() - t0 < 10:
            time.sleep(0.05)
        assert not st._active
