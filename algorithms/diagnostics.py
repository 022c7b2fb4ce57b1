"""Diagnostics algorithm — the feature-test workload (``v6 test`` analog).

Reference parity: the separate ``v6-diagnostics`` repo invoked by
``v6 test``: walks every part of the algorithm runtime
contract — env entries, data mounts, proxy/client reachability, subtask
fan-out, temporary-folder persistence across runs of one job — and returns
a pass/fail report per feature.
"""
from __future__ import annotations

import os
from pathlib import Path

from vantage6_amd.algorithm import algorithm_client, metadata


@algorithm_client
def central_diagnostics(client):
    org_ids = [o["id"] for o in client.organization.list()]
    report = {"n_organizations": len(org_ids), "features": {}}

    # 1. subtask fan-out + result collection
    task = client.task.create(
        input_={"method": "partial_diagnostics", "kwargs": {}},
        organizations=org_ids, name="diag partials",
    )
    partials = client.wait_for_results(task["id"])
    report["features"]["subtask_fanout"] = len(partials) == len(org_ids)
    report["features"]["partials"] = partials

    # 2. temp-folder persistence: second wave of subtasks reads what the
    #    first wave wrote (TEMPORARY_FOLDER shared across runs of one job)
    task2 = client.task.create(
        input_={"method": "partial_read_marker", "kwargs": {}},
        organizations=org_ids, name="diag marker readback",
    )
    markers = client.wait_for_results(task2["id"])
    report["features"]["temp_folder_shared"] = all(m.get("marker_seen") for m in markers)

    # 3. client identity
    report["features"]["client_identity"] = {
        "organization_id": client.util.get_organization_id(),
        "collaboration_id": client.util.get_collaboration_id(),
    }
    report["ok"] = (
        report["features"]["subtask_fanout"]
        and report["features"]["temp_folder_shared"]
        and all(p["env_ok"] for p in partials)
    )
    return report


@metadata
def partial_diagnostics(meta):
    env_ok = all(
        v is not None
        for v in (meta.task_id, meta.organization_id, meta.collaboration_id,
                  meta.temporary_directory)
    )
    marker_ok = False
    if meta.temporary_directory:
        marker = Path(meta.temporary_directory) / f"diag_marker_{meta.organization_id}"
        marker.write_text("seen")
        marker_ok = marker.exists()
    # env contract entries visible to plain os.environ users too? (we pass env
    # via the dispatch context, not the process env — report what we got)
    return {
        "env_ok": env_ok,
        "marker_written": marker_ok,
        "organization_id": meta.organization_id,
        "task_id": meta.task_id,
    }


@metadata
def partial_read_marker(meta):
    seen = False
    if meta.temporary_directory:
        marker = Path(meta.temporary_directory) / f"diag_marker_{meta.organization_id}"
        seen = marker.exists()
    return {"marker_seen": seen, "organization_id": meta.organization_id}


def partial_spin(seconds: float = 300.0, marker_file: str | None = None):
    """Busy-loop for the kill-preemption test: burns CPU in a Python loop
    until ``seconds`` elapse (far longer than any test timeout). Writes
    ``marker_file`` once started so the test knows the loop is live."""
    import time

    t0 = time.monotonic()
    if marker_file:
        Path(marker_file).write_text("spinning")
    x = 0
    while time.monotonic() - t0 < seconds:
        x += 1  # pure-Python loop: interruptible at every bytecode
    return {"iterations": x}
