"""MockAlgorithmClient — run federated algorithms fully in-process, no infra.

Reference parity: ``vantage6-algorithm-tools/vantage6/algorithm/tools/
mock_client.py`` — constructed with per-organization
datasets and the algorithm module; ``task.create`` executes the target
method synchronously for each organization; results are kept in dicts; the
sub-client API mirrors :class:`AlgorithmClient` so algorithm code runs
unchanged. Both a product feature (algorithm development) and the unit-test
vehicle.
"""
from __future__ import annotations

import importlib
import itertools
from typing import Any

from ..common.serialization import deserialize, serialize
from .wrap import dispatch


class MockAlgorithmClient:
    def __init__(
        self,
        datasets: list[list[dict]],
        module: str,
        collaboration_id: int = 1,
        organization_ids: list[int] | None = None,
    ):
        """`datasets[i]` is the list of database dicts for organization i,
        each ``{"database": <uri-or-object>, "type": "csv" | ...}`` or a
        pre-loaded object under key ``"data"``."""
        self.module = importlib.import_module(module) if isinstance(module, str) else module
        self.datasets = datasets
        self.n_orgs = len(datasets)
        self.organization_ids = organization_ids or list(range(1, self.n_orgs + 1))
        self.collaboration_id = collaboration_id
        self._task_counter = itertools.count(1)
        self._run_counter = itertools.count(1)
        self._tasks: dict[int, dict] = {}
        self._runs: dict[int, dict] = {}
        self.organization_id = self.organization_ids[0]
        self.task_id = 0
        self.task = self.Task(self)
        self.result = self.Result(self)
        self.run = self.Run(self)
        self.organization = self.Organization(self)
        self.util = self.Util(self)

    # ------------------------------------------------------------- execution
    def _env_for_org(self, org_id: int) -> dict[str, str]:
        idx = self.organization_ids.index(org_id)
        env = {"ORGANIZATION_ID": str(org_id),
               "COLLABORATION_ID": str(self.collaboration_id)}
        for i, db in enumerate(self.datasets[idx]):
            label = db.get("label", "default" if i == 0 else f"db{i}")
            env[f"{label.upper()}_DATABASE_URI"] = str(db.get("database", ""))
            env[f"{label.upper()}_DATABASE_TYPE"] = db.get("type", "csv")
            if db.get("preprocessing"):
                import json as _json

                env[f"{label.upper()}_PREPROCESSING"] = _json.dumps(
                    db["preprocessing"])
            if label == "default":
                env["DATABASE_URI"] = str(db.get("database", ""))
                env["DATABASE_TYPE"] = db.get("type", "csv")
        return env

    def _execute(self, input_: dict, org_id: int) -> Any:
        # serialize round-trip so the mock exercises the same wire format
        payload = deserialize(serialize(input_))
        sub = MockAlgorithmClient(
            self.datasets, self.module, self.collaboration_id, self.organization_ids
        )
        sub.organization_id = org_id
        return dispatch(self.module, payload, self._env_for_org(org_id), sub)

    # ------------------------------------------------------------ subclients
    class _Sub:
        def __init__(self, parent: "MockAlgorithmClient"):
            self.parent = parent

    class Task(_Sub):
        def create(self, input_: dict, organizations: list[int],
                   name: str = "mock", description: str = "") -> dict:
            p = self.parent
            task_id = next(p._task_counter)
            run_ids = []
            for org_id in organizations:
                run_id = next(p._run_counter)
                try:
                    result = p._execute(input_, org_id)
                    status = "completed"
                except Exception as exc:  # noqa: BLE001
                    result, status = None, "crashed"
                    import traceback

                    traceback.print_exc()
                p._runs[run_id] = {
                    "id": run_id, "task_id": task_id, "organization_id": org_id,
                    "result": result, "status": status,
                }
                run_ids.append(run_id)
            p._tasks[task_id] = {
                "id": task_id, "name": name, "status": (
                    "completed" if all(p._runs[r]["status"] == "completed" for r in run_ids)
                    else "crashed"),
                "runs": run_ids,
            }
            return dict(p._tasks[task_id])

        def get(self, task_id: int) -> dict:
            return dict(self.parent._tasks[task_id])

    class Result(_Sub):
        def from_task(self, task_id: int) -> list[Any]:
            p = self.parent
            return [p._runs[r]["result"] for r in p._tasks[task_id]["runs"]]

        def get(self, run_id: int) -> Any:
            return self.parent._runs[run_id]["result"]

    class Run(_Sub):
        def from_task(self, task_id: int) -> list[dict]:
            p = self.parent
            return [dict(p._runs[r]) for r in p._tasks[task_id]["runs"]]

    class Organization(_Sub):
        def list(self) -> list[dict]:
            return [{"id": i, "name": f"mock-org-{i}"} for i in self.parent.organization_ids]

        def get(self, org_id: int) -> dict:
            return {"id": org_id, "name": f"mock-org-{org_id}"}

    class Util(_Sub):
        def get_organization_id(self) -> int:
            return self.parent.organization_id

        def get_task_id(self) -> int:
            return self.parent.task_id

        def get_collaboration_id(self) -> int:
            return self.parent.collaboration_id

    def wait_for_results(self, task_id: int, interval: float = 0.0,
                         timeout: float = 0.0, allow_failures: bool = False):
        results = self.result.from_task(task_id)
        if allow_failures:
            return [r for r in results if r is not None]
        return results
