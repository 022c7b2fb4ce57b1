"""AlgorithmClient — used inside algorithm code on a station.

Reference parity: ``vantage6-algorithm-tools/vantage6/algorithm/client/
__init__.py`` — sub-clients ``.task`` (subtask
fan-out), ``.result``/``.run`` (collect partial results), ``.organization``,
plus ``wait_for_results``. All traffic goes through the station's *proxy*
object (:class:`vantage6_amd.node.proxy.NodeProxy`): the proxy encrypts
subtask inputs per destination organization and decrypts results with the
node's private key, exactly like the reference's node-local Flask proxy
(``vantage6-node/vantage6/node/proxy_server.py``), but as in-process calls.
"""
from __future__ import annotations

import time
from typing import Any

from ..common.task_status import has_task_failed, has_task_finished
from .exceptions import CollectResultsError


class AlgorithmClient:
    def __init__(self, proxy):
        """`proxy` implements the NodeProxy interface (see node/proxy.py)."""
        self._proxy = proxy
        self.task = self.Task(self)
        self.result = self.Result(self)
        self.run = self.Run(self)
        self.organization = self.Organization(self)
        self.util = self.Util(self)

    @classmethod
    def from_http(cls, token: str, host: str, port, api_path: str = "/api"):
        from ..client.http_proxy import HttpProxy

        return cls(HttpProxy(token=token, base_url=f"{host}:{port}{api_path}"))

    # ---------------------------------------------------------------- waiting
    def wait_for_results(self, task_id: int, interval: float = 0.1,
                         timeout: float = 300.0, allow_failures: bool = False):
        """Block until all runs of `task_id` finish; return decrypted results.

        Reference: ``AlgorithmClient.wait_for_results`` (poll loop). The
        in-process deployment polls the orchestrator directly, so `interval`
        can be far below the reference's ~1 s default.

        With ``allow_failures=True`` (elastic federated rounds, SURVEY.md
        §6.3: a crashed station simply drops out of the weighted mean) the
        call waits until every run is terminal and returns only the
        COMPLETED runs' results; inspect ``run.from_task`` for statuses.
        """
        from ..common.task_status import RunStatus

        t0 = time.monotonic()
        while True:
            if allow_failures:
                runs = self._proxy.list_runs(task_id)
                if runs and all(has_task_finished(r["status"]) for r in runs):
                    completed = [r for r in runs
                                 if r["status"] == RunStatus.COMPLETED.value]
                    if not completed:
                        raise CollectResultsError(
                            f"task {task_id}: every run failed"
                        )
                    return [self._proxy.get_result(r["id"]) for r in completed]
            else:
                task = self._proxy.get_task(task_id)
                status = task["status"]
                if has_task_finished(status):
                    if has_task_failed(status):
                        # surface the failed subtask logs so the parent
                        # run's log names the real error (reference: child
                        # logs are fetched by the researcher; algorithms
                        # only see the status otherwise)
                        logs = ""
                        try:
                            runs = self._proxy.list_runs(task_id)
                            logs = "\n---\n".join(
                                (r.get("log") or "") for r in runs
                                if has_task_failed(r["status"]))
                        except Exception:  # noqa: BLE001
                            pass
                        raise CollectResultsError(
                            f"task {task_id} ended with status {status!r}; "
                            f"subtask logs:\n{logs}"
                        )
                    return self.result.from_task(task_id)
            if time.monotonic() - t0 > timeout:
                raise CollectResultsError(f"timeout waiting for task {task_id}")
            time.sleep(interval)

    # ------------------------------------------------------------- subclients
    class _Sub:
        def __init__(self, parent: "AlgorithmClient"):
            self.parent = parent
            self._proxy = parent._proxy

    class Task(_Sub):
        def create(
            self,
            input_: dict,
            organizations: list[int],
            name: str = "subtask",
            description: str = "",
        ) -> dict:
            """Fan out a subtask to sibling organizations."""
            return self._proxy.create_subtask(
                input_=input_, organization_ids=organizations, name=name,
                description=description,
            )

        def get(self, task_id: int) -> dict:
            return self._proxy.get_task(task_id)

    class Result(_Sub):
        def from_task(self, task_id: int) -> list[Any]:
            """Decrypted + deserialized results of every run of a task."""
            return self._proxy.get_results(task_id)

        def get(self, run_id: int) -> Any:
            return self._proxy.get_result(run_id)

    class Run(_Sub):
        def from_task(self, task_id: int) -> list[dict]:
            return self._proxy.list_runs(task_id)

    class Organization(_Sub):
        def list(self) -> list[dict]:
            """Organizations in this collaboration (for fan-out targeting)."""
            return self._proxy.list_organizations()

        def get(self, org_id: int) -> dict:
            for org in self._proxy.list_organizations():
                if org["id"] == org_id:
                    return org
            raise KeyError(org_id)

    class Util(_Sub):
        def get_organization_id(self) -> int:
            return self._proxy.organization_id

        def get_task_id(self) -> int:
            return self._proxy.task_id

        def get_collaboration_id(self) -> int:
            return self._proxy.collaboration_id
