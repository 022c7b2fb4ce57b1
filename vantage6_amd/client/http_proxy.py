"""HTTP proxy transport for AlgorithmClient (container token over REST).

Reference parity: algorithm containers talk to the node-local Flask proxy
over HTTP. In the in-process deployment algorithms get
the direct :class:`NodeProxy`; this transport exists for *external*
algorithm processes that only have a container token + the façade URL
(``AlgorithmClient.from_http``). Inputs/results travel base64-encoded
(unencrypted collaborations; encrypted ones require the in-process proxy,
which holds the node private key).
"""
from __future__ import annotations

from typing import Any

import requests

from ..common import base64s_to_bytes, bytes_to_base64s
from ..common.serialization import deserialize, serialize


class HttpProxy:
    def __init__(self, token: str, base_url: str):
        self.token = token
        self.base_url = base_url.rstrip("/")
        self.http = requests.Session()
        # decode identity claims from the container JWT payload
        import base64 as _b64
        import json as _json

        p = token.split(".")[1]
        p += "=" * (-len(p) % 4)
        claims = _json.loads(_b64.urlsafe_b64decode(p))
        self.organization_id = claims.get("org")
        self.collaboration_id = claims.get("collab")
        self.task_id = claims.get("task_id")
        self.image = claims.get("image", "")

    def _req(self, verb: str, path: str, **kw):
        r = self.http.request(
            verb, f"{self.base_url}/{path.lstrip('/')}",
            headers={"Authorization": f"Bearer {self.token}"}, **kw,
        )
        if r.status_code >= 400:
            raise RuntimeError(f"proxy {verb} {path}: {r.status_code} {r.text}")
        return r.json() if r.text else None

    # ------------------------------------------------------------------ API
    def create_subtask(self, input_: dict, organization_ids: list[int],
                       name: str = "subtask", description: str = "") -> dict:
        blob = bytes_to_base64s(serialize(input_))
        return self._req("POST", "task", json={
            "name": name, "description": description, "image": self.image,
            "collaboration_id": self.collaboration_id,
            "organizations": [{"id": i, "input": blob} for i in organization_ids],
            "method": str(input_.get("method", "")),
        })

    def get_task(self, task_id: int) -> dict:
        return self._req("GET", f"task/{task_id}")

    def list_runs(self, task_id: int) -> list[dict]:
        return self._req("GET", "run", params={"task_id": task_id})["data"]

    def _open(self, result_b64: str) -> Any:
        if not result_b64:
            return None
        return deserialize(base64s_to_bytes(result_b64))

    def get_results(self, task_id: int) -> list[Any]:
        rows = self._req("GET", "result", params={"task_id": task_id})["data"]
        return [self._open(r["result"]) for r in rows]

    def get_result(self, run_id: int) -> Any:
        run = self._req("GET", f"run/{run_id}")
        return self._open(run["result"])

    def list_organizations(self) -> list[dict]:
        """Organizations of THIS collaboration (fan-out targets)."""
        collabs = self._req("GET", "collaboration", params={"per_page": 1000})["data"]
        for c in collabs:
            if c["id"] == self.collaboration_id:
                return [{"id": o["id"], "name": o["name"]}
                        for o in c["organizations"]]
        return []
