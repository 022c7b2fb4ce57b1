"""NodeProxy — the algorithms' only gateway to the orchestrator.

Reference parity: ``vantage6-node/vantage6/node/proxy_server.py``
 — algorithm containers reach the central server ONLY
through the node's proxy, which (a) injects the container JWT, (b) encrypts
subtask inputs per destination organization with their public keys, and
(c) decrypts results with the node's private key before handing them to the
algorithm. Here the proxy is an in-process object holding a reference to the
:class:`ServerCore` and the node's cryptor; the responsibilities are
identical, the HTTP hop is gone.
"""
from __future__ import annotations

from typing import Any

from ..common import bytes_to_base64s
from ..common.encryption import CryptorBase
from ..common.serialization import deserialize, serialize
from ..server.core import Identity, ServerCore


class NodeProxy:
    def __init__(
        self,
        core: ServerCore,
        container_identity: Identity,
        cryptor: CryptorBase,
        encrypted_collaboration: bool,
        image: str,
    ):
        self.core = core
        self.ident = container_identity
        self.cryptor = cryptor
        self.encrypted = encrypted_collaboration
        self.image = image

    # identity accessors used by AlgorithmClient.util
    @property
    def organization_id(self) -> int:
        return self.ident.organization_id

    @property
    def collaboration_id(self) -> int:
        return self.ident.collaboration_id

    @property
    def task_id(self) -> int:
        return self.ident.task_id

    # ------------------------------------------------------------------ tasks
    def _encrypt_for_org(self, blob: bytes, org_id: int) -> str:
        if self.encrypted:
            pubkey = self.core.organization_public_key(org_id)
            if not pubkey:
                raise ValueError(f"organization {org_id} has no public key uploaded")
            return self.cryptor.encrypt_bytes_to_str(blob, pubkey)
        return bytes_to_base64s(blob)

    def create_subtask(
        self,
        input_: dict,
        organization_ids: list[int],
        name: str = "subtask",
        description: str = "",
    ) -> dict:
        blob = serialize(input_)
        organizations = [
            {"id": org_id, "input": self._encrypt_for_org(blob, org_id)}
            for org_id in organization_ids
        ]
        return self.core.create_task(
            self.ident,
            name=name,
            description=description,
            image=self.image,
            collaboration_id=self.ident.collaboration_id,
            organizations=organizations,
            method=str(input_.get("method", "")),
        )

    def get_task(self, task_id: int) -> dict:
        return self.core.get_task(self.ident, task_id)

    def list_runs(self, task_id: int) -> list[dict]:
        return self.core.list_runs(self.ident, task_id=task_id)["data"]

    # ---------------------------------------------------------------- results
    def _open_result(self, result_b64: str) -> Any:
        if not result_b64:
            return None
        blob = self.cryptor.decrypt_str_to_bytes(result_b64)
        return deserialize(blob)

    def get_results(self, task_id: int) -> list[Any]:
        rows = self.core.list_results(self.ident, task_id=task_id)["data"]
        return [self._open_result(r["result"]) for r in rows]

    def get_result(self, run_id: int) -> Any:
        run = self.core.get_run(self.ident, run_id)
        return self._open_result(run["result"])

    def list_organizations(self) -> list[dict]:
        collab = self.core.get_collaboration(self.ident,
                                             self.ident.collaboration_id)
        return [{"id": o["id"], "name": o["name"]}
                for o in collab["organizations"]]
