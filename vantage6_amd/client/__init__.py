"""Researcher client ("UserClient").

Reference parity: ``vantage6-client/vantage6/client/__init__.py``
 — ``Client.authenticate``, ``setup_encryption``,
sub-clients ``.task``, ``.result``, ``.run``, ``.node``, ``.organization``,
``.collaboration``, ``.user``, ``.study`` and ``wait_for_results``.

Transports: the in-process gateway drives :class:`ServerCore` directly
(MI355X single-node deployment), and the HTTP gateway speaks the same JSON
schema against the FastAPI façade — researcher code is identical either way.
"""
from __future__ import annotations

import time
from typing import Any

from ..common import bytes_to_base64s
from ..common.encryption import CryptorBase, RSACryptor
from ..common.serialization import deserialize, serialize
from ..common.task_status import has_task_failed, has_task_finished


class CoreGateway:
    """In-process transport: call the orchestrator service layer directly."""

    def __init__(self, core):
        self.core = core
        self.token: str | None = None
        self.refresh_token_value: str | None = None

    def authenticate(self, username: str, password: str) -> dict:
        out = self.core.authenticate_user(username, password)
        self.token = out["access_token"]
        self.refresh_token_value = out["refresh_token"]
        return out

    @property
    def ident(self):
        if self.token is None:
            raise RuntimeError("client is not authenticated")
        return self.core.identity_from_token(self.token)

    def call(self, _core_method: str, *args, **kwargs):
        return getattr(self.core, _core_method)(self.ident, *args, **kwargs)


class HttpGateway:
    """HTTP transport against the FastAPI façade (same JSON schema)."""

    def __init__(self, base_url: str):
        import requests

        self.http = requests.Session()
        self.base_url = base_url.rstrip("/")
        if not self.base_url.endswith("/api"):
            self.base_url += "/api"
        self.token: str | None = None
        self.refresh_token_value: str | None = None

    def authenticate(self, username: str, password: str) -> dict:
        r = self.http.post(
            f"{self.base_url}/token/user", json={"username": username, "password": password}
        )
        self._raise_for(r)
        out = r.json()
        self.token = out["access_token"]
        self.refresh_token_value = out.get("refresh_token")
        return out

    def _raise_for(self, r):
        if r.status_code >= 400:
            try:
                msg = r.json().get("detail", r.text)
            except Exception:  # noqa: BLE001
                msg = r.text
            raise RuntimeError(f"server returned {r.status_code}: {msg}")

    max_retries = 3  # transient connection errors (reference ClientBase)

    def _send(self, verb, url, **kw):
        import time as _time

        import requests

        last = None
        for attempt in range(self.max_retries):
            try:
                return self.http.request(verb, url, **kw)
            except requests.exceptions.ConnectionError as exc:
                last = exc
                _time.sleep(0.2 * (attempt + 1))
        raise last

    def request(self, verb: str, path: str, json: dict | None = None, params=None):
        headers = {"Authorization": f"Bearer {self.token}"} if self.token else {}
        r = self._send(
            verb, f"{self.base_url}/{path.lstrip('/')}", json=json, params=params,
            headers=headers,
        )
        if r.status_code == 401 and self.refresh_token_value:
            rr = self.http.post(
                f"{self.base_url}/token/refresh",
                json={"refresh_token": self.refresh_token_value},
            )
            if rr.status_code < 400:
                self.token = rr.json()["access_token"]
                self.refresh_token_value = rr.json().get("refresh_token")
                headers = {"Authorization": f"Bearer {self.token}"}
                r = self._send(
                    verb, f"{self.base_url}/{path.lstrip('/')}", json=json, params=params,
                    headers=headers,
                )
        self._raise_for(r)
        return r.json() if r.text else None


class UserClient:
    """The researcher SDK. ``Client`` alias kept for parity."""

    def __init__(self, core=None, url: str | None = None):
        if core is not None:
            self.gw = CoreGateway(core)
        elif url is not None:
            self.gw = HttpGateway(url)
        else:
            raise ValueError("pass core= (in-process) or url= (HTTP façade)")
        self.cryptor: CryptorBase = CryptorBase()
        self.whoami: dict = {}
        self.task = self.Task(self)
        self.result = self.Result(self)
        self.run = self.Run(self)
        self.organization = self.Organization(self)
        self.collaboration = self.Collaboration(self)
        self.node = self.NodeSub(self)
        self.user = self.UserSub(self)
        self.study = self.StudySub(self)
        self.role = self.RoleSub(self)
        self.rule = self.RuleSub(self)
        self.store = self.StoreSub(self)

    # ----------------------------------------------------------------- auth
    def authenticate(self, username: str, password: str) -> dict:
        out = self.gw.authenticate(username, password)
        self.whoami = {"username": username}
        try:
            me = self._my_org_id()
            self.whoami["organization_id"] = me
            if isinstance(self.gw, CoreGateway):
                ident = self.gw.ident
                self.whoami["user_id"] = ident.id
        except Exception:  # noqa: BLE001 — whoami is best-effort metadata
            pass
        return out

    def setup_encryption(self, private_key_file: str | None, key_bits: int = 2048) -> None:
        """Load/create the researcher org's private key & upload public key."""
        if private_key_file is None:
            self.cryptor = CryptorBase()
            return
        self.cryptor = RSACryptor(private_key_file, bits=key_bits)
        # upload our org public key so stations can encrypt results for us
        me_org = self._my_org_id()
        self._call("set_organization_public_key", "PATCH", f"organization/{me_org}",
                   org_id=me_org, public_key=self.cryptor.public_key_str,
                   json={"public_key": self.cryptor.public_key_str})

    def _my_org_id(self) -> int:
        if isinstance(self.gw, CoreGateway):
            return self.gw.ident.organization_id
        # HTTP: decode our own token payload (no signature check client-side)
        import base64, json as _json

        p = self.gw.token.split(".")[1]
        p += "=" * (-len(p) % 4)
        return _json.loads(base64.urlsafe_b64decode(p))["org"]

    # ------------------------------------------------------------ transport
    def _call(self, core_method: str, verb: str, path: str, json: dict | None = None,
              params=None, **core_kwargs):
        if isinstance(self.gw, CoreGateway):
            return self.gw.call(core_method, **core_kwargs)
        return self.gw.request(verb, path, json=json, params=params)

    # --------------------------------------------------------------- waiting
    def wait_for_results(self, task_id: int, interval: float = 0.1,
                         timeout: float = 600.0) -> list[Any]:
        t0 = time.monotonic()
        while True:
            task = self.task.get(task_id)
            if has_task_finished(task["status"]):
                if has_task_failed(task["status"]):
                    runs = self.run.from_task(task_id)
                    logs = "\n---\n".join(r.get("log") or "" for r in runs)
                    raise RuntimeError(
                        f"task {task_id} ended with status {task['status']!r}; logs:\n{logs}"
                    )
                return self.result.from_task(task_id)
            if time.monotonic() - t0 > timeout:
                raise TimeoutError(f"timeout waiting for task {task_id}")
            time.sleep(interval)

    # ------------------------------------------------------------ subclients
    class _Sub:
        def __init__(self, parent: "UserClient"):
            self.parent = parent

    class Task(_Sub):
        def create(
            self,
            collaboration: int,
            organizations: list[int],
            name: str,
            image: str,
            input_: dict,
            description: str = "",
            databases: list[dict] | None = None,
            study: int | None = None,
        ) -> dict:
            p = self.parent
            orgs = []
            encrypted = p.collaboration.get(collaboration)["encrypted"]
            blob = serialize(input_) if input_ is not None else None
            for org in organizations:
                # an org entry may be a bare id (input_ is serialized and
                # encrypted per org here) or a prepared {"id", "input"}
                # dict (caller did its own serialization/encryption —
                # the reference's node/proxy shape)
                if isinstance(org, dict):
                    orgs.append({"id": org["id"], "input": org.get("input", "")})
                    continue
                if blob is None:
                    raise ValueError("input_ is required for bare org ids")
                if encrypted:
                    pubkey = p.organization.get(org)["public_key"]
                    if not pubkey:
                        raise RuntimeError(f"organization {org} has no public key")
                    enc = p.cryptor.encrypt_bytes_to_str(blob, pubkey)
                else:
                    enc = bytes_to_base64s(blob)
                orgs.append({"id": org, "input": enc})
            body = {
                "name": name,
                "image": image,
                "collaboration_id": collaboration,
                "description": description,
                "databases": databases or [],
                "study_id": study,
                "organizations": orgs,
                "method": str(input_.get("method", ""))
                          if isinstance(input_, dict) else "",
            }
            return p._call("create_task", "POST", "task", json=body, **body)

        def get(self, task_id: int) -> dict:
            return self.parent._call("get_task", "GET", f"task/{task_id}", task_id=task_id)

        def list(self, **filters) -> dict:
            return self.parent._call("list_tasks", "GET", "task", params=filters, **filters)

        def delete(self, task_id: int) -> dict:
            """Kill a task (DELETE /api/task/<id>)."""
            return self.parent._call("kill_task", "DELETE", f"task/{task_id}", task_id=task_id)

    class Result(_Sub):
        def from_task(self, task_id: int) -> list[Any]:
            p = self.parent
            rows = p._call("list_results", "GET", "result", params={"task_id": task_id},
                           task_id=task_id)["data"]
            out = []
            for row in rows:
                if not row["result"]:
                    out.append(None)
                    continue
                blob = p.cryptor.decrypt_str_to_bytes(row["result"])
                out.append(deserialize(blob))
            return out

    class Run(_Sub):
        def from_task(self, task_id: int) -> list[dict]:
            return self.parent._call("list_runs", "GET", "run", params={"task_id": task_id},
                                     task_id=task_id)["data"]

        def get(self, run_id: int) -> dict:
            return self.parent._call("get_run", "GET", f"run/{run_id}", run_id=run_id)

    class Organization(_Sub):
        def list(self, **params) -> dict:
            return self.parent._call("list_organizations", "GET", "organization",
                                     params=params, **params)

        def get(self, org_id: int) -> dict:
            return self.parent._call("get_organization", "GET", f"organization/{org_id}",
                                     org_id=org_id)

        def update(self, org_id: int, **fields) -> dict:
            return self.parent._call("update_organization", "PATCH",
                                     f"organization/{org_id}", json=fields,
                                     org_id=org_id, **fields)

        def delete(self, org_id: int, delete_dependents: bool = False) -> dict:
            return self.parent._call(
                "delete_organization", "DELETE",
                f"organization/{org_id}?delete_dependents="
                f"{str(delete_dependents).lower()}",
                org_id=org_id, delete_dependents=delete_dependents)

        def create(self, name: str, **kw) -> dict:
            body = {"name": name, **kw}
            return self.parent._call("create_organization", "POST", "organization",
                                     json=body, name=name, **kw)

    class Collaboration(_Sub):
        def list(self, **params) -> dict:
            return self.parent._call("list_collaborations", "GET", "collaboration",
                                     params=params, **params)

        def get(self, cid: int) -> dict:
            return self.parent._call("get_collaboration", "GET",
                                     f"collaboration/{cid}", collab_id=cid)

        def update(self, cid: int, **fields) -> dict:
            return self.parent._call("update_collaboration", "PATCH",
                                     f"collaboration/{cid}", json=fields,
                                     collab_id=cid, **fields)

        def delete(self, cid: int, delete_dependents: bool = False) -> dict:
            return self.parent._call(
                "delete_collaboration", "DELETE",
                f"collaboration/{cid}?delete_dependents="
                f"{str(delete_dependents).lower()}",
                collab_id=cid, delete_dependents=delete_dependents)

        def create(self, name: str, organization_ids: list[int], encrypted=False) -> dict:
            body = {"name": name, "organization_ids": organization_ids, "encrypted": encrypted}
            return self.parent._call("create_collaboration", "POST", "collaboration",
                                     json=body, **body)

    class NodeSub(_Sub):
        def list(self, **params) -> dict:
            return self.parent._call("list_nodes", "GET", "node", params=params, **params)

        def update(self, node_id: int, **fields) -> dict:
            return self.parent._call("update_node", "PATCH", f"node/{node_id}",
                                     json=fields, node_id=node_id, **fields)

        def delete(self, node_id: int) -> dict:
            return self.parent._call("delete_node", "DELETE", f"node/{node_id}",
                                     node_id=node_id)

        def reset_api_key(self, node_id: int) -> dict:
            """Rotate the node's api key; the new key is returned once
            (reference /api/recover/node)."""
            return self.parent._call("reset_node_api_key", "POST",
                                     "recover/node", json={"id": node_id},
                                     node_id=node_id)

        def columns(self, node_id: int, label: str = "default") -> list[str]:
            """Data-column discovery (reference /api/column)."""
            return self.parent._call("get_columns", "GET", f"column/{node_id}",
                                     params={"label": label}, node_id=node_id,
                                     label=label)

        def create(self, name: str, collaboration_id: int, organization_id: int,
                   device: str = "cpu") -> dict:
            body = {"name": name, "collaboration_id": collaboration_id,
                    "organization_id": organization_id, "device": device}
            return self.parent._call("create_node", "POST", "node", json=body, **body)

    class UserSub(_Sub):
        def create(self, username: str, password: str, organization_id: int,
                   roles: list[str] | None = None, **kw) -> dict:
            body = {"username": username, "password": password,
                    "organization_id": organization_id, "roles": roles or [], **kw}
            return self.parent._call("create_user", "POST", "user", json=body, **body)

        def list(self, **params) -> dict:
            return self.parent._call("list_users", "GET", "user", params=params,
                                     **params)

        def get(self, user_id: int) -> dict:
            return self.parent._call("get_user", "GET", f"user/{user_id}",
                                     user_id=user_id)

        def update(self, user_id: int, **fields) -> dict:
            return self.parent._call("update_user", "PATCH", f"user/{user_id}",
                                     json=fields, user_id=user_id, **fields)

        def delete(self, user_id: int) -> dict:
            return self.parent._call("delete_user", "DELETE", f"user/{user_id}",
                                     user_id=user_id)

    class RoleSub(_Sub):
        def list(self, **params) -> dict:
            return self.parent._call("list_roles", "GET", "role", params=params, **params)

        def create(self, name: str, rule_ids: list[int],
                   organization_id: int | None = None) -> dict:
            body = {"name": name, "rule_ids": rule_ids,
                    "organization_id": organization_id}
            return self.parent._call("create_role", "POST", "role", json=body, **body)

        def get(self, role_id: int) -> dict:
            return self.parent._call("get_role", "GET", f"role/{role_id}",
                                     role_id=role_id)

        def update(self, role_id: int, **fields) -> dict:
            return self.parent._call("update_role", "PATCH",
                                     f"role/{role_id}", json=fields,
                                     role_id=role_id, **fields)

        def delete(self, role_id: int) -> dict:
            return self.parent._call("delete_role", "DELETE",
                                     f"role/{role_id}", role_id=role_id)

    class RuleSub(_Sub):
        def list(self, **params) -> dict:
            return self.parent._call("list_rules", "GET", "rule", params=params, **params)

        def get(self, rule_id: int) -> dict:
            return self.parent._call("get_rule", "GET", f"rule/{rule_id}",
                                     rule_id=rule_id)

    class StoreSub(_Sub):
        """Algorithm-store sub-client (reference ``client.store``, 4.2+)."""

        def submit(self, name: str, image: str, description: str = "") -> dict:
            body = {"name": name, "image": image, "description": description}
            return self.parent._call("store_submit_algorithm", "POST", "algorithm",
                                     json=body, **body)

        def review(self, algorithm_id: int, approve: bool, comment: str = "") -> dict:
            body = {"approve": approve, "comment": comment}
            return self.parent._call("store_review_algorithm", "PATCH",
                                     f"algorithm/{algorithm_id}/review", json=body,
                                     algorithm_id=algorithm_id, **body)

        def list(self, **params) -> dict:
            return self.parent._call("store_list_algorithms", "GET", "algorithm",
                                     params=params, **params)

    class StudySub(_Sub):
        def list(self, **params) -> dict:
            return self.parent._call("list_studies", "GET", "study",
                                     params=params, **params)

        def get(self, study_id: int) -> dict:
            return self.parent._call("get_study", "GET", f"study/{study_id}",
                                     study_id=study_id)

        def create(self, name: str, collaboration_id: int,
                   organization_ids: list[int]) -> dict:
            body = {"name": name, "collaboration_id": collaboration_id,
                    "organization_ids": organization_ids}
            return self.parent._call("create_study", "POST", "study", json=body, **body)

        def update(self, study_id: int, **fields) -> dict:
            return self.parent._call("update_study", "PATCH",
                                     f"study/{study_id}", json=fields,
                                     study_id=study_id, **fields)

        def delete(self, study_id: int) -> dict:
            return self.parent._call("delete_study", "DELETE",
                                     f"study/{study_id}", study_id=study_id)


Client = UserClient
