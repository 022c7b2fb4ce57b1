"""Remote-station transport: run a :class:`~vantage6_amd.node.station.Station`
against an orchestrator on ANOTHER host, over HTTP + WebSocket.

Reference parity: this is the reference's actual deployment shape — the
node daemon authenticates with its api-key over HTTPS
(``vantage6-node/vantage6/node/__init__.py``), receives ``new_task`` pushes
over Socket.IO, and PATCHes results back over REST.
The MI355X default keeps stations in-process (one per GPU), but
:class:`RemoteCore` makes the same ``Station`` class work across machines:
it implements the exact subset of the :class:`ServerCore` surface that the
station and its :class:`NodeProxy` call, translating each call to the
FastAPI façade (:mod:`vantage6_amd.server.api`) and running a WebSocket
listener thread for the event rooms.

Usage::

    core = RemoteCore("http://orchestrator-host:7601")
    station = Station(core, config, api_key="...")
    station.start()
"""
from __future__ import annotations

import base64
import json
import logging
import threading
from typing import Callable

from ..client import HttpGateway
from ..server.core import Identity

log = logging.getLogger(__name__)


def _decode_payload(token: str) -> dict:
    """Client-side JWT payload decode (no signature check — the server
    verifies; the station only needs its own ids for room names)."""
    part = token.split(".")[1]
    part += "=" * (-len(part) % 4)
    return json.loads(base64.urlsafe_b64decode(part))


class _WsEvents:
    """`events.subscribe(room, cb)` backed by the `/api/ws` WebSocket.

    The server already scopes pushed events to the authenticated identity's
    rooms, so every received message is relevant; callbacks are deduped (the
    station registers the same handler for both of its rooms).
    """

    def __init__(self, remote: "RemoteCore"):
        self._remote = remote
        self._callbacks: list[Callable] = []
        self._thread: threading.Thread | None = None
        self._stop = threading.Event()

    def subscribe(self, room: str, callback: Callable) -> None:  # noqa: ARG002
        if callback not in self._callbacks:
            self._callbacks.append(callback)
        if self._thread is None:
            self._thread = threading.Thread(
                target=self._run, name="remote-core-ws", daemon=True)
            self._thread.start()

    def unsubscribe(self, room: str, callback: Callable) -> None:  # noqa: ARG002
        if callback in self._callbacks:
            self._callbacks.remove(callback)

    def stop(self) -> None:
        self._stop.set()

    def _run(self) -> None:
        import websockets.sync.client as _ws_sync  # type: ignore

        self._run_with(_ws_sync)

    def _run_with(self, ws_mod) -> None:
        url = self._remote.ws_url()
        while not self._stop.is_set():
            try:
                with ws_mod.connect(url) as ws:
                    while not self._stop.is_set():
                        msg = json.loads(ws.recv(timeout=1.0))
                        for cb in list(self._callbacks):
                            try:
                                cb(msg["event"], msg["data"])
                            except Exception:  # noqa: BLE001
                                log.exception("event callback failed")
            except TimeoutError:
                continue
            except Exception:  # noqa: BLE001
                if self._stop.is_set():
                    return
                log.debug("ws reconnect", exc_info=True)
                self._stop.wait(0.5)


class _PollingEvents(_WsEvents):
    """Fallback event source when no WebSocket client library is available:
    poll open runs every ``interval`` seconds and synthesize ``new_task``
    events (the station dedupes via its queue/kill logic)."""

    interval = 1.0

    def _dispatch(self, event: str, data: dict) -> None:
        for cb in list(self._callbacks):
            try:
                cb(event, data)
            except Exception:  # noqa: BLE001
                log.exception("event callback failed")

    def _run(self) -> None:
        seen: set[int] = set()
        open_now: dict[int, int] = {}  # run id -> task id
        while not self._stop.wait(self.interval):
            try:
                runs = self._remote._request(
                    "GET", "run", params={"state": "open"})["data"]
            except Exception:  # noqa: BLE001
                continue
            ids = {r["id"] for r in runs}
            for r in runs:
                if r["id"] in seen:
                    continue
                seen.add(r["id"])
                open_now[r["id"]] = r["task_id"]
                self._dispatch("new_task", {
                    "task_id": r["task_id"], "run_id": r["id"],
                    # node-token listings only contain OUR runs
                    "organization_id": self._remote.org_id,
                })
            # a run that LEFT the open set without us finishing it was
            # terminated server-side — check for a user kill
            for rid in [x for x in open_now if x not in ids]:
                tid = open_now.pop(rid)
                try:
                    status = self._remote._request(
                        "GET", f"run/{rid}")["status"]
                except Exception:  # noqa: BLE001
                    continue
                if status == "killed by user":
                    self._dispatch("kill_containers", {
                        "task_id": tid,
                        "kill_list": [{
                            "run_id": rid, "task_id": tid,
                            "organization_id": self._remote.org_id}],
                    })


class RemoteCore:
    """HTTP implementation of the ServerCore surface used by Station and
    NodeProxy. All ``ident`` parameters are accepted for signature parity
    and ignored — the bearer token IS the identity server-side."""

    def __init__(self, url: str, use_websocket: bool | None = None):
        self.gw = HttpGateway(url)
        if use_websocket is None:
            try:
                import websockets.sync.client  # noqa: F401

                use_websocket = True
            except ImportError:
                use_websocket = False
        self.events = (_WsEvents(self) if use_websocket
                       else _PollingEvents(self))

    # ------------------------------------------------------------ transport
    def ws_url(self) -> str:
        base = self.gw.base_url.replace("http://", "ws://").replace(
            "https://", "wss://")
        return f"{base}/ws?token={self.gw.token}"

    def close(self) -> None:
        self.events.stop()

    # ----------------------------------------------------------------- auth
    org_id: int | None = None  # set at authenticate; used by event synth
    _api_key: str | None = None

    def authenticate_node(self, api_key: str) -> dict:
        self._api_key = api_key
        out = self.gw.request("POST", "token/node", json={"api_key": api_key})
        self.gw.token = out["access_token"]
        self.gw.refresh_token_value = out.get("refresh_token")
        self.org_id = _decode_payload(out["access_token"]).get("org")
        return out

    def _request(self, verb: str, path: str, ident=None, **kw):
        """Gateway call that survives full token expiry: on a 401 (even
        after the gateway's own refresh attempt), re-authenticate with the
        stored api-key and retry once (reference: node re-auths on the
        ``expired_token`` socket event).

        When ``ident`` carries a container token (proxy calls on behalf of
        a running algorithm), the request is sent with THAT bearer — the
        server enforces container semantics (same-collaboration subtasks,
        image match) on it."""
        token = getattr(ident, "token", None)
        if token is not None and getattr(ident, "type", None) == "container":
            r = self.gw._send(
                verb, f"{self.gw.base_url}/{path.lstrip('/')}",
                json=kw.get("json"), params=kw.get("params"),
                headers={"Authorization": f"Bearer {token}"})
            self.gw._raise_for(r)
            return r.json() if r.text else None
        try:
            return self.gw.request(verb, path, **kw)
        except RuntimeError as exc:
            if "401" not in str(exc) or self._api_key is None:
                raise
            self.authenticate_node(self._api_key)
            return self.gw.request(verb, path, **kw)

    def identity_from_token(self, token: str) -> Identity:
        p = _decode_payload(token)
        ident = Identity(
            type=p["type"], id=p["sub"], organization_id=p.get("org"),
            collaboration_id=p.get("collab"), task_id=p.get("task_id"),
            image=p.get("image"),
        )
        # keep the raw token so calls made AS this identity (the NodeProxy
        # acting for a running algorithm container) carry the right bearer
        ident.token = token
        return ident

    def token_for_container(self, ident, task_id: int, image: str) -> dict:
        return self._request("POST", "token/container",
                             json={"task_id": task_id, "image": image})

    # ---------------------------------------------------------- node runtime
    def node_ping(self, ident, info: dict | None = None) -> None:
        self._request("POST", "ping", json={"info": info or {}})

    def node_set_offline(self, node_id: int) -> None:
        try:
            self._request("POST", "ping", json={"info": {}, "offline": True})
        except Exception:  # noqa: BLE001 — best-effort during shutdown
            pass

    def node_sync_open_runs(self, ident) -> list[dict]:
        return self._request("GET", "run", params={"state": "open"})["data"]

    def set_organization_public_key(self, ident, org_id: int,
                                    public_key: str) -> dict:
        return self._request("PATCH", f"organization/{org_id}",
                               json={"public_key": public_key})

    # ------------------------------------------------------- task/run plane
    def get_task(self, ident, task_id: int) -> dict:
        return self._request("GET", f"task/{task_id}", ident=ident)

    def list_runs(self, ident, **params) -> dict:
        params = {k: v for k, v in params.items() if v is not None}
        return self._request("GET", "run", ident=ident, params=params)

    def get_run(self, ident, run_id: int) -> dict:
        return self._request("GET", f"run/{run_id}", ident=ident)

    def patch_run(self, ident, run_id: int, **fields) -> dict:
        return self._request("PATCH", f"run/{run_id}", json=fields)

    def list_results(self, ident, task_id: int, **kw) -> dict:
        return self._request("GET", "result", ident=ident,
                             params={"task_id": task_id})

    def create_task(self, ident, **body) -> dict:
        return self._request("POST", "task", ident=ident, json=body)

    # --------------------------------------------------------------- lookups
    def organization_public_key(self, org_id: int) -> str:
        return self._request("GET", f"organization/{org_id}").get(
            "public_key") or ""

    def get_collaboration(self, ident, collab_id: int) -> dict:
        return self._request("GET", f"collaboration/{collab_id}", ident=ident)

    def algorithm_is_approved(self, image: str) -> bool:
        rows = self._request("GET", "algorithm",
                               params={"status": "approved",
                                       "per_page": 1000})["data"]
        return any(a["image"] == image for a in rows)
