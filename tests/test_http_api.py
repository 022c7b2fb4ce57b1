"""REST façade tests: the vantage6 client schema over real HTTP.

Boots the orchestrator + stations in-process, serves the FastAPI façade on
a uvicorn thread, and drives the full federated column-mean through
``UserClient(url=...)`` — the same researcher code as in-process mode.
"""
import threading
import time

import numpy as np
import pandas as pd
import pytest
import requests
import uvicorn

from vantage6_amd.cli.dev import DevNetwork
from vantage6_amd.client import UserClient
from vantage6_amd.common.config import DatabaseConfig
from vantage6_amd.server.api import create_app

PORT = 7655


@pytest.fixture
def net(tmp_path):
    rng = np.random.default_rng(3)
    paths = []
    for i in range(2):
        df = pd.DataFrame({"v": rng.normal(10, 2, 40)})
        p = tmp_path / f"s{i}.csv"
        df.to_csv(p, index=False)
        paths.append(p)
    network = DevNetwork(
        n_stations=2,
        databases=[[DatabaseConfig(label="default", uri=str(p), type="csv")]
                   for p in paths],
        name="httpnet",
    )
    network.start()
    yield network
    network.stop()


@pytest.fixture
def server(net):
    app = create_app(net.core)
    net.http_app = app  # tests reach app.state (mail outbox)
    config = uvicorn.Config(app, host="127.0.0.1", port=PORT, log_level="error")
    srv = uvicorn.Server(config)
    th = threading.Thread(target=srv.run, daemon=True)
    th.start()
    for _ in range(100):
        try:
            requests.get(f"http://127.0.0.1:{PORT}/api/version", timeout=1)
            break
        except Exception:
            time.sleep(0.05)
    yield f"http://127.0.0.1:{PORT}"
    srv.should_exit = True
    th.join(timeout=5)


def test_version_health(server):
    v = requests.get(f"{server}/api/version").json()
    assert "version" in v
    h = requests.get(f"{server}/api/health").json()
    assert h["status"] == "ok"


def test_auth_required(server):
    r = requests.get(f"{server}/api/task")
    assert r.status_code == 401


def test_bad_login(server):
    r = requests.post(f"{server}/api/token/user",
                      json={"username": "nobody", "password": "x"})
    assert r.status_code == 401


def test_e2e_column_mean_over_http(server, net):
    client = UserClient(url=server)
    client.authenticate("httpnet-researcher", "password")
    task = client.task.create(
        collaboration=net.collaboration["id"],
        organizations=[net.org_ids[0]],
        name="http column mean",
        image="algorithms.column_mean",
        input_={"method": "central_mean", "kwargs": {"columns": ["v"]}},
    )
    assert task["status"] == "pending"
    results = client.wait_for_results(task["id"], interval=0.05, timeout=60)
    assert results[0]["v"] == pytest.approx(10, abs=2)
    # run listing + logs over HTTP
    runs = client.run.from_task(task["id"])
    assert runs and runs[0]["status"] == "completed"


def test_token_refresh_roundtrip(server):
    r = requests.post(f"{server}/api/token/user",
                      json={"username": "root", "password": "root"})
    tokens = r.json()
    rr = requests.post(f"{server}/api/token/refresh",
                       json={"refresh_token": tokens["refresh_token"]})
    assert rr.status_code == 200 and "access_token" in rr.json()


def test_node_token_and_open_runs(server, net):
    # nodes authenticate with api keys over HTTP too
    root = UserClient(url=server)
    root.authenticate("root", "root")
    node = root.node.create(name="extra-node",
                            collaboration_id=net.collaboration["id"],
                            organization_id=net.org_ids[0])
    r = requests.post(f"{server}/api/token/node", json={"api_key": node["api_key"]})
    assert r.status_code == 200
    tok = r.json()["access_token"]
    rr = requests.get(f"{server}/api/run", params={"state": "open"},
                      headers={"Authorization": f"Bearer {tok}"})
    assert rr.status_code == 200


def test_websocket_event_push(net):
    """Socket.IO-parity: a websocket subscriber receives new_task events."""
    from starlette.testclient import TestClient

    from vantage6_amd.server.api import create_app

    app = create_app(net.core)
    tok = net.client.gw.token if hasattr(net.client.gw, "token") else None
    client = TestClient(app)
    with client.websocket_connect(f"/api/ws?token={tok}") as ws:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="ws-test", image="algorithms.column_mean",
            input_={"method": "partial_mean", "kwargs": {"columns": ["v"]}},
        )
        # researcher subscribes to the collaboration room: receives the run
        # status stream (initializing -> active -> completed); new_task goes
        # to the per-organization room (the stations'), like the reference
        statuses = []
        for _ in range(10):
            msg = ws.receive_json()
            if msg["event"] == "algorithm_status_change":
                statuses.append(msg["data"]["status"])
                if msg["data"]["status"] == "completed":
                    break
        assert "completed" in statuses
        net.client.wait_for_results(task["id"], timeout=30)


def test_websocket_rejects_bad_token(net):
    from starlette.testclient import TestClient

    from vantage6_amd.server.api import create_app

    client = TestClient(create_app(net.core))
    import pytest as _pytest

    with _pytest.raises(Exception):
        with client.websocket_connect("/api/ws?token=garbage") as ws:
            ws.receive_json()


def test_external_container_via_wrap_and_http(server, net, tmp_path):
    """A.3 contract end-to-end as an *external* process would run it:
    INPUT/TOKEN/OUTPUT files + env map + AlgorithmClient over HTTP."""
    from vantage6_amd.algorithm.wrap import wrap_algorithm
    from vantage6_amd.common.serialization import deserialize, serialize

    # a node forges a container token for a running task (as the station does)
    core = net.core
    root = UserClient(core=core)
    root.authenticate("root", "root")
    task = net.client.task.create(
        collaboration=net.collaboration["id"], organizations=[net.org_ids[0]],
        name="host-task", image="algorithms.column_mean",
        input_={"method": "partial_mean", "kwargs": {"columns": ["v"]}},
    )
    net.stations[0].join_idle()
    node_ident = net.stations[0].identity
    ctok = core.token_for_container(node_ident, task["id"],
                                    "algorithms.column_mean")["container_token"]

    input_file = tmp_path / "input"
    output_file = tmp_path / "output"
    token_file = tmp_path / "token"
    input_file.write_bytes(serialize(
        {"method": "central_mean", "kwargs": {"columns": ["v"]}}))
    token_file.write_text(ctok)
    env = {
        "INPUT_FILE": str(input_file), "OUTPUT_FILE": str(output_file),
        "TOKEN_FILE": str(token_file), "PKG_NAME": "algorithms.column_mean",
        "HOST": "http://127.0.0.1", "PORT": str(PORT), "API_PATH": "/api",
    }
    wrap_algorithm(environ=env)
    out = deserialize(output_file.read_bytes())
    assert out["v"] == pytest.approx(10, abs=2)


def test_role_and_rule_listings(server):
    root = UserClient(url=server)
    root.authenticate("root", "root")
    roles = root.gw.request("GET", "role")
    names = {r["name"] for r in roles["data"]}
    assert {"Root", "Organization-Admin", "Researcher", "Viewer"} <= names
    rules = root.gw.request("GET", "rule", params={"per_page": 500})
    assert rules["links"]["total"] >= 200  # resources × scopes × operations
    assert {"name", "scope", "operation"} <= set(rules["data"][0])


def test_algorithm_store_http(server):
    root = UserClient(url=server)
    root.authenticate("root", "root")
    a = root.gw.request("POST", "algorithm", json={
        "name": "diag", "image": "algorithms.diagnostics"})
    assert a["status"] == "submitted"
    r = root.gw.request("POST", f"algorithm/{a['id']}/review",
                        json={"approve": True, "comment": "ok"})
    assert r["status"] == "approved"
    listed = root.gw.request("GET", "algorithm", params={"status": "approved"})
    assert any(x["image"] == "algorithms.diagnostics" for x in listed["data"])


def test_recover_password_and_mfa_over_http(server, net):
    net.root_client.user.create("webrec", "old", organization_id=net.org_ids[0],
                                roles=["Researcher"], email="w@example.org")
    r = requests.post(f"{server}/api/recover/lost", json={"username": "webrec"})
    assert r.status_code == 200
    outbox = net.http_app.state.mail.outbox
    token = outbox[-1]["body"].split(": ")[-1]
    r = requests.post(f"{server}/api/recover/reset",
                      json={"token": token, "password": "new"})
    assert r.status_code == 200
    r = requests.post(f"{server}/api/token/user",
                      json={"username": "webrec", "password": "new"})
    assert r.status_code == 200
    # 2FA reset round-trip
    r = requests.post(f"{server}/api/recover/2fa/lost", json={"username": "webrec"})
    assert r.status_code == 200
    token = net.http_app.state.mail.outbox[-1]["body"].split(": ")[-1]
    r = requests.post(f"{server}/api/recover/2fa/reset", json={"token": token})
    assert r.status_code == 200 and "mfa_secret" in r.json()


def test_column_discovery_over_http(server, net):
    # trigger an immediate heartbeat so column info is on the Node row
    for st in net.stations:
        st.core.node_ping(st.identity, {"columns": st._database_columns()})
    tok = requests.post(f"{server}/api/token/user",
                        json={"username": "root", "password": "root"}).json()
    h = {"Authorization": f"Bearer {tok['access_token']}"}
    nodes = requests.get(f"{server}/api/node", headers=h).json()["data"]
    r = requests.get(f"{server}/api/column/{nodes[0]['id']}", headers=h)
    assert r.status_code == 200 and r.json() == ["v"]


def test_remote_station_over_http(tmp_path):
    """Full reference deployment shape: orchestrator behind HTTP, a Station
    attached via RemoteCore (api-key auth, event/polling pickup, result
    PATCH), researcher over the HTTP client — column mean end-to-end with
    NOTHING sharing memory with the server core."""
    from vantage6_amd.client import UserClient
    from vantage6_amd.common.config import DatabaseConfig, StationConfig
    from vantage6_amd.node.remote import RemoteCore
    from vantage6_amd.node.station import Station
    from vantage6_amd.server.core import ServerCore

    df = pd.DataFrame({"v": [1.0, 2.0, 3.0, 4.0]})
    csv = tmp_path / "remote.csv"
    df.to_csv(csv, index=False)

    core = ServerCore(password_policy=False)
    core.create_root_user("root", "root")
    root = UserClient(core=core)
    root.authenticate("root", "root")
    org = root.organization.create(name="remote-org")
    collab = root.collaboration.create(name="remote-c",
                                       organization_ids=[org["id"]])
    node = root.node.create(name="remote-n", collaboration_id=collab["id"],
                            organization_id=org["id"])

    app = create_app(core)
    port = PORT + 7
    config = uvicorn.Config(app, host="127.0.0.1", port=port, log_level="error")
    srv = uvicorn.Server(config)
    th = threading.Thread(target=srv.run, daemon=True)
    th.start()
    url = f"http://127.0.0.1:{port}"
    for _ in range(100):
        try:
            requests.get(f"{url}/api/version", timeout=1)
            break
        except Exception:
            time.sleep(0.05)

    remote = RemoteCore(url)
    remote.events.interval = 0.1  # fast polling for the test
    station = Station(remote, StationConfig(
        name="remote-station", organization_id=org["id"], device="cpu",
        databases=[DatabaseConfig(label="default", uri=str(csv), type="csv")]),
        api_key=node["api_key"])
    try:
        station.start()
        web = UserClient(url=url)
        web.authenticate("root", "root")
        # CENTRAL task: exercises the proxy's subtask fan-out over HTTP
        # with the CONTAINER token (a node bearer would be rejected)
        task = web.task.create(
            collaboration=collab["id"], organizations=[org["id"]],
            name="remote-mean", image="algorithms.column_mean",
            input_={"method": "central_mean", "kwargs": {"columns": ["v"]}})
        (res,) = web.wait_for_results(task["id"], timeout=30)
        assert res["v"] == pytest.approx(2.5)
    finally:
        station.stop()
        remote.close()
        srv.should_exit = True
        th.join(timeout=5)


def test_encrypted_e2e_over_http(tmp_path):
    """Encrypted collaboration with the researcher on the HTTP client:
    key upload via PATCH /organization, per-org hybrid-RSA task input,
    station decrypt, result encrypted back to the initiating org."""
    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.client import UserClient
    from vantage6_amd.common.config import DatabaseConfig

    df = pd.DataFrame({"v": [2.0, 4.0, 6.0]})
    csv = tmp_path / "enc.csv"
    df.to_csv(csv, index=False)
    net = DevNetwork(
        n_stations=1, encrypted=True,
        databases=[[DatabaseConfig(label="default", uri=str(csv), type="csv")]],
        name="enc-http")
    net.start()
    app = create_app(net.core)
    port = PORT + 11
    srv = uvicorn.Server(uvicorn.Config(app, host="127.0.0.1", port=port,
                                        log_level="error"))
    th = threading.Thread(target=srv.run, daemon=True)
    th.start()
    url = f"http://127.0.0.1:{port}"
    for _ in range(100):
        try:
            requests.get(f"{url}/api/version", timeout=1)
            break
        except Exception:
            time.sleep(0.05)
    try:
        web = UserClient(url=url)
        web.authenticate("enc-http-researcher", "password")
        web.setup_encryption(str(net._tmp / "station0.key"), key_bits=1024)
        task = web.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="enc-mean", image="algorithms.column_mean",
            input_={"method": "partial_mean", "kwargs": {"columns": ["v"]}})
        (res,) = web.wait_for_results(task["id"], timeout=30)
        assert res["sums"]["v"] == pytest.approx(12.0)
        # the stored result on the server is NOT plaintext-b64 (it is the
        # hybrid envelope: key$nonce$mac$ciphertext)
        raw = net.root_client.run.from_task(task["id"])[0]
        run_full = net.root_client.run.get(raw["id"])
        assert "$" in (run_full["result"] or "")
    finally:
        net.stop()
        srv.should_exit = True
        th.join(timeout=5)


def test_remote_core_reauth_after_token_loss(tmp_path):
    """RemoteCore must recover from a dead/expired token by re-authenticating
    with its api-key (reference: node re-auth on expired_token)."""
    from vantage6_amd.client import UserClient
    from vantage6_amd.node.remote import RemoteCore
    from vantage6_amd.server.core import ServerCore

    core = ServerCore(password_policy=False)
    core.create_root_user("root", "root")
    root = UserClient(core=core)
    root.authenticate("root", "root")
    org = root.organization.create(name="reauth-org")
    collab = root.collaboration.create(name="reauth-c",
                                       organization_ids=[org["id"]])
    node = root.node.create(name="reauth-n", collaboration_id=collab["id"],
                            organization_id=org["id"])
    app = create_app(core)
    port = PORT + 13
    srv = uvicorn.Server(uvicorn.Config(app, host="127.0.0.1", port=port,
                                        log_level="error"))
    th = threading.Thread(target=srv.run, daemon=True)
    th.start()
    url = f"http://127.0.0.1:{port}"
    for _ in range(100):
        try:
            requests.get(f"{url}/api/version", timeout=1)
            break
        except Exception:
            time.sleep(0.05)
    try:
        remote = RemoteCore(url)
        remote.authenticate_node(node["api_key"])
        assert remote.node_sync_open_runs(None) == []
        # simulate token loss: both tokens invalidated
        remote.gw.token = "garbage"
        remote.gw.refresh_token_value = "garbage"
        assert remote.node_sync_open_runs(None) == []  # re-auths transparently
        assert remote.gw.token != "garbage"
        # store policy check with a node bearer over HTTP
        assert remote.algorithm_is_approved("algorithms.column_mean") is False
    finally:
        srv.should_exit = True
        th.join(timeout=5)


def test_pagination_link_headers(server, net):
    tok = requests.post(f"{server}/api/token/user",
                        json={"username": "root", "password": "root"}).json()
    h = {"Authorization": f"Bearer {tok['access_token']}"}
    for _ in range(3):
        net.root_client.task.create(
            collaboration=net.collaboration["id"], organizations=[net.org_ids[0]],
            name="pg", image="algorithms.column_mean",
            input_={"method": "partial_mean", "kwargs": {"columns": ["v"]}})
    r = requests.get(f"{server}/api/task", params={"per_page": 2, "page": 1},
                     headers=h)
    assert r.status_code == 200
    assert 'rel="next"' in r.headers.get("Link", "")
    assert int(r.headers["Total-Count"]) >= 3


def test_pagination_link_headers(server):
    """GET listings carry RFC 8288 Link headers (first/last/next/prev) and
    X-Total style links block in the body, like the reference's Pagination
    helper."""
    client = UserClient(url=server)
    client.authenticate("root", "root")
    for i in range(5):
        client.organization.create(name=f"pg-http-{i}")
    tok = client.gw.token
    r = requests.get(f"{server}/api/organization",
                     params={"page": 2, "per_page": 2},
                     headers={"Authorization": f"Bearer {tok}"}, timeout=5)
    assert r.status_code == 200
    body = r.json()
    assert len(body["data"]) == 2
    assert body["links"]["page"] == 2
    link = r.headers.get("link", "")
    assert 'rel="first"' in link and 'rel="last"' in link
    assert 'rel="next"' in link and 'rel="prev"' in link
    assert "page=3" in link and "page=1" in link


def test_permission_errors_are_403_not_500_on_user_study_listings(server, net):
    """user/study listings previously bypassed the exception-mapping
    wrapper: a core PermissionError_ surfaced as 500."""
    core = net.core
    from vantage6_amd.client import UserClient as UC
    root = UC(core=core)
    root.authenticate("root", "root")
    org = root.organization.create(name="noperm-org")
    root.user.create("noperm", "pw", organization_id=org["id"], roles=[])
    http = UserClient(url=server)
    http.authenticate("noperm", "pw")
    for path in ("user", "study", "study/999"):
        r = requests.get(f"{server}/api/{path}",
                         headers={"Authorization": f"Bearer {http.gw.token}"},
                         timeout=5)
        assert r.status_code in (401, 403, 404), (path, r.status_code)


def test_all_entity_routes_map_core_errors(server, net):
    """Every entity route maps core exceptions: 404 for missing ids, never
    500 (user/node patch+delete, collaboration get, 2fa reset, columns)."""
    http = UserClient(url=server)
    http.authenticate("root", "root")
    hdr = {"Authorization": f"Bearer {http.gw.token}"}
    checks = [
        ("GET", "user/99999", None),
        ("PATCH", "user/99999", {"email": "x@y.z"}),
        ("DELETE", "user/99999", None),
        ("PATCH", "node/99999", {"name": "n"}),
        ("DELETE", "node/99999", None),
        ("GET", "collaboration/99999", None),
        ("GET", "column/99999?label=default", None),
    ]
    for verb, path, body in checks:
        r = requests.request(verb, f"{server}/api/{path}", json=body,
                             headers=hdr, timeout=5)
        assert r.status_code in (400, 401, 403, 404), (verb, path, r.status_code)
    r = requests.post(f"{server}/api/recover/2fa/reset",
                      json={"token": "garbage.token.here"}, timeout=5)
    assert r.status_code in (400, 401), r.status_code


def test_auth_rate_limit_429(net):
    """Credential endpoints are per-IP rate limited (429 + Retry-After)
    upstream of the per-account lockout."""
    from starlette.testclient import TestClient

    from vantage6_amd.server.api import create_app

    app = create_app(net.core, auth_rate_limit=5, auth_rate_window=60.0)
    client = TestClient(app)
    codes = []
    for _ in range(8):
        r = client.post("/api/token/user",
                        json={"username": "nosuch", "password": "x"})
        codes.append(r.status_code)
    assert codes[:5] == [401] * 5
    assert codes[5:] == [429] * 3
    assert "retry-after" in {k.lower() for k in r.headers}
    # non-auth endpoints are not limited
    assert client.get("/api/version").status_code == 200


def test_entity_crud_over_http(server):
    """The new PATCH/DELETE routes work over the HTTP gateway too."""
    http = UserClient(url=server)
    http.authenticate("root", "root")
    org = http.organization.create(name="http-crud-org")
    upd = http.organization.update(org["id"], name="http-crud-org2",
                                   country="DE")
    assert upd["name"] == "http-crud-org2" and upd["country"] == "DE"
    c = http.collaboration.create(name="http-crud-c",
                                  organization_ids=[org["id"]])
    assert http.collaboration.update(c["id"], name="http-crud-c2")["name"] \
        == "http-crud-c2"
    st = http.study.create(name="http-st", collaboration_id=c["id"],
                           organization_ids=[org["id"]])
    http.study.update(st["id"], name="http-st2")
    http.study.delete(st["id"])
    n = http.node.create(name="http-n", collaboration_id=c["id"],
                         organization_id=org["id"])
    rot = http.node.reset_api_key(n["id"])
    assert rot["api_key"] and rot["id"] == n["id"]
    # guarded delete over HTTP: 400 until forced
    import pytest as _pytest
    with _pytest.raises(RuntimeError, match="dependents"):
        http.collaboration.delete(c["id"])
    http.collaboration.delete(c["id"], delete_dependents=True)
    # give the org a dependent of its own (the collab cascade removed its
    # node) so the org guard actually fires
    http.user.create("http-crud-u", "Str0ngPass1", organization_id=org["id"],
                     roles=["Researcher"])
    with _pytest.raises(RuntimeError, match="dependents"):
        http.organization.delete(org["id"])
    http.organization.delete(org["id"], delete_dependents=True)
