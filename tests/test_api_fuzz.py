"""API fuzz: random/malformed bodies and params must never produce a 500
— every core exception maps to a 4xx (the wrapper contract)."""
import random

import pytest

from vantage6_amd.server.api import create_app
from vantage6_amd.server.core import ServerCore


@pytest.mark.timeout(300)
def test_no_endpoint_returns_500():
    from starlette.testclient import TestClient

    core = ServerCore(password_policy=False)
    core.create_root_user("root", "root")
    app = create_app(core, auth_rate_limit=0)  # no limiter in the fuzz
    client = TestClient(app, raise_server_exceptions=False)
    tok = client.post("/api/token/user",
                      json={"username": "root", "password": "root"}
                      ).json()["access_token"]
    hdr = {"Authorization": f"Bearer {tok}"}
    rng = random.Random(13)

    def rnd_value(depth=0):
        choices = [
            lambda: rng.randint(-2**40, 2**40),
            lambda: rng.choice(["", "x", "0", "-1", "ñ€", "a" * 300,
                                "<script>", "../../etc/passwd", "null"]),
            lambda: rng.choice([True, False, None]),
            lambda: rng.random() * 1e6,
        ]
        if depth < 2:
            choices.append(lambda: [rnd_value(depth + 1)
                                    for _ in range(rng.randint(0, 3))])
            choices.append(lambda: {rng.choice(["id", "name", "x", "input"]):
                                    rnd_value(depth + 1)
                                    for _ in range(rng.randint(0, 3))})
        return rng.choice(choices)()

    routes = []
    for r in app.routes:
        methods = getattr(r, "methods", None)
        path = getattr(r, "path", "")
        if not methods or not path.startswith("/api"):
            continue
        for m in methods - {"HEAD", "OPTIONS"}:
            routes.append((m, path))
    assert len(routes) > 40

    bad = []
    for trial in range(600):
        m, path = rng.choice(routes)
        # fill path params with junk ids
        url = path
        for seg in ("task_id", "run_id", "org_id", "user_id", "node_id",
                    "collab_id", "study_id", "role_id", "rule_id",
                    "algorithm_id"):
            url = url.replace("{%s}" % seg,
                              str(rng.choice([0, 1, -5, 999999, "abc"])))
        kwargs = {"headers": hdr if rng.random() < 0.8 else
                  {"Authorization": "Bearer garbage.token.here"}}
        if m in ("POST", "PATCH"):
            kwargs["json"] = rnd_value()
        if rng.random() < 0.4:
            kwargs["params"] = {rng.choice(["page", "per_page", "state",
                                            "name", "task_id"]): rnd_value()}
        resp = client.request(m, url, **kwargs)
        if resp.status_code >= 500:
            bad.append((m, url, kwargs.get("json"), resp.status_code,
                        resp.text[:200]))
    assert not bad, bad[:5]
    core.close()


@pytest.mark.timeout(300)
def test_station_survives_garbage_tasks():
    """Malformed task inputs (junk b64, non-dict payloads, wrong kwarg
    types, unknown methods/images) mark their runs failed — and the
    station worker survives to run a good task afterwards."""
    import base64 as b64
    import time

    from vantage6_amd.cli.dev import DevNetwork
    from vantage6_amd.common.config import DatabaseConfig
    from vantage6_amd.common.serialization import serialize

    dbs = [[DatabaseConfig(label="default",
                           uri="synthetic://tabular?n=64&d=4&seed=0",
                           type="synthetic")]]
    with DevNetwork(n_stations=1, databases=dbs, name="fuzznet") as net:
        cl = net.client
        cases = [
            ("not even base64 °°°", "algorithms.column_mean"),
            (b64.b64encode(b"not json").decode(), "algorithms.column_mean"),
            (b64.b64encode(serialize([1, 2, 3])).decode(),
             "algorithms.column_mean"),                      # non-dict payload
            (b64.b64encode(serialize({"no_method": 1})).decode(),
             "algorithms.column_mean"),
            (b64.b64encode(serialize({"method": "partial_mean",
                                      "kwargs": {"columns": 42}})).decode(),
             "algorithms.column_mean"),                      # wrong kwarg type
            (b64.b64encode(serialize({"method": "nope"})).decode(),
             "algorithms.column_mean"),
            (b64.b64encode(serialize({"method": "partial_mean"})).decode(),
             "algorithms.does_not_exist"),                   # bad image
        ]
        ids = []
        for i, (inp, image) in enumerate(cases):
            t = cl.task.create(collaboration=net.collaboration["id"],
                               organizations=[{"id": net.org_ids[0],
                                               "input": inp}],
                               name=f"junk{i}", image=image, input_=None)
            ids.append(t["id"])
        deadline = time.monotonic() + 60
        while time.monotonic() < deadline:
            terminal = {"failed", "no image", "crashed", "not allowed",
                        "start failed", "non-existing algorithm",
                        "killed by user"}
            st = {cl.task.get(tid)["status"] for tid in ids}
            if st <= terminal:
                break
            time.sleep(0.1)
        for tid in ids:
            status = cl.task.get(tid)["status"]
            assert status in terminal, (tid, status)
        # worker is still alive: a good task completes
        good = cl.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]], name="good",
            image="algorithms.fedavg",
            input_={"method": "partial_fit", "kwargs": {
                "model": "logreg",
                "model_kwargs": {"in_features": 4, "num_classes": 2},
                "local_steps": 1, "batch_size": 16}})
        (res,) = cl.wait_for_results(good["id"], timeout=60)
        assert "n" in res or res
