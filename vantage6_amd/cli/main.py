"""`v6amd` CLI — lifecycle + dev + test + bench commands.

Reference parity: the ``v6`` console entrypoint:
``v6 server start`` → ``v6amd server``, ``v6 dev create/start-demo-network``
→ ``v6amd dev``, ``v6 test`` → ``v6amd test``, plus ``v6amd bench`` for the
MI355X benchmark harness. Config wizard: ``v6amd new`` writes a template
network YAML.
"""
from __future__ import annotations

import json
import sys
from pathlib import Path

import click

# repo root importable when invoked as a module
sys.path.insert(0, str(Path(__file__).resolve().parents[2]))


@click.group()
def cli():
    """vantage6_amd — MI355X-native federated learning infrastructure."""


@cli.command()
@click.option("--config", "config_path", required=True, type=click.Path())
@click.option("--stations", default=2, help="stations in the template")
def new(config_path, stations):
    """Write a template network YAML (configuration wizard analog)."""
    from ..common.config import (DatabaseConfig, NetworkConfig, ServerConfig,
                                 StationConfig, save_config)

    cfg = NetworkConfig(
        name=Path(config_path).stem,
        server=ServerConfig(),
        stations=[
            StationConfig(
                name=f"station-{i}",
                device="auto",
                databases=[DatabaseConfig(
                    label="default",
                    uri="synthetic://tabular?n=4096&d=32&seed=1",
                    type="synthetic",
                )],
            )
            for i in range(stations)
        ],
    )
    save_config(cfg, config_path)
    click.echo(f"wrote {config_path}")


@cli.command()
@click.option("--config", "config_path", type=click.Path(exists=True), default=None)
@click.option("--port", default=7601)
@click.option("--stations", default=0, help="also start N in-process stations")
def server(config_path, port, stations):
    """Start the orchestrator + REST façade (uvicorn)."""
    import uvicorn

    from ..cli.dev import DevNetwork
    from ..common.config import NetworkConfig, load_config
    from ..server.api import create_app
    from ..server.core import ServerCore

    if config_path:
        cfg = load_config(config_path)
        net = DevNetwork(
            n_stations=len(cfg.stations),
            encrypted=cfg.encrypted,
            databases=[s.databases for s in cfg.stations],
            devices=[s.device for s in cfg.stations],
            uri=cfg.server.uri,
            name=cfg.name,
        ).start()
        core, port = net.core, cfg.server.port
    elif stations:
        net = DevNetwork(n_stations=stations).start()
        core = net.core
    else:
        core = ServerCore()
        core.create_root_user()
    app = create_app(core)
    click.echo(f"orchestrator listening on 127.0.0.1:{port}/api (root/root)")
    uvicorn.run(app, host="127.0.0.1", port=port, log_level="info")


@cli.command()
@click.option("--config", "config_path", required=True,
              type=click.Path(exists=True), help="StationConfig YAML")
@click.option("--server", "server_url", required=True,
              help="orchestrator URL, e.g. http://host:7601")
@click.option("--api-key", envvar="V6AMD_API_KEY", required=True,
              help="node api-key (or env V6AMD_API_KEY)")
@click.option("--private-key", "private_key", type=click.Path(), default=None,
              help="org RSA key file for encrypted collaborations")
def node(config_path, server_url, api_key, private_key):
    """Start a data station against a REMOTE orchestrator (reference:
    ``v6 node start`` — the node-daemon deployment shape). Runs until
    interrupted."""
    import time as _time

    from ..common.config import StationConfig, load_config
    from ..node.remote import RemoteCore
    from ..node.station import Station

    cfg = load_config(config_path, model=StationConfig)
    core = RemoteCore(server_url)
    station = Station(core, cfg, api_key=api_key,
                      private_key_file=private_key)
    station.start()
    click.echo(f"station {cfg.name!r} online against {server_url} "
               f"(device {cfg.device}); ctrl-c to stop")
    try:
        while True:
            _time.sleep(1.0)
    except KeyboardInterrupt:
        pass
    finally:
        station.stop()
        core.close()


@cli.group()
def dev():
    """Demo-network commands (integration harness)."""


@dev.command("demo")
@click.option("--stations", default=2)
@click.option("--encrypted", is_flag=True)
def dev_demo(stations, encrypted):
    """Boot a demo network and run the federated column-mean through it."""
    import numpy as np
    import pandas as pd
    import tempfile

    from ..cli.dev import DevNetwork
    from ..common.config import DatabaseConfig

    tmp = Path(tempfile.mkdtemp(prefix="v6amd_demo_"))
    rng = np.random.default_rng(0)
    dbs = []
    for i in range(stations):
        df = pd.DataFrame({"age": rng.integers(20, 90, 100),
                           "bmi": rng.normal(25, 4, 100)})
        p = tmp / f"shard{i}.csv"
        df.to_csv(p, index=False)
        dbs.append([DatabaseConfig(label="default", uri=str(p), type="csv")])
    with DevNetwork(n_stations=stations, encrypted=encrypted, databases=dbs) as net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="demo column mean", image="algorithms.column_mean",
            input_={"method": "central_mean", "kwargs": {"columns": ["age", "bmi"]}},
        )
        res = net.client.wait_for_results(task["id"], timeout=120)
    click.echo(json.dumps({"demo": "column_mean", "result": res[0]}, indent=2))


@cli.command()
@click.option("--uri", default="sqlite://", help="database URI")
@click.option("--root-password", default="root")
def shell(uri, root_password):
    """Interactive shell with an orchestrator session bound (reference:
    ``vserver-local shell`` — IPython with the models; here a stdlib
    ``code.interact`` with ServerCore, the model classes and a root
    client in scope)."""
    import code

    from ..client import UserClient
    from ..server import model as model_module
    from ..server.core import ServerCore

    core = ServerCore(uri=uri)
    try:
        core.create_root_user("root", root_password)
    except Exception:  # noqa: BLE001 — existing DB keeps its root
        pass
    client = UserClient(core=core)
    client.authenticate("root", root_password)
    ns = {"core": core, "session": core.session, "client": client,
          "models": model_module}
    for name in ("User", "Node", "Organization", "Collaboration", "Task",
                 "Run", "Role", "Rule", "Study"):
        ns[name] = getattr(model_module, name)
    banner = ("vantage6_amd shell — objects: core, session, client, models, "
              "User/Node/Organization/Collaboration/Task/Run/Role/Rule/Study")
    code.interact(banner=banner, local=ns)


@cli.command(name="create-private-key")
@click.option("--output", "output_path", required=True, type=click.Path(),
              help="where to write the RSA private key")
@click.option("--bits", default=2048)
def create_private_key(output_path, bits):
    """Generate an organization RSA keypair (reference:
    ``v6 node create-private-key``). The public key is printed so it can be
    uploaded via ``client.organization`` / the station does it on start."""
    from ..common.encryption import RSACryptor

    cryptor = RSACryptor(output_path, bits=bits)
    click.echo(f"private key written to {output_path}")
    click.echo(f"public key (b64):\n{cryptor.public_key_str}")


@cli.command(name="import")
@click.option("--config", "config_path", required=True,
              type=click.Path(exists=True), help="entity fixture YAML")
@click.option("--uri", default="sqlite://", help="orchestrator database URI")
@click.option("--root-password", default="root")
def import_entities(config_path, uri, root_password):
    """Load organizations/collaborations/users/nodes from a YAML fixture
    (reference: ``v6 server import`` — SURVEY.md §3.6). Prints created
    node api-keys so stations can be configured."""
    import yaml as _yaml

    from ..client import UserClient
    from ..server.core import ServerCore

    with open(config_path) as fh:
        fixture = _yaml.safe_load(fh) or {}
    # fixture import is admin-controlled bootstrap (reference `v6 server
    # import` also seeds literal YAML passwords) — no interactive policy
    core = ServerCore(uri=uri, password_policy=False)
    core.create_root_user("root", root_password)
    root = UserClient(core=core)
    root.authenticate("root", root_password)
    org_ids = {}
    for org in fixture.get("organizations", []):
        created = root.organization.create(name=org["name"],
                                           public_key=org.get("public_key", ""))
        org_ids[org["name"]] = created["id"]
        click.echo(f"organization {org['name']!r}: id {created['id']}")
    collab_ids = {}
    for col in fixture.get("collaborations", []):
        created = root.collaboration.create(
            name=col["name"],
            organization_ids=[org_ids[n] for n in col["organizations"]],
            encrypted=col.get("encrypted", False))
        collab_ids[col["name"]] = created["id"]
        click.echo(f"collaboration {col['name']!r}: id {created['id']}")
    for user in fixture.get("users", []):
        root.user.create(username=user["username"], password=user["password"],
                         organization_id=org_ids[user["organization"]],
                         roles=user.get("roles", ["Researcher"]))
        click.echo(f"user {user['username']!r} created")
    for node in fixture.get("nodes", []):
        created = root.node.create(name=node["name"],
                                   collaboration_id=collab_ids[node["collaboration"]],
                                   organization_id=org_ids[node["organization"]])
        click.echo(f"node {node['name']!r}: api_key {created['api_key']}")
    core.close()


@cli.command()
@click.option("--stations", default=2)
def test(stations):
    """Run the diagnostics workload against a fresh demo network."""
    from ..cli.dev import DevNetwork

    with DevNetwork(n_stations=stations) as net:
        task = net.client.task.create(
            collaboration=net.collaboration["id"],
            organizations=[net.org_ids[0]],
            name="diagnostics", image="algorithms.diagnostics",
            input_={"method": "central_diagnostics"},
        )
        report = net.client.wait_for_results(task["id"], timeout=120)[0]
    click.echo(json.dumps(report, indent=2, default=str))
    sys.exit(0 if report.get("ok") else 1)


@cli.group()
def algorithm():
    """Algorithm-package tools (`v6 algorithm` analog)."""


@algorithm.command("create")
@click.option("--name", required=True, help="package name (python identifier)")
@click.option("--dir", "out_dir", default="algorithms", type=click.Path())
def algorithm_create(name, out_dir):
    """Scaffold a new algorithm package (central/partial pair + mock test)."""
    pkg = Path(out_dir) / f"{name}.py"
    if pkg.exists():
        raise click.ClickException(f"{pkg} exists")
    pkg.parent.mkdir(parents=True, exist_ok=True)
    pkg.write_text(f'''"""Algorithm package `{name}` — scaffolded by v6amd algorithm create."""
from vantage6_amd.algorithm import algorithm_client, data


@algorithm_client
def central_{name}(client, **kwargs):
    """Central part: fan out partials and aggregate."""
    org_ids = [org["id"] for org in client.organization.list()]
    task = client.task.create(
        input_={{"method": "partial_{name}", "kwargs": kwargs}},
        organizations=org_ids,
    )
    results = client.wait_for_results(task["id"])
    return {{"n_partials": len(results), "partials": results}}


@data(1)
def partial_{name}(df, **kwargs):
    """Partial part: computed against this station's local data."""
    return {{"n_rows": int(len(df))}}
''')
    test = Path("tests") / f"test_algo_{name}.py"
    if not test.exists() and Path("tests").is_dir():
        test.write_text(f'''"""Mock-client test for `{name}` (scaffolded)."""
from vantage6_amd.algorithm import MockAlgorithmClient


def test_{name}_mock(tmp_path):
    import pandas as pd

    p = tmp_path / "d.csv"
    pd.DataFrame({{"x": [1, 2, 3]}}).to_csv(p, index=False)
    mock = MockAlgorithmClient(
        datasets=[[{{"label": "default", "database": str(p), "type": "csv"}}]],
        module="algorithms.{name}",
    )
    task = mock.task.create(input_={{"method": "central_{name}"}}, organizations=[1])
    out = mock.result.from_task(task["id"])[0]
    assert out["n_partials"] == 1
''')
    click.echo(f"scaffolded {pkg}" + (f" and {test}" if test.exists() else ""))


@cli.command()
@click.option("--model", default="resnet18")
@click.option("--steps", default=20)
@click.option("--warmup", default=5)
@click.option("--device", default=None)
def bench(model, steps, warmup, device):
    """Run the flagship benchmark (wrapper over bench.py)."""
    import subprocess

    cmd = [sys.executable, str(Path(__file__).resolve().parents[2] / "bench.py"),
           "--model", model, "--steps", str(steps), "--warmup", str(warmup)]
    if device:
        cmd += ["--device", device]
    raise SystemExit(subprocess.call(cmd))


if __name__ == "__main__":
    cli()
