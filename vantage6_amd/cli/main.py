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
