"""Diagnostics workload + CLI tests (click CliRunner, reference test style)."""
from click.testing import CliRunner

from vantage6_amd.cli.main import cli


def test_diagnostics_via_cli():
    r = CliRunner().invoke(cli, ["test", "--stations", "2"])
    assert r.exit_code == 0, r.output
    assert '"ok": true' in r.output


def test_new_config_template(tmp_path):
    p = tmp_path / "net.yaml"
    r = CliRunner().invoke(cli, ["new", "--config", str(p), "--stations", "3"])
    assert r.exit_code == 0, r.output
    from vantage6_amd.common.config import load_config

    cfg = load_config(p)
    assert len(cfg.stations) == 3
    assert cfg.stations[0].databases[0].type == "synthetic"


def test_demo_network_cli():
    r = CliRunner().invoke(cli, ["dev", "demo", "--stations", "2"])
    assert r.exit_code == 0, r.output
    assert "column_mean" in r.output


def test_algorithm_scaffold(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    (tmp_path / "tests").mkdir()
    r = CliRunner().invoke(cli, ["algorithm", "create", "--name", "my_count"])
    assert r.exit_code == 0, r.output
    assert (tmp_path / "algorithms" / "my_count.py").exists()
    assert (tmp_path / "tests" / "test_algo_my_count.py").exists()
    src = (tmp_path / "algorithms" / "my_count.py").read_text()
    assert "central_my_count" in src and "partial_my_count" in src


def test_shell_command_exists_and_binds_namespace(monkeypatch):
    """`v6amd shell` (reference: vserver-local shell) drops into an
    interactive session with core/session/client/models bound."""
    import code as code_module

    from click.testing import CliRunner

    from vantage6_amd.cli.main import cli

    captured = {}

    def fake_interact(banner=None, local=None):
        captured.update(local or {})

    monkeypatch.setattr(code_module, "interact", fake_interact)
    r = CliRunner().invoke(cli, ["shell"])
    assert r.exit_code == 0, r.output
    assert {"core", "session", "client", "models", "Task", "Run"} <= set(captured)
    # the bound client is authenticated and usable
    assert captured["client"].whoami["username"] == "root"


def test_every_cli_command_help_works():
    """--help for every command/group imports and renders (catches broken
    imports in command bodies without running them)."""
    from click.testing import CliRunner

    from vantage6_amd.cli.main import cli

    runner = CliRunner()
    top = runner.invoke(cli, ["--help"])
    assert top.exit_code == 0, top.output
    import re

    section = top.output.split("Commands:", 1)[1]
    cmds = re.findall(r"^  ([a-z][a-z0-9-]*)", section, re.M)
    assert len(cmds) >= 8, cmds
    for c in cmds:
        r = runner.invoke(cli, [c, "--help"])
        assert r.exit_code == 0, (c, r.output)


def test_examples_run():
    """examples/ scripts execute end-to-end on CPU."""
    import subprocess
    import sys
    from pathlib import Path

    root = Path(__file__).resolve().parents[1]
    for name in ("quickstart.py", "federated_glm.py", "fedavg_training.py"):
        r = subprocess.run([sys.executable, str(root / "examples" / name)],
                           capture_output=True, text=True, timeout=240)
        assert r.returncode == 0, (name, r.stderr[-1500:])
