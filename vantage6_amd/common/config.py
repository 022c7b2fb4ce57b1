"""YAML configuration management.

Reference parity: ``vantage6-common/vantage6/common/configuration_manager.py``
and ``.../context.py`` — load/validate/store YAML configs and
resolve data/log directories. Validation uses pydantic (available in this
image) instead of the reference's ``schema`` package.
"""
from __future__ import annotations

from pathlib import Path
from typing import Any

import yaml
from pydantic import BaseModel, Field


class DatabaseConfig(BaseModel):
    """A data source mounted at a station (node config `databases:` entry)."""

    label: str = "default"
    uri: str
    type: str = "csv"  # csv | parquet | excel | sql | tensor | synthetic


class StationConfig(BaseModel):
    """One data station (the reference's node YAML)."""

    name: str
    api_key: str = ""
    organization_id: int | None = None
    device: str = "auto"  # "auto" | "cpu" | "cuda:<i>" (HIP device on ROCm)
    databases: list[DatabaseConfig] = Field(default_factory=list)
    policies: dict[str, Any] = Field(default_factory=dict)  # allowed_algorithms etc.


class ServerConfig(BaseModel):
    """Orchestrator config (the reference's server YAML)."""

    ip: str = "127.0.0.1"
    port: int = 7601
    api_path: str = "/api"
    uri: str = "sqlite://"  # task/run store; ":memory:" or a file path
    jwt_secret: str = ""
    token_expires_seconds: float = 21600.0


class KernelConfig(BaseModel):
    """MI355X compute-plane knobs (net-new vs the reference)."""

    dtype: str = "bf16"
    use_hip_ops: bool = True  # hand-written CDNA4 kernels on the fit path
    bucket_bytes: int = 64 << 20  # RCCL bucket size for chunked overlap
    streams: int = 2


class NetworkConfig(BaseModel):
    """A whole dev-network: orchestrator + N stations (``v6 dev`` analog)."""

    name: str = "demo"
    server: ServerConfig = Field(default_factory=ServerConfig)
    stations: list[StationConfig] = Field(default_factory=list)
    kernels: KernelConfig = Field(default_factory=KernelConfig)
    encrypted: bool = False


def load_config(path: str | Path, model: type[BaseModel] = NetworkConfig) -> BaseModel:
    with open(path) as fh:
        raw = yaml.safe_load(fh) or {}
    return model.model_validate(raw)


def save_config(cfg: BaseModel, path: str | Path) -> None:
    Path(path).parent.mkdir(parents=True, exist_ok=True)
    with open(path, "w") as fh:
        yaml.safe_dump(cfg.model_dump(), fh, sort_keys=False)


class AppContext:
    """Resolve per-instance folders (config/data/log), reference parity with
    ``vantage6.common.context.AppContext`` minus the appdirs system/user split
    (single-node deployment: everything under one root)."""

    def __init__(self, name: str, root: str | Path | None = None):
        self.name = name
        self.root = Path(root) if root else Path.home() / ".vantage6_amd" / name
        self.root.mkdir(parents=True, exist_ok=True)

    @property
    def data_dir(self) -> Path:
        p = self.root / "data"
        p.mkdir(exist_ok=True)
        return p

    @property
    def log_dir(self) -> Path:
        p = self.root / "log"
        p.mkdir(exist_ok=True)
        return p

    @property
    def config_file(self) -> Path:
        return self.root / "config.yaml"
