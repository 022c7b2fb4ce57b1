"""Demo-network harness.

Reference parity: ``vantage6/vantage6/cli/dev.py`` —
``v6 dev create-demo-network`` spins up a server + N org/node configs on one
host; it is the de-facto integration harness. Here :class:`DevNetwork`
boots the in-process orchestrator, N stations (CPU or one per GPU) and an
authenticated researcher client in one call — used by the CLI, the tests
and the diagnostics workload.
"""
from __future__ import annotations

import tempfile
from pathlib import Path

from ..common.config import DatabaseConfig, StationConfig
from ..client import UserClient
from ..node.station import Station
from ..server.core import ServerCore


class DevNetwork:
    """Orchestrator + N stations + researcher client, all in-process."""

    def __init__(
        self,
        n_stations: int = 2,
        encrypted: bool = False,
        databases: list[list[DatabaseConfig]] | None = None,
        devices: list[str] | None = None,
        uri: str = "sqlite://",
        name: str = "demo",
        key_bits: int = 2048,
    ):
        self.core = ServerCore(uri=uri, password_policy=False)
        self.encrypted = encrypted
        self.core.create_root_user("root", "root")
        root = UserClient(core=self.core)
        root.authenticate("root", "root")
        self.root_client = root

        org_ids = []
        for i in range(n_stations):
            org = root.organization.create(name=f"{name}-org-{i}")
            org_ids.append(org["id"])
        self.collaboration = root.collaboration.create(
            name=name, organization_ids=org_ids, encrypted=encrypted,
        )
        cid = self.collaboration["id"]

        self._tmp = Path(tempfile.mkdtemp(prefix=f"v6amd_net_{name}_"))
        self.stations: list[Station] = []
        for i, org_id in enumerate(org_ids):
            node = root.node.create(
                name=f"{name}-station-{i}", collaboration_id=cid,
                organization_id=org_id,
                device=(devices[i] if devices else "cpu"),
            )
            cfg = StationConfig(
                name=f"{name}-station-{i}",
                organization_id=org_id,
                device=(devices[i] if devices else "cpu"),
                databases=databases[i] if databases else [],
            )
            station = Station(
                self.core, cfg, api_key=node["api_key"],
                private_key_file=(
                    str(self._tmp / f"station{i}.key") if encrypted else None
                ),
                key_bits=key_bits,
            )
            self.stations.append(station)

        # researcher user belongs to station org 0 (vantage6 model: one
        # keypair per organization, shared by the org's node and users)
        root.user.create(
            username=f"{name}-researcher", password="password",
            organization_id=org_ids[0], roles=["Researcher"],
        )
        self.researcher_org_id = org_ids[0]
        self.org_ids = org_ids
        self.client = UserClient(core=self.core)
        self.client.authenticate(f"{name}-researcher", "password")
        if encrypted:
            # same key file as station 0 — org-level keypair
            self.client.setup_encryption(str(self._tmp / "station0.key"),
                                         key_bits=key_bits)

    def start(self) -> "DevNetwork":
        for s in self.stations:
            s.start()
        return self

    def stop(self) -> None:
        for s in self.stations:
            s.stop()
        self.core.close()

    def __enter__(self) -> "DevNetwork":
        return self.start()

    def __exit__(self, *exc) -> None:
        self.stop()
