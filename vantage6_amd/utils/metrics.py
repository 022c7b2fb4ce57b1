"""Observability: per-round timing table + Prometheus endpoint.

Reference parity: the reference exposes run logs and node
online status but no metrics endpoint; this adds the per-round timing
table (round #, fit ms, aggregate ms, loss) and an optional
``prometheus_client`` HTTP endpoint with rounds/sec and phase timings —
the net-new observability the MI355X deployment needs.
"""
from __future__ import annotations

from typing import Optional


class RoundMetrics:
    """Collects FedAvgTrainer.run_round() stats; optionally exports them."""

    def __init__(self, prometheus_port: Optional[int] = None):
        self.rows: list[dict] = []
        self._prom = None
        if prometheus_port is not None:
            import prometheus_client as pc

            self._prom = {
                "rounds": pc.Counter("v6amd_rounds_total", "completed FedAvg rounds"),
                "fit_ms": pc.Gauge("v6amd_fit_ms", "local fit ms (last round)"),
                "agg_ms": pc.Gauge("v6amd_agg_ms", "aggregation ms (last round)"),
                "round_ms": pc.Gauge("v6amd_round_ms", "total round ms"),
                "loss": pc.Gauge("v6amd_loss", "mean local loss (last round)"),
                "overlap_pct": pc.Gauge(
                    "v6amd_overlap_pct",
                    "fraction of the aggregation collective hidden under "
                    "the next round's forward (fit/agg overlap)"),
                "agg_comm_ms": pc.Gauge(
                    "v6amd_agg_comm_ms", "collective wall ms (last round)"),
                "hbm_used": pc.Gauge(
                    "v6amd_hbm_used_bytes", "torch HBM allocated on this rank"),
            }
            pc.start_http_server(prometheus_port)

    def record(self, stats: dict) -> None:
        self.rows.append(dict(stats))
        if self._prom:
            self._prom["rounds"].inc()
            self._prom["fit_ms"].set(stats.get("fit_ms", 0))
            self._prom["agg_ms"].set(stats.get("agg_ms", 0))
            self._prom["round_ms"].set(stats.get("round_ms", 0))
            self._prom["loss"].set(stats.get("loss", 0))
            self._prom["overlap_pct"].set(stats.get("overlap_pct", 0))
            self._prom["agg_comm_ms"].set(stats.get("agg_comm_ms", 0))
            import torch

            if torch.cuda.is_available():
                self._prom["hbm_used"].set(torch.cuda.memory_allocated())

    def table(self, last: int = 20) -> str:
        overlap = any("overlap_pct" in r for r in self.rows)
        hdr = f"{'round':>6} {'loss':>10} {'fit ms':>9} {'agg ms':>9} {'total ms':>9}"
        if overlap:
            hdr += f" {'comm ms':>9} {'overlap%':>9}"
        lines = [hdr, "-" * len(hdr)]
        for r in self.rows[-last:]:
            line = (
                f"{r.get('round', 0):>6} {r.get('loss', 0):>10.4f} "
                f"{r.get('fit_ms', 0):>9.2f} {r.get('agg_ms', 0):>9.2f} "
                f"{r.get('round_ms', 0):>9.2f}"
            )
            if overlap:
                line += (f" {r.get('agg_comm_ms', 0):>9.2f} "
                         f"{r.get('overlap_pct', 0):>9.1f}")
            lines.append(line)
        if self.rows:
            import statistics

            med = statistics.median(r.get("round_ms", 0) for r in self.rows)
            lines.append(f"median round: {med:.2f} ms "
                         f"({1000.0 / med if med else 0:.2f} rounds/s)")
        return "\n".join(lines)
