"""Prometheus metrics for long-running transfer workloads (soak,
deployment monitoring).  Optional: degrades to no-ops when
prometheus_client is absent.

Usage:
    m = TransferMetrics(port=9109)      # starts the HTTP exporter
    m.observe_bytes("write", nbytes)
    m.observe_audit(ok=True)
or scrape-free:
    m = TransferMetrics(port=None); ...; text = m.render()
"""
from __future__ import annotations

try:
    from prometheus_client import (CollectorRegistry, Counter, Gauge,
                                   generate_latest, start_http_server)

    _HAVE_PROM = True
except ImportError:  # pragma: no cover
    _HAVE_PROM = False


class TransferMetrics:
    def __init__(self, port: int | None = None, prefix: str = "rocp2p"):
        self.enabled = _HAVE_PROM
        if not self.enabled:
            return
        self.registry = CollectorRegistry()
        self.bytes_total = Counter(
            f"{prefix}_bytes_total", "bytes transferred", ["direction"],
            registry=self.registry)
        self.msgs_total = Counter(
            f"{prefix}_messages_total", "messages transferred",
            ["direction"], registry=self.registry)
        self.audits_total = Counter(
            f"{prefix}_integrity_audits_total", "integrity audits run",
            registry=self.registry)
        self.audit_failures_total = Counter(
            f"{prefix}_integrity_failures_total",
            "integrity audits that found corruption",
            registry=self.registry)
        self.bandwidth = Gauge(
            f"{prefix}_bandwidth_gbps", "last measured bandwidth",
            ["direction"], registry=self.registry)
        if port:
            start_http_server(port, registry=self.registry)

    def observe_bytes(self, direction: str, nbytes: int, msgs: int = 0):
        if not self.enabled:
            return
        self.bytes_total.labels(direction).inc(nbytes)
        if msgs:
            self.msgs_total.labels(direction).inc(msgs)

    def observe_bandwidth(self, direction: str, gbps: float):
        if self.enabled:
            self.bandwidth.labels(direction).set(gbps)

    def observe_audit(self, ok: bool):
        if not self.enabled:
            return
        self.audits_total.inc()
        if not ok:
            self.audit_failures_total.inc()

    def render(self) -> bytes:
        if not self.enabled:
            return b""
        return generate_latest(self.registry)
