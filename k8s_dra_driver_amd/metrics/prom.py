"""Prometheus metrics for the plugin's prepare hot path.

The reference exposes metrics only on the controller; the plugin — whose
latency IS the north-star metric — has none (SURVEY.md §5.5 calls this a
gap to fix). Metrics are registry-scoped so tests and multi-instance
benches never collide on the global default registry.
"""

from __future__ import annotations

import contextlib
import time
from typing import Optional

from prometheus_client import (
    CollectorRegistry,
    Counter,
    Gauge,
    Histogram,
    start_http_server,
)

_BUCKETS = (
    0.0005,
    0.001,
    0.0025,
    0.005,
    0.01,
    0.025,
    0.05,
    0.1,
    0.25,
    0.5,
    1.0,
    2.5,
    5.0,
)


class PluginMetrics:
    def __init__(self, registry: Optional[CollectorRegistry] = None):
        self.registry = registry or CollectorRegistry()
        self.prepare_seconds = Histogram(
            "dra_prepare_seconds",
            "NodePrepareResources per-claim latency",
            buckets=_BUCKETS,
            registry=self.registry,
        )
        self.unprepare_seconds = Histogram(
            "dra_unprepare_seconds",
            "NodeUnprepareResources per-claim latency",
            buckets=_BUCKETS,
            registry=self.registry,
        )
        self.prepared_claims = Counter(
            "dra_prepared_claims_total",
            "Successfully prepared claims",
            registry=self.registry,
        )
        self.prepare_errors = Counter(
            "dra_prepare_errors_total",
            "Failed claim preparations",
            registry=self.registry,
        )
        self.unprepare_errors = Counter(
            "dra_unprepare_errors_total",
            "Failed claim unpreparations",
            registry=self.registry,
        )
        self.allocatable_devices = Gauge(
            "dra_allocatable_devices",
            "Devices currently published in ResourceSlices",
            registry=self.registry,
        )
        self.repartitions = Counter(
            "dra_repartitions_total",
            "Dynamic partition mode switches performed",
            registry=self.registry,
        )
        self.isolation_violations = Counter(
            "dra_shared_isolation_violations_total",
            "Shared-GPU processes caught with stripped/altered CU masks",
            registry=self.registry,
        )
        self.slice_heals = Counter(
            "dra_resourceslice_heals_total",
            "ResourceSlice republications triggered by external drift",
            registry=self.registry,
        )
        self.deferred_restores = Gauge(
            "dra_deferred_mode_restores",
            "GPUs whose partition-mode restore is pending a drain",
            registry=self.registry,
        )

    @contextlib.contextmanager
    def time_prepare(self):
        t0 = time.perf_counter()
        try:
            yield
        finally:
            self.prepare_seconds.observe(time.perf_counter() - t0)

    @contextlib.contextmanager
    def time_unprepare(self):
        t0 = time.perf_counter()
        try:
            yield
        finally:
            self.unprepare_seconds.observe(time.perf_counter() - t0)

    def serve(self, port: int) -> None:
        """Expose /metrics (controller parity: reference main.go:194-214)."""
        start_http_server(port, registry=self.registry)
