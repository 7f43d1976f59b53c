"""Prometheus metrics for the operator (the reference shipped none despite
vendoring client_golang — SURVEY.md §5 'Metrics')."""
from __future__ import annotations

import logging

log = logging.getLogger(__name__)


class OperatorMetrics:
    def __init__(self, port: int = 0):
        from prometheus_client import Counter, Gauge, Histogram, start_http_server
        self.sync_duration = Histogram(
            "aitj_sync_duration_seconds",
            "Duration of one reconcile sync",
            buckets=(0.001, 0.005, 0.02, 0.1, 0.5, 2.0))
        self.syncs_total = Counter(
            "aitj_syncs_total", "Reconcile syncs by resulting phase",
            ["phase"])
        self.restarts_total = Counter(
            "aitj_restarts_total", "Pod restarts triggered", ["scope"])
        self.jobs_by_phase = Gauge(
            "aitj_jobs", "Jobs currently in phase", ["phase"])
        self.elastic_resizes_total = Counter(
            "aitj_elastic_resizes_total",
            "EdlPolicy=Auto world resizes by direction", ["direction"])
        if port:
            start_http_server(port)
            log.info("metrics on :%d/metrics", port)

    def observe_sync(self, seconds: float, phase: str) -> None:
        self.sync_duration.observe(seconds)
        self.syncs_total.labels(phase=phase or "None").inc()
