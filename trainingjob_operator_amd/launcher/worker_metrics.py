"""Worker-side Prometheus metrics: the north-star tokens/sec plus step/loss
gauges, exposed by rank 0 (SURVEY.md §5 'Metrics': "worker-side tokens/sec +
step-time metrics (the north-star metric)")."""
from __future__ import annotations

import logging

log = logging.getLogger(__name__)


class WorkerMetrics:
    def __init__(self, port: int):
        from prometheus_client import Gauge, start_http_server
        self.tokens_per_sec = Gauge(
            "aitj_worker_tokens_per_sec",
            "Whole-job training throughput (tokens/s, all ranks)")
        self.step = Gauge("aitj_worker_step", "Current optimizer step")
        self.loss = Gauge("aitj_worker_loss", "Last training loss")
        self.eval_loss = Gauge("aitj_worker_eval_loss",
                               "Last held-out evaluation loss")
        start_http_server(port)
        log.info("worker metrics on :%d/metrics", port)

    def observe(self, step: int, loss: float, tokens_per_sec: float) -> None:
        self.step.set(step)
        self.loss.set(loss)
        self.tokens_per_sec.set(tokens_per_sec)

    def observe_eval(self, loss: float) -> None:
        self.eval_loss.set(loss)
