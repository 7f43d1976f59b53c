"""Operator bootstrap: flags -> client -> leader election -> controller
(reference: cmd/main.go + cmd/app/server.go:26-109).

    python -m trainingjob_operator_amd.controller.server --namespace ml
"""
from __future__ import annotations

import argparse
import logging
import signal
import sys
import threading

from ..kube.client import RealKubeApi
from .core import TrainingJobController
from .leaderelect import LeaderElector
from .options import OperatorOptions

log = logging.getLogger("operator")


def setup_signal_handler() -> threading.Event:
    """SIGINT/SIGTERM -> stop event; second signal exits hard
    (reference: pkg/signals/signal.go:29)."""
    stop = threading.Event()

    def handler(signum, frame):
        if stop.is_set():
            sys.exit(1)
        log.info("shutdown requested")
        stop.set()

    signal.signal(signal.SIGINT, handler)
    signal.signal(signal.SIGTERM, handler)
    return stop


def run(options: OperatorOptions, api=None) -> None:
    stop = setup_signal_handler()
    if api is None:
        api = RealKubeApi(base_url=options.master or None)
    metrics = None
    if options.metrics_port:
        from .metrics import OperatorMetrics
        metrics = OperatorMetrics(options.metrics_port)
    controller = TrainingJobController(api, options, metrics=metrics)

    if options.leader_elect:
        elector = LeaderElector(
            api, options.lease_namespace, options.lease_name,
            lease_duration=options.lease_duration,
            renew_deadline=options.renew_deadline,
            retry_period=options.retry_period)
        elector.run(lambda: controller.run(stop), stop)
    else:
        controller.run(stop)


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(
        description="MI355X-native AITrainingJob operator")
    OperatorOptions.add_flags(ap)
    args = ap.parse_args(argv)
    logging.basicConfig(
        level=logging.INFO,
        format="%(asctime)s %(name)s [%(levelname)s] %(message)s")
    run(OperatorOptions.from_args(args))
    return 0


if __name__ == "__main__":
    sys.exit(main())
