"""Node health agent (DaemonSet): publishes per-GPU health as a node
condition the controller's NodeFail path consumes.

Replaces the reference's only health signal (NodeReady polled via full node
LIST, reference pod.go:439-455) with GPU-granular health: the agent sets an
``EDLGPUHealthy`` condition + a detail annotation on its node; the
controller treats EDLGPUHealthy=False like NotReady (pods on that node hit
the NodeFail phase/restart policies within one probe period).
"""
from __future__ import annotations

import argparse
import logging
import os
import threading

from ..api import constants as C
from ..kube.client import KubeApi, RealKubeApi
from ..utils.k8stime import format_time
from . import gpu_health

log = logging.getLogger("node-agent")

GPU_HEALTH_CONDITION = "EDLGPUHealthy"
GPU_HEALTH_ANNOTATION = f"{C.CRD_GROUP}/gpu-health"


class NodeAgent:
    def __init__(self, api: KubeApi, node_name: str,
                 expected_gpus: int = 0, period: float = 10.0):
        self.api = api
        self.node_name = node_name
        self.expected = expected_gpus or None
        self.period = period
        self._last_healthy = None

    def probe_and_publish(self, now: float = None) -> gpu_health.NodeGpuReport:
        report = gpu_health.probe(self.expected)
        healthy = report.healthy
        ts = format_time(now)
        self.api.patch_node_status(self.node_name, {
            "conditions": [{
                "type": GPU_HEALTH_CONDITION,
                "status": "True" if healthy else "False",
                "reason": "GPUHealthy" if healthy else "GPUUnhealthy",
                "message": report.summary(),
                "lastHeartbeatTime": ts,
                "lastTransitionTime": ts,
            }],
        })
        self.api.annotate_node(self.node_name,
                               {GPU_HEALTH_ANNOTATION: report.to_json()})
        if healthy != self._last_healthy:
            log.info("node %s GPU health -> %s (%s)", self.node_name,
                     healthy, report.summary())
            self._last_healthy = healthy
        return report

    def run(self, stop: threading.Event) -> None:
        while not stop.is_set():
            try:
                self.probe_and_publish()
            except Exception:
                log.exception("probe failed")
            stop.wait(self.period)


def main(argv=None) -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--node-name",
                    default=os.environ.get("NODE_NAME", ""))
    ap.add_argument("--expected-gpus", type=int,
                    default=int(os.environ.get("EXPECTED_GPUS", "0")))
    ap.add_argument("--period", type=float, default=10.0)
    args = ap.parse_args(argv)
    logging.basicConfig(level=logging.INFO)
    if not args.node_name:
        ap.error("--node-name or NODE_NAME required")
    agent = NodeAgent(RealKubeApi(), args.node_name, args.expected_gpus,
                      args.period)
    agent.run(threading.Event())
    return 0


if __name__ == "__main__":
    import sys
    sys.exit(main())
