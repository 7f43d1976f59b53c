"""Kubernetes Event recorder (reference: controller.go:88-102 event
broadcaster/recorder)."""
from __future__ import annotations

import itertools
import logging

from ..api import constants as C
from ..kube.client import KubeApi
from ..utils.k8stime import format_time

log = logging.getLogger(__name__)


class EventRecorder:
    def __init__(self, api: KubeApi, component: str = C.CONTROLLER_NAME):
        self.api = api
        self.component = component
        self._seq = itertools.count(1)

    def event(self, job, etype: str, reason: str, message: str) -> None:
        log.info("event %s %s %s: %s", job.key, etype, reason, message)
        now = format_time()
        self.api.create_event(job.namespace, {
            "apiVersion": "v1",
            "kind": "Event",
            "metadata": {
                "name": f"{job.name}.{next(self._seq)}",
                "namespace": job.namespace,
            },
            "involvedObject": {
                "apiVersion": C.API_VERSION,
                "kind": C.CRD_KIND,
                "name": job.name,
                "namespace": job.namespace,
                "uid": job.uid,
            },
            "reason": reason,
            "message": message,
            "type": etype,
            "source": {"component": self.component},
            "firstTimestamp": now,
            "lastTimestamp": now,
            "count": 1,
        })
