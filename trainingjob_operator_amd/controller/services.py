"""Headless-service reconciler: one service per pod index for stable DNS
(reference: pkg/controller/service.go:117-240)."""
from __future__ import annotations

import logging
from typing import List

from ..api import constants as C
from ..api.types import AITrainingJob, gen_general_name
from ..kube import objects as ko
from ..kube.client import ApiError, KubeApi
from .envinject import ports_from_spec

log = logging.getLogger(__name__)


def filter_services_for_replica_type(services: List[dict],
                                     rt: str) -> List[dict]:
    """reference: service.go:198-218."""
    return [s for s in services
            if ko.labels_of(s).get(C.LABEL_REPLICA_NAME) == rt]


def service_slices(services: List[dict], replicas: int) -> List[List[dict]]:
    """reference: service.go:220-240."""
    slices: List[List[dict]] = [[] for _ in range(replicas)]
    for s in services:
        try:
            idx = int(ko.labels_of(s).get(C.LABEL_REPLICA_INDEX, "-1"))
        except ValueError:
            idx = -1
        if 0 <= idx < replicas:
            slices[idx].append(s)
    return slices


class ServiceReconciler:
    def __init__(self, api: KubeApi, recorder=None, expectations=None):
        self.api = api
        self.recorder = recorder
        self.expectations = expectations

    def reconcile(self, job: AITrainingJob, all_services: List[dict],
                  rtype: str) -> None:
        """Gap-fill one headless service per index when the replica type has
        aitj-* ports (reference: service.go:117-146)."""
        rt = rtype.lower()
        spec = job.spec.replica_specs[rtype]
        ports = ports_from_spec(spec)
        if not ports:
            return
        replicas = spec.replicas or 0
        services = filter_services_for_replica_type(all_services, rt)
        for index, sslice in enumerate(service_slices(services, replicas)):
            if not sslice:
                self._create(job, rt, index, ports)

    def _create(self, job: AITrainingJob, rt: str, index: int,
                ports: List[int]) -> None:
        """reference: service.go:148-196 (ClusterIP None; selector pins the
        exact replica index -> per-pod DNS {job}-{rt}-{i}.{ns})."""
        labels = ko.gen_labels(job.name)
        labels[C.LABEL_REPLICA_NAME] = rt
        labels[C.LABEL_REPLICA_INDEX] = str(index)
        svc = {
            "apiVersion": "v1",
            "kind": "Service",
            "metadata": {
                "name": gen_general_name(job.name, rt, index),
                "labels": labels,
                "ownerReferences": [ko.gen_owner_reference(job)],
            },
            "spec": {
                "clusterIP": "None",
                "selector": labels,
                "ports": [{"name": f"{C.PORT_PREFIX}{p}", "port": p}
                          for p in ports],
            },
        }
        try:
            if self.expectations:
                self.expectations.expect_creation(
                    job.key, f"svc/{svc['metadata']['name']}")
            try:
                self.api.create_service(job.namespace, svc)
                if self.expectations:
                    self.expectations.creation_observed(
                        job.key, f"svc/{svc['metadata']['name']}")
            except ApiError:
                if self.expectations:
                    self.expectations.creation_observed(
                        job.key, f"svc/{svc['metadata']['name']}")
                raise
        except ApiError as e:
            if not e.already_exists:
                raise
        if self.recorder:
            self.recorder.event(job, "Normal", "SuccessfulCreateService",
                                f"created service {svc['metadata']['name']}")
