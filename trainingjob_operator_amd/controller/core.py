"""TrainingJobController: the reconcile loop
(reference: pkg/controller/controller.go:37-440).

Tests drive ``sync_once(key)`` synchronously against a FakeKubeApi; ``run``
wires watch threads -> workqueue -> N workers for production.
"""
from __future__ import annotations

import logging
import threading
import time
from typing import Dict, List, Optional

from ..api import constants as C
from ..api.crd import crd_manifest
from ..api.defaults import set_defaults
from ..api.types import ACTIVE_PHASES, AITrainingJob, Phase
from ..api.validation import validate
from ..kube import objects as ko
from ..kube.client import ApiError, KubeApi
from ..kube.workqueue import RateLimitedQueue
from .events import EventRecorder
from .expectations import ControllerExpectations
from .gc import GarbageCollector
from .options import OperatorOptions
from .pods import PodReconciler, node_ready_map
from .services import ServiceReconciler
from .status import StatusEngine

log = logging.getLogger(__name__)


class TrainingJobController:
    def __init__(self, api: KubeApi,
                 options: Optional[OperatorOptions] = None,
                 metrics=None):
        self.api = api
        self.options = options or OperatorOptions()
        self.queue = RateLimitedQueue()
        self.recorder = EventRecorder(api)
        self.expectations = ControllerExpectations()
        self.pod_reconciler = PodReconciler(api, self.options, self.recorder,
                                            self.expectations)
        self.service_reconciler = ServiceReconciler(api, self.recorder,
                                                    self.expectations)
        self.status_engine = StatusEngine(api, self.recorder, self._enqueue)
        self.gc = GarbageCollector(api, self.options.namespace)
        self.metrics = metrics
        from ..utils.tracing import tracer
        self.trace = tracer("controller")
        self._stop = threading.Event()

    # ------------------------------------------------------------------
    # enqueue paths (reference: controller.go:406-422)
    # ------------------------------------------------------------------
    def _enqueue(self, job: AITrainingJob, rate_limited: bool,
                 delay: float) -> None:
        key = job.key
        if delay > 0:
            self.queue.add_after(key, delay)
        elif rate_limited:
            self.queue.add_rate_limited(key)
        else:
            self.queue.add(key)

    def enqueue_key(self, key: str) -> None:
        self.queue.add(key)

    # ------------------------------------------------------------------
    # sync (reference: controller.go:270-312)
    # ------------------------------------------------------------------
    def sync_once(self, key: str, now: Optional[float] = None) -> None:
        if now is None:
            now = time.time()
        t0 = time.perf_counter()
        namespace, name = key.split("/", 1)
        try:
            job_dict = self.api.get_job(namespace, name)
        except ApiError as e:
            if e.not_found:
                self.queue.forget(key)
                return
            raise
        job = AITrainingJob.from_dict(job_dict)
        set_defaults(job)
        errors = validate(job)
        if errors:
            self.recorder.event(job, "Warning", "ValidationFailed",
                                "; ".join(errors))
            return
        # gate on active phases (controller.go:298-304)
        if job.status.phase not in ACTIVE_PHASES:
            return
        phase_before = job.status.phase
        self.reconcile(job, now)
        if self.metrics:
            self.metrics.observe_sync(time.perf_counter() - t0,
                                      job.status.phase)
        self.trace.event("sync", job=key,
                         duration_ms=round((time.perf_counter() - t0) * 1e3, 3),
                         phase=job.status.phase)
        if job.status.phase != phase_before:
            self.trace.event("phase_transition", job=key,
                             from_phase=phase_before, to=job.status.phase)

    # ------------------------------------------------------------------
    # reconcile (reference: controller.go:314-388)
    # ------------------------------------------------------------------
    def reconcile(self, job: AITrainingJob, now: float) -> None:
        original_status = job.status.to_dict()
        original_annotations = dict(job.annotations)
        selector = ko.job_selector(job.name)
        pods = self._claim(self.api.list_pods(job.namespace, selector), job)
        services = self._claim(
            self.api.list_services(job.namespace, selector), job)

        # in-flight-operation gate (reference: controller.go:295,390-404):
        # the fresh list first settles any expectation it can prove, then
        # pending creates/deletes suppress this sync (a delayed requeue
        # keeps progress even if the watch event is lost)
        # names are kind-prefixed: per-index headless services share the
        # pod names, so unprefixed sets would mask pod deletions
        present = {f"pod/{ko.name_of(o)}" for o in pods} | \
            {f"svc/{ko.name_of(o)}" for o in services}
        self.expectations.observe_list(job.key, present)
        if not self.expectations.satisfied(job.key, now):
            self.queue.add_after(job.key, 1.0)
            return

        ending_phases: Dict[str, str] = {}
        message = ""
        if not job.status.restart_replica_name:
            node_ready = node_ready_map(self.api)
            for rtype in sorted(job.spec.replica_specs):
                phase, msg = self.pod_reconciler.reconcile(
                    job, pods, rtype, node_ready, now)
                ending_phases[rtype] = phase
                if msg:
                    message = msg
                if phase == Phase.RESTARTING:
                    # two-sync restart dance (controller.go:362-366):
                    # mark, go Terminating, wait for pods to vanish
                    from ..policy import engine
                    engine.update_job_conditions(job, Phase.TERMINATING,
                                                 msg, now)
                    job.status.restart_replica_name = rtype
                    if self.metrics:
                        self.metrics.restarts_total.labels(
                            scope=job.spec.replica_specs[rtype]
                            .restart_scope or "All").inc()
                    break
                self.service_reconciler.reconcile(job, services, rtype)

        # updateStatus always runs; its restart wait-gate handles the syncs
        # between Terminating and Restarting (controller.go:380, status.go:113)
        self.status_engine.update_status(job, pods, services, ending_phases,
                                         message, now)

        if self.metrics:
            from .pods import TARGET_ANNOTATION
            for k, v in job.annotations.items():
                prev = original_annotations.get(k)
                if k.startswith(TARGET_ANNOTATION) and prev not in (None, v):
                    try:
                        direction = "up" if int(v) > int(prev) else "down"
                    except ValueError:
                        continue
                    self.metrics.elastic_resizes_total.labels(
                        direction=direction).inc()

        if job.status.to_dict() != original_status or \
                dict(job.annotations) != original_annotations:
            self.status_engine.persist(job)

    def _claim(self, objs: List[dict], job: AITrainingJob) -> List[dict]:
        """Ownership filter with adoption (reference PodControllerRefManager,
        pod.go:134-150): selector-matched objects with our controllerRef are
        managed; label-matching STRAYS with no controller at all are adopted
        by patching our ownerReference onto them (pods only — services are
        cheap to recreate); objects owned by another controller are left
        alone. Release-on-label-mismatch cannot trigger here because the
        list is already selector-filtered (documented divergence)."""
        out = []
        for o in objs:
            ref = ko.controller_ref(o)
            if ref is None:
                if ko.is_deleting(o) or not job.uid:
                    continue
                if o.get("kind") != "Pod" and "spec" not in o:
                    continue
                is_pod = "containers" in (o.get("spec") or {})
                if not is_pod:
                    continue
                try:
                    patched = self.api.patch_pod_metadata(
                        ko.namespace_of(o), ko.name_of(o),
                        {"ownerReferences": [ko.gen_owner_reference(job)]})
                    self.recorder.event(job, "Normal", "AdoptedPod",
                                        f"adopted stray pod {ko.name_of(o)}")
                    out.append(patched)
                except ApiError:
                    continue
                continue
            if ref.get("kind") == C.CRD_KIND and ref.get("name") == job.name:
                if not job.uid or ref.get("uid") in (None, job.uid):
                    out.append(o)
        return out

    # ------------------------------------------------------------------
    # run loop (reference: controller.go:182-268)
    # ------------------------------------------------------------------
    def ensure_crd(self) -> None:
        """Self-register the CRD at startup (controller.go:210-234; ours
        carries a real OpenAPI v3 schema, the reference's had none)."""
        self.api.ensure_crd(crd_manifest())

    def run(self, stop: Optional[threading.Event] = None) -> None:
        stop = stop or self._stop
        self.ensure_crd()
        threads: List[threading.Thread] = []

        def watcher(watch_fn, to_key, observe=False):
            for evt_type, obj in watch_fn(self.options.namespace or None,
                                          stop):
                if evt_type == "RELIST":
                    # watch rv expired (410 Gone): re-list so nothing
                    # that happened during the gap waits for the resync
                    try:
                        for j in self.api.list_jobs(
                                self.options.namespace or None):
                            self.queue.add(job_key(j))
                    except Exception:
                        log.exception("relist after 410 failed")
                    continue
                key = to_key(obj)
                if key:
                    if observe:
                        kind = "svc" if obj.get("kind") == "Service" else \
                            "pod"
                        if evt_type == "ADDED":
                            self.expectations.creation_observed(
                                key, f"{kind}/{ko.name_of(obj)}")
                        elif evt_type == "DELETED":
                            self.expectations.deletion_observed(
                                key, f"{kind}/{ko.name_of(obj)}")
                    self.queue.add(key)

        def job_key(obj):
            m = obj.get("metadata", {})
            return f"{m.get('namespace', 'default')}/{m.get('name', '')}"

        def owned_key(obj):
            ref = ko.controller_ref(obj)
            if ref and ref.get("kind") == C.CRD_KIND:
                return f"{ko.namespace_of(obj)}/{ref.get('name')}"
            return None

        for fn, keyer, obs in ((self.api.watch_jobs, job_key, False),
                               (self.api.watch_pods, owned_key, True),
                               (self.api.watch_services, owned_key, True)):
            t = threading.Thread(target=watcher, args=(fn, keyer, obs),
                                 daemon=True)
            t.start()
            threads.append(t)

        def resync():
            while not stop.wait(self.options.resync_period):
                try:
                    for j in self.api.list_jobs(self.options.namespace
                                                or None):
                        self.queue.add(job_key(j))
                except Exception:
                    log.exception("resync list failed")

        threading.Thread(target=resync, daemon=True).start()

        def gc_loop():
            while not stop.wait(self.options.gc_period):
                try:
                    self.gc.clean_garbage_pods(time.time())
                except Exception:
                    log.exception("gc sweep failed")

        threading.Thread(target=gc_loop, daemon=True).start()

        def worker():
            while not stop.is_set():
                key = self.queue.get(timeout=0.5)
                if key is None:
                    continue
                try:
                    self.sync_once(key)
                    self.queue.forget(key)
                except Exception:
                    log.exception("sync %s failed", key)
                    self.queue.add_rate_limited(key)
                finally:
                    self.queue.done(key)

        workers = [threading.Thread(target=worker, daemon=True)
                   for _ in range(self.options.thread_num)]
        for w in workers:
            w.start()
        log.info("controller running with %d workers", len(workers))
        stop.wait()
        self.queue.shut_down()

    def stop(self) -> None:
        self._stop.set()
