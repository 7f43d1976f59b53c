"""Status engine / job phase machine (reference: pkg/controller/status.go:101-305).

Improvement over the reference (SURVEY.md §7 hard part 4): phase updates go
through the /status subresource with conflict retry + re-read, instead of
blind Update() x5 on the main resource.
"""
from __future__ import annotations

import logging
from typing import Dict, List

from ..api.types import (
    AITrainingJob, CleanPodPolicy, ENDING_PHASES, Phase, RestartScope,
)
from ..kube import objects as ko
from ..kube.client import ApiError, KubeApi
from ..policy import engine
from ..utils.k8stime import format_time, parse_time
from .pods import filter_pods_for_replica_type

log = logging.getLogger(__name__)


class StatusEngine:
    def __init__(self, api: KubeApi, recorder=None, enqueue=None):
        self.api = api
        self.recorder = recorder
        # enqueue(job, rate_limited: bool, delay_seconds: float)
        self.enqueue = enqueue or (lambda job, rl, delay: None)

    # ------------------------------------------------------------------
    def update_status(self, job: AITrainingJob, pods: List[dict],
                      services: List[dict],
                      ending_phases: Dict[str, str], message: str,
                      now: float) -> None:
        """reference: status.go:101-253."""
        for rtype in job.spec.replica_specs:
            rt = rtype.lower()
            job.status.replica_statuses[rtype] = \
                engine.count_replica_statuses(
                    job.status.restart_counts.get(rtype, 0),
                    filter_pods_for_replica_type(pods, rt))

        # restart wait-gate: hold until the scoped pods are gone, then flip
        # to Restarting and clear the marker (status.go:113-143)
        if job.status.restart_replica_name:
            rname = job.status.restart_replica_name
            spec = job.spec.replica_specs.get(rname)
            if spec is None:
                job.status.restart_replica_name = ""
                return
            scope = spec.restart_scope
            replica_pods = filter_pods_for_replica_type(pods, rname.lower())
            if scope == RestartScope.ALL and not pods:
                engine.update_job_conditions(
                    job, Phase.RESTARTING, "All pods are restarting now", now)
                job.status.restart_replica_name = ""
            elif scope == RestartScope.REPLICA and not replica_pods:
                engine.update_job_conditions(
                    job, Phase.RESTARTING,
                    f"{rname.lower()} pods are restarting now", now)
                job.status.restart_replica_name = ""
            elif scope == RestartScope.POD and \
                    len(replica_pods) < (spec.replicas or 0):
                engine.update_job_conditions(
                    job, Phase.RESTARTING, "pod is restarting now", now)
                job.status.restart_replica_name = ""
            return

        # job-level ending aggregation (status.go:144-174)
        decision = engine.aggregate_job_ending(job, ending_phases, message)
        if decision.terminate:
            self.terminate(job, pods, services, decision.phase,
                           decision.message, now)
            return

        # deferred finalization: termination annotation + pods gone ->
        # final phase (status.go:176-187)
        for phase in ENDING_PHASES:
            if phase in job.annotations:
                if not pods:
                    job.status.end_time = format_time(now)
                    msg = f"{job.annotations[phase]}; deleted pods"
                    engine.update_job_conditions(job, phase, msg, now)
                else:
                    self.enqueue(job, True, 0)
                return

        # TimeLimit (status.go:189-198)
        if job.spec.time_limit is not None and job.status.start_running_time:
            started = parse_time(job.status.start_running_time) or now
            if now - started >= job.spec.time_limit:
                msg = (f"started at {job.status.start_running_time}, "
                       f"timeLimit is {job.spec.time_limit} second")
                self.terminate(job, pods, services, Phase.TIMEOUT, msg, now)
                return

        # phase derivation from counters (status.go:200-244)
        is_scheduled = True
        is_creating = False
        is_running = True
        is_restarting = False
        for rtype, spec in job.spec.replica_specs.items():
            replicas = spec.replicas or 0
            rs = job.status.replica_statuses[rtype]
            accounted = (rs.scheduled + rs.active + rs.succeeded + rs.failed
                         + rs.restarting)
            is_scheduled = is_scheduled and accounted == replicas
            is_creating = is_creating or rs.scheduled > 0
            is_restarting = is_restarting or rs.restarting > 0
            is_running = is_running and rs.active == replicas

        if job.status.phase != Phase.RUNNING and is_running:
            if job.status.start_running_time is None:
                job.status.start_running_time = format_time(now)
            engine.update_job_conditions(job, Phase.RUNNING,
                                         "all pods are running", now)
        if is_creating and is_scheduled and \
                job.status.phase != Phase.RESTARTING:
            engine.update_job_conditions(job, Phase.CREATING, message, now)
        if is_restarting and job.status.phase != Phase.RESTARTING:
            engine.update_job_conditions(job, Phase.RESTARTING, message, now)
        if not is_scheduled and not is_restarting and \
                job.status.phase != Phase.RESTARTING:
            if job.status.start_time is None:
                job.status.start_time = format_time(now)
            engine.update_job_conditions(
                job, Phase.PENDING, "all pods are waiting for scheduling",
                now)

        # schedule the delayed sync that will fire the TimeLimit
        # (status.go:246-252)
        if job.spec.time_limit is not None and job.status.start_running_time:
            started = parse_time(job.status.start_running_time) or now
            remaining = job.spec.time_limit - (now - started)
            self.enqueue(job, False, max(remaining, 0.0))

    # ------------------------------------------------------------------
    def terminate(self, job: AITrainingJob, pods: List[dict],
                  services: List[dict], ending_phase: str, message: str,
                  now: float) -> None:
        """reference: status.go:256-283. CleanPodPolicy None keeps pods and
        finalizes immediately; otherwise annotate the pending final phase,
        delete pods+services, and go Terminating until they are gone."""
        if self.recorder:
            self.recorder.event(
                job, "Normal" if ending_phase == Phase.SUCCEEDED
                else "Warning", str(ending_phase), message)
        if job.spec.clean_pod_policy == CleanPodPolicy.NONE:
            job.status.end_time = format_time(now)
            engine.update_job_conditions(job, ending_phase, message, now)
            return
        job.annotations[str(ending_phase)] = message
        for pod in pods:
            try:
                self.api.delete_pod(ko.namespace_of(pod), ko.name_of(pod))
            except ApiError as e:
                if not e.not_found:
                    raise
        for svc in services:
            try:
                self.api.delete_service(ko.namespace_of(svc),
                                        ko.name_of(svc))
            except ApiError as e:
                if not e.not_found:
                    raise
        engine.update_job_conditions(job, Phase.TERMINATING, message, now)

    # ------------------------------------------------------------------
    def persist(self, job: AITrainingJob, retries: int = 5) -> None:
        """Write status (+ annotations, which carry the pending-termination
        marker) back to the API with optimistic-conflict retry
        (reference: status.go:285-305, improved to re-read on conflict)."""
        # real semantics for status.lastReconcileTime (declared but never
        # set in the reference): the time of the last
        # status write — stamped here, not per-sync, to avoid API churn
        from ..utils.k8stime import format_time
        job.status.last_reconcile_time = format_time()
        # the CRD declares a status subresource, so .status MUST go through
        # update_job_status (a main-resource PUT silently drops it on a real
        # API server); annotations (the deferred-termination marker) go
        # through a main-resource update only when they changed
        for attempt in range(retries):
            try:
                current = self.api.get_job(job.namespace, job.name)
            except ApiError as e:
                if e.not_found:
                    return
                raise
            try:
                cur_ann = current.get("metadata", {}).get("annotations") or {}
                want_ann = {**cur_ann, **job.annotations}
                if want_ann != cur_ann:
                    current.setdefault("metadata", {})["annotations"] = \
                        want_ann
                    current = self.api.update_job(job.namespace, job.name,
                                                  current)
                current["status"] = job.status.to_dict()
                self.api.update_job_status(job.namespace, job.name, current)
                return
            except ApiError as e:
                if e.conflict and attempt < retries - 1:
                    continue
                raise
