"""Pod reconciler: gap-fill creation, restart scopes, ending policies,
elastic resize (reference: pkg/controller/pod.go:152-326,483-546; elastic
semantics are new — the reference declared min/max/edlPolicy but never read
them).
"""
from __future__ import annotations

import copy
import logging
from typing import Dict, List, Tuple

from ..api import constants as C
from ..api.types import (
    AITrainingJob, EdlPolicy, EndingPolicy, Phase, ReplicaSpec, RestartScope,
    gen_general_name,
)
from ..kube import objects as ko
from ..kube.client import ApiError, KubeApi
from ..policy import engine
from .envinject import inject_env

log = logging.getLogger(__name__)

EPOCH_ANNOTATION = f"{C.CRD_GROUP}/rendezvous-epoch"
WORLD_SIZE_ANNOTATION = f"{C.CRD_GROUP}/world-size"
# EdlPolicy=Auto state, per replica type (suffix "-<rt>"): the controller's
# chosen effective world size and the last auto-resize timestamp
TARGET_ANNOTATION = f"{C.CRD_GROUP}/target-replicas"
LAST_RESIZE_ANNOTATION = f"{C.CRD_GROUP}/last-resize"


def filter_pods_for_replica_type(pods: List[dict], rt: str) -> List[dict]:
    """reference: pod.go:654-674."""
    return [p for p in pods
            if ko.labels_of(p).get(C.LABEL_REPLICA_NAME) == rt]


def pod_slices(pods: List[dict], replicas: int) -> List[List[dict]]:
    """Group by the replica-index label; out-of-range pods are returned
    separately by callers that care (reference: pod.go:676-696 drops them).
    """
    slices: List[List[dict]] = [[] for _ in range(replicas)]
    for p in pods:
        try:
            idx = int(ko.labels_of(p).get(C.LABEL_REPLICA_INDEX, "-1"))
        except ValueError:
            idx = -1
        if 0 <= idx < replicas:
            slices[idx].append(p)
    return slices


def out_of_range_pods(pods: List[dict], replicas: int) -> List[dict]:
    out = []
    for p in pods:
        try:
            idx = int(ko.labels_of(p).get(C.LABEL_REPLICA_INDEX, "-1"))
        except ValueError:
            idx = -1
        if idx < 0 or idx >= replicas:
            out.append(p)
    return out


def node_ready_map(api: KubeApi) -> Dict[str, bool]:
    """Map of healthy nodes (absent == failed). Extends the reference's
    NodeReady check (pod.go:439-455) with the node agent's GPU-granular
    ``EDLGPUHealthy`` condition: a node whose agent reports GPU-lost / ECC /
    thermal / xGMI-down is treated as failed even while kubelet-Ready, so
    pods there hit the NodeFail phase + restart policies."""
    ready = {}
    for node in api.list_nodes():
        is_ready = False
        gpu_healthy = True
        for cond in (node.get("status") or {}).get("conditions") or []:
            if cond.get("type") == "Ready" and cond.get("status") == "True":
                is_ready = True
            if cond.get("type") == "EDLGPUHealthy" and \
                    cond.get("status") == "False":
                gpu_healthy = False
        if is_ready and gpu_healthy:
            ready[node["metadata"]["name"]] = True
    return ready


def rendezvous_epoch(job: AITrainingJob) -> int:
    try:
        return int(job.annotations.get(EPOCH_ANNOTATION, "0"))
    except ValueError:
        return 0


class PodReconciler:
    def __init__(self, api: KubeApi, options, recorder=None,
                 expectations=None):
        self.expectations = expectations
        self.api = api
        self.options = options
        self.recorder = recorder

    # -- creation ---------------------------------------------------------
    def create_new_pod(self, job: AITrainingJob, rt: str, index: int,
                       restart_count: int, spec: ReplicaSpec) -> None:
        """reference: pod.go:483-546."""
        labels = ko.gen_labels(job.name)
        labels[C.LABEL_JOB_NAME_SHORT] = job.name
        labels[C.LABEL_POD_ROLE] = rt
        labels[C.LABEL_RESTART_COUNT] = str(restart_count)
        labels[C.LABEL_REPLICA_NAME] = rt
        labels[C.LABEL_REPLICA_INDEX] = str(index)
        if job.spec.priority:
            labels[C.LABEL_PRIORITY] = job.spec.priority
        if job.spec.framework_type:
            labels[C.LABEL_FRAMEWORK] = job.spec.framework_type

        template = copy.deepcopy(spec.template)
        m = template.setdefault("metadata", {})
        m["name"] = gen_general_name(job.name, rt, index)
        tmpl_labels = m.setdefault("labels", {})
        tmpl_labels.update(labels)
        for k, v in (job.metadata.get("labels") or {}).items():
            tmpl_labels.setdefault(k, v)
        m["ownerReferences"] = [ko.gen_owner_reference(job)]
        m["annotations"] = {
            **(m.get("annotations") or {}),
            WORLD_SIZE_ANNOTATION: str(spec.replicas or 0),
            EPOCH_ANNOTATION: str(rendezvous_epoch(job)),
        }

        pspec = template.setdefault("spec", {})
        if job.spec.scheduler_name:
            pspec["schedulerName"] = job.spec.scheduler_name
        if spec.restart_policy:
            # the operator owns restarts; the kubelet must not race it
            # (reference: pod.go:532-535)
            pspec["restartPolicy"] = "Never"

        inject_env(template, job, rt, index, restart_count,
                   epoch=rendezvous_epoch(job))

        pod = {"apiVersion": "v1", "kind": "Pod", "metadata": m,
               "spec": pspec}
        # expectation raised before the call, settled on ANY definitive
        # answer (success ack or clean ApiError) -- our reads are live
        # LISTs, not an informer cache, so the ack is authoritative; the
        # guard protects exactly the ambiguous window (a transport error
        # where the create may or may not have landed), unlike the
        # reference whose cache lag needs the watch event
        # (controller.go:390-404, pod.go:489-494)
        if self.expectations:
            self.expectations.expect_creation(job.key, f"pod/{m['name']}")
        try:
            self.api.create_pod(job.namespace, pod)
            if self.expectations:
                self.expectations.creation_observed(job.key,
                                                    f"pod/{m['name']}")
        except ApiError as e:
            if self.expectations:   # clean failure: nothing in flight
                self.expectations.creation_observed(job.key,
                                                    f"pod/{m['name']}")
            if not e.already_exists:
                raise
        if self.recorder:
            self.recorder.event(job, "Normal", "SuccessfulCreatePod",
                                f"created pod {m['name']}")

    def _delete_pod(self, job: AITrainingJob, pod: dict,
                    force: bool = False) -> None:
        grace = 0 if force else None
        if self.expectations:
            self.expectations.expect_deletion(job.key,
                                              f"pod/{ko.name_of(pod)}")
        try:
            self.api.delete_pod(ko.namespace_of(pod), ko.name_of(pod),
                                grace_period=grace)
            if self.expectations:
                self.expectations.deletion_observed(
                    job.key, f"pod/{ko.name_of(pod)}")
        except ApiError as e:
            if self.expectations:
                self.expectations.deletion_observed(
                    job.key, f"pod/{ko.name_of(pod)}")
            if not e.not_found:
                raise
        if self.recorder:
            self.recorder.event(job, "Normal", "SuccessfulDeletePod",
                                f"deleted pod {ko.name_of(pod)}")

    # -- reconcile one replica type ---------------------------------------
    def reconcile(self, job: AITrainingJob, all_pods: List[dict], rtype: str,
                  node_ready: Dict[str, bool],
                  now: float) -> Tuple[str, str]:
        """Returns (ending_phase, message) exactly like the reference's
        reconcilePods (pod.go:152-326); mutates job.status counters."""
        if job.status.phase == Phase.TERMINATING:
            return Phase.TERMINATING, ""
        if Phase.PREEMPTED in job.annotations:
            return Phase.PREEMPTED, job.annotations[Phase.PREEMPTED]
        if Phase.FAILED in job.annotations:
            return Phase.FAILED, job.annotations[Phase.FAILED]

        rt = rtype.lower()
        spec = job.spec.replica_specs[rtype]
        replicas = spec.replicas or 0
        replica_pods = filter_pods_for_replica_type(all_pods, rt)
        job.status.replica_statuses[rtype] = engine.ReplicaStatus()
        job.status.restart_counts.setdefault(rtype, 0)

        # --- EdlPolicy=Auto: controller-chosen world size in
        # [minReplicas, maxReplicas] (new semantics).
        # Scale-down when pods sit unschedulable past the grace period;
        # +1 scale-up probes toward maxReplicas once everything runs and
        # the probe interval elapsed. The chosen size is sticky via the
        # target-replicas annotation; the world-size stale check below
        # then drives the actual restart-at-new-size dance.
        if spec.edl_policy == EdlPolicy.AUTO:
            effective = self._auto_target(job, rtype, spec,
                                          replica_pods, now)
            if effective != replicas:
                # in-memory only: spec is never persisted, but this sync's
                # pod creation/env injection and the services reconciler
                # must all see the effective size
                spec.replicas = effective
                replicas = effective

        # --- elastic resize detection (new semantics) ---
        if spec.edl_policy and spec.edl_policy != EdlPolicy.NEVER:
            stale = [p for p in replica_pods
                     if (p["metadata"].get("annotations") or {})
                     .get(WORLD_SIZE_ANNOTATION) not in (None, str(replicas))]
            stale += out_of_range_pods(replica_pods, replicas)
            if stale:
                msg = (f"elastic resize of {rt} to {replicas} replicas: "
                       "restarting world")
                log.info("%s: %s", job.key, msg)
                for p in replica_pods:
                    self._delete_pod(job, p)
                job.annotations[EPOCH_ANNOTATION] = str(
                    rendezvous_epoch(job) + 1)
                self._count(job, rtype, replica_pods)
                return Phase.RESTARTING, msg

        slices = pod_slices(replica_pods, replicas)
        message = ""
        failed_reason: List[str] = []
        failed_phase = Phase.FAILED
        creating_msgs: Dict[str, List[str]] = {}
        creating_cond = engine.get_condition(job.status, Phase.CREATING)
        creating_transition = None
        if creating_cond is not None and creating_cond.status == "True":
            from ..utils.k8stime import parse_time
            creating_transition = parse_time(creating_cond.last_transition_time)

        for index, pslice in enumerate(slices):
            if not pslice:
                log.info("%s: creating pod %s-%d", job.key, rt, index)
                self.create_new_pod(job, rt, index,
                                    job.status.restart_counts[rtype], spec)
                continue
            pod = pslice[0]
            sched_msg = self._scheduling_message(pod)
            if sched_msg:
                message = f"{rt}: {sched_msg} "
            d = engine.container_decision(
                pod, spec.restart_policy, job.spec.restarting_exit_code,
                node_ready, creating_transition,
                self.options.creating_failure_policy(), now,
                fault_tolerant=job.spec.fault_tolerant)
            if d.message and d.phase == Phase.FAILED:
                failed_reason.append(d.message)

            if d.is_restart:
                force = d.phase == Phase.NODE_FAIL
                if engine.restart_allowed(job, rtype):
                    engine.bump_restart_count(job, rtype)
                    msg = (f"restart times is "
                           f"{job.status.restart_counts[rtype]}, {d.message} ")
                    if spec.restart_scope == RestartScope.POD:
                        self._delete_pod(job, pod, force)
                        self._count(job, rtype, replica_pods)
                        return Phase.RESTARTING, msg
                    if spec.restart_scope == RestartScope.REPLICA:
                        for p in replica_pods:
                            self._delete_pod(job, p, force)
                        self._count(job, rtype, replica_pods)
                        return Phase.RESTARTING, msg
                    if spec.restart_scope == RestartScope.ALL:
                        for p in all_pods:
                            self._delete_pod(job, p, force)
                        for urt in job.spec.replica_specs:
                            self._count(job, urt,
                                        filter_pods_for_replica_type(
                                            all_pods, urt.lower()))
                        return Phase.RESTARTING, msg

            if d.phase == Phase.CREATING:
                creating_msgs.setdefault(d.message, []).append(
                    ko.name_of(pod))

            if (d.phase == Phase.SUCCEEDED
                    and ko.pod_phase(pod) == "Succeeded"
                    and spec.complete_policy == EndingPolicy.ANY):
                return d.phase, f"pod {ko.name_of(pod)} have completed"

            if (d.phase in (Phase.FAILED, Phase.NODE_FAIL)
                    and spec.fail_policy == EndingPolicy.ANY):
                return d.phase, (f"pod {ko.name_of(pod)} is failed, "
                                 f"{d.message}")

            if index == 0:
                if (d.phase == Phase.SUCCEEDED
                        and ko.pod_phase(pod) == "Succeeded"
                        and spec.complete_policy == EndingPolicy.RANK0):
                    return Phase.SUCCEEDED, \
                        f"rank0 pod {ko.name_of(pod)} have completed"
                if (d.phase in (Phase.FAILED, Phase.NODE_FAIL)
                        and spec.fail_policy == EndingPolicy.RANK0):
                    return d.phase, (f"rank0 pod {ko.name_of(pod)} is "
                                     f"failed, {d.message}")

            if d.phase == Phase.NODE_FAIL:
                failed_phase = Phase.NODE_FAIL

        self._count(job, rtype, replica_pods)
        rs = job.status.replica_statuses[rtype]

        if spec.complete_policy == EndingPolicy.ALL \
                and rs.succeeded == replicas:
            return Phase.SUCCEEDED, f"All {rtype} pods have completed"
        if spec.fail_policy == EndingPolicy.ALL and rs.failed == replicas:
            if failed_reason:
                message = ", ".join(failed_reason)
            return failed_phase, f"All {rtype} pods are failed, {message}"

        if creating_msgs:
            parts = [f"pods {names} {msg}"
                     for msg, names in creating_msgs.items()]
            return Phase.NONE, ", ".join(parts)
        return Phase.NONE, message

    # -- helpers ----------------------------------------------------------
    def _count(self, job: AITrainingJob, rtype: str, pods: List[dict]):
        job.status.replica_statuses[rtype] = engine.count_replica_statuses(
            job.status.restart_counts.get(rtype, 0), pods)

    def _auto_target(self, job: AITrainingJob, rtype: str,
                     spec: ReplicaSpec, replica_pods: List[dict],
                     now: float) -> int:
        """EdlPolicy=Auto control loop: returns the effective replica
        count for this sync and records it (plus the last-resize time) in
        job annotations so it is sticky across syncs and restarts."""
        from ..utils.k8stime import format_time, parse_time
        rt = rtype.lower()
        base = spec.replicas or 0
        lo = spec.min_replicas if spec.min_replicas is not None else base
        hi = spec.max_replicas if spec.max_replicas is not None else base
        key = f"{TARGET_ANNOTATION}-{rt}"
        tkey = f"{LAST_RESIZE_ANNOTATION}-{rt}"
        try:
            target = int(job.annotations[key])
        except (KeyError, ValueError):
            target = base
        target = max(lo, min(hi, target))

        def resize(new: int, why: str) -> int:
            job.annotations[key] = str(new)
            job.annotations[tkey] = format_time(now)
            msg = (f"EdlPolicy=Auto: {rt} {target} -> {new} replicas "
                   f"({why})")
            log.info("%s: %s", job.key, msg)
            if self.recorder:
                self.recorder.event(job, "Normal", "ElasticScale", msg)
            return new

        # a LIVE spec.replicas patch on an Auto role resets the target:
        # explicit user intent outranks the control loop's last choice
        bkey = f"{TARGET_ANNOTATION}-base-{rt}"
        seen_base = job.annotations.get(bkey)
        if seen_base != str(base):
            job.annotations[bkey] = str(base)
            if seen_base is not None:
                return resize(max(lo, min(hi, base)),
                              "manual spec.replicas change")

        grace = self.options.elastic_unschedulable_grace
        stuck = [p for p in replica_pods
                 if (s := self._unschedulable_since(p)) is not None
                 and now - s >= grace]
        if stuck and target > lo:
            scheduled = len([p for p in replica_pods if ko.pod_node(p)])
            new = max(lo, min(target - 1, scheduled))
            return resize(new, f"{len(stuck)} pods unschedulable "
                               f">{grace:.0f}s")

        if not stuck and target < hi:
            last = parse_time(job.annotations.get(tkey))
            if last is None:
                # start the probe clock on first sight, don't scale yet
                job.annotations[tkey] = format_time(now)
            else:
                running = len([p for p in replica_pods
                               if ko.pod_phase(p) == "Running"])
                if (running >= target
                        and now - last >= self.options
                        .elastic_scaleup_interval):
                    return resize(min(hi, target + 1), "capacity probe")

        job.annotations[key] = str(target)
        return target

    @staticmethod
    def _unschedulable_since(pod: dict):
        """Epoch seconds since the scheduler marked the pod unschedulable
        (PodScheduled=False while Pending and unbound), else None."""
        if ko.pod_phase(pod) != "Pending" or ko.pod_node(pod):
            return None
        from ..utils.k8stime import parse_time
        for cond in (pod.get("status") or {}).get("conditions") or []:
            if cond.get("type") == "PodScheduled" and \
                    cond.get("status") == "False":
                return (parse_time(cond.get("lastTransitionTime"))
                        or parse_time((pod.get("metadata") or {})
                                      .get("creationTimestamp"))
                        or 0.0)
        return None

    @staticmethod
    def _scheduling_message(pod: dict) -> str:
        """reference: pod.go:457-467."""
        if ko.pod_phase(pod) == "Pending" and not ko.pod_node(pod):
            for cond in (pod.get("status") or {}).get("conditions") or []:
                if cond.get("type") == "PodScheduled" and \
                        cond.get("status") == "False":
                    return cond.get("message", "")
        return ""
