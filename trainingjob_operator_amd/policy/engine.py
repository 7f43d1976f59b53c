"""Pure-function policy engine: restart / ending / phase decisions.

This is the behavioral core of the reference's pod reconciler and status
machine, factored into side-effect-free functions so the full policy matrix
(6 restart policies x 3 scopes x exit-code lists x ending policies) is
unit-testable without any cluster (the reference has zero tests).

Reference semantics reproduced from:
  * pkg/controller/pod.go:328-437      (reconcileContainers)
  * pkg/controller/controller.go:442-462 (isRetryableExitCode / checkExitCode)
  * pkg/controller/status.go:13-99     (condition helpers)
  * pkg/controller/status.go:307-380   (replica status counters)
  * pkg/controller/status.go:144-174   (job-level ending aggregation)

Pods are plain dicts in corev1 shape.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional

from ..api.constants import CONTAINER_PREFIX, ERROR_CONTAINER_STATUS
from ..api.types import (
    AITrainingJob,
    Condition,
    EndingPolicy,
    ENDING_PHASES,
    Phase,
    PHASE_REASON,
    ReplicaStatus,
    RestartPolicy,
    RestartScope,
    TrainingJobStatus,
)
from ..utils.k8stime import format_time, parse_time


# ---------------------------------------------------------------------------
# Exit codes
# ---------------------------------------------------------------------------

def is_retryable_exit_code(exit_codes: List[int], restarting_exit_code: str) -> bool:
    """True iff every collected exit code appears in the comma-separated
    retryable list (reference: controller.go:442-453 — note the AND over all
    codes, and False for an empty code list)."""
    if not exit_codes:
        return False
    allowed = {part.strip() for part in restarting_exit_code.split(",")}
    return all(str(code) in allowed for code in exit_codes)


# ---------------------------------------------------------------------------
# Container-level decision (the reconcileContainers core)
# ---------------------------------------------------------------------------

@dataclass
class CreatingFailurePolicy:
    """Knobs from the operator options (reference: cmd/app/options/options.go:17-19,39-49).

    creating_restart_seconds: window (since the job's Creating condition
      transition) inside which creation errors trigger pod restarts; past it,
      with enable_creating_failed, the job is failed outright.
    creating_duration_seconds: how long a pod may sit in a creation-error
      waiting state before a restart fires.
    """

    creating_restart_seconds: float = 5.0
    creating_duration_seconds: float = 15 * 60.0
    enable_creating_failed: bool = False


@dataclass
class ContainerDecision:
    phase: str = Phase.NONE
    is_restart: bool = False
    message: str = ""


def container_decision(
    pod: dict,
    restart_policy: str,
    restarting_exit_code: str,
    node_ready: Dict[str, bool],
    creating_condition_transition: Optional[float],
    policy: CreatingFailurePolicy,
    now: float,
    fault_tolerant: bool = False,
) -> ContainerDecision:
    """Map one pod's container statuses to (phase, is_restart, message).

    Mirrors reference pkg/controller/pod.go:328-437.

    node_ready maps node name -> True for Ready nodes; a pod bound to a node
    absent from the map is treated as on a failed node (reference
    pod.go:439-455 builds the map only from Ready nodes).
    creating_condition_transition: unix time of the job's Creating condition
    lastTransitionTime, or None if the job has no True Creating condition.
    fault_tolerant: spec.faultTolerant — widens node-failure restarts to
    every restart policy except Never (new semantics for a dead ref field).
    """
    status = pod.get("status") or {}
    pod_spec = pod.get("spec") or {}
    exit_codes: List[int] = []
    failed_reason: List[str] = []
    is_restart = False
    is_succeeded = True
    is_creating = False

    for cstatus in status.get("containerStatuses") or []:
        state = cstatus.get("state") or {}
        terminated = state.get("terminated")
        if (cstatus.get("name") or "").startswith(CONTAINER_PREFIX):
            is_succeeded = is_succeeded and terminated is not None
            if terminated is not None:
                code = int(terminated.get("exitCode", 0))
                is_succeeded = is_succeeded and code == 0
                exit_codes.append(code)
                if code != 0:
                    failed_reason.append(
                        "container %s on node %s exited with reason %s exitcode %s"
                        % (cstatus.get("name"), pod_spec.get("nodeName"),
                           terminated.get("reason"), code)
                    )
        waiting = state.get("waiting")
        if waiting is not None:
            is_creating = True
            reason = waiting.get("reason", "")
            if reason in ERROR_CONTAINER_STATUS:
                if creating_condition_transition is not None:
                    if now - creating_condition_transition < policy.creating_restart_seconds:
                        start = parse_time(status.get("startTime"))
                        if start is not None and \
                                now - start > policy.creating_duration_seconds:
                            is_restart = True
                    elif policy.enable_creating_failed:
                        msg = (
                            "pod %s create container failed[%s] and has been "
                            "retrying for %s seconds"
                            % (pod.get("metadata", {}).get("name"), reason,
                               policy.creating_restart_seconds)
                        )
                        return ContainerDecision(Phase.FAILED, is_restart, msg)
                failed_reason.append(reason)

    if status.get("phase") == "Failed":
        if (restart_policy in (RestartPolicy.EXIT_CODE,
                               RestartPolicy.ON_NODE_FAIL_WITH_EXIT_CODE)
                and is_retryable_exit_code(exit_codes, restarting_exit_code)) \
                or restart_policy in (RestartPolicy.ON_FAILURE, RestartPolicy.ALWAYS):
            is_restart = True
        if failed_reason:
            message = "; ".join(failed_reason)
        elif status.get("reason"):
            message = status["reason"]
            if status.get("message"):
                message = f"{status['reason']}, {status['message']}"
        else:
            message = ""
        return ContainerDecision(Phase.FAILED, is_restart, message)

    node_name = pod_spec.get("nodeName") or ""
    if node_name and node_name not in node_ready:
        if restart_policy in (RestartPolicy.ON_NODE_FAIL_WITH_EXIT_CODE,
                              RestartPolicy.ON_NODE_FAIL, RestartPolicy.ALWAYS):
            is_restart = True
        elif fault_tolerant and restart_policy != RestartPolicy.NEVER:
            # spec.faultTolerant (declared but dead in the reference,
            # SURVEY.md C15): node loss is retryable under ANY restart
            # policy except Never — the job survives node failures even
            # when its policy only covers container exits
            is_restart = True
        return ContainerDecision(
            Phase.NODE_FAIL, is_restart,
            f"Node {node_name} is failed and offline")

    if is_creating:
        if failed_reason:
            return ContainerDecision(Phase.CREATING, is_restart,
                                     "; ".join(failed_reason))
        return ContainerDecision(Phase.CREATING, is_restart, "creating containers")

    if is_succeeded:
        return ContainerDecision(Phase.SUCCEEDED, is_restart, "")

    return ContainerDecision(Phase.NONE, is_restart, "")


# ---------------------------------------------------------------------------
# Restart accounting
# ---------------------------------------------------------------------------

def bump_restart_count(job: AITrainingJob, rtype: str) -> None:
    """Scope All bumps EVERY replica type's counter (reference:
    status.go:322-330)."""
    spec = job.spec.replica_specs[rtype]
    if spec.restart_scope == RestartScope.ALL:
        for rt in job.spec.replica_specs:
            job.status.restart_counts[rt] = job.status.restart_counts.get(rt, 0) + 1
    else:
        job.status.restart_counts[rtype] = job.status.restart_counts.get(rtype, 0) + 1


def restart_allowed(job: AITrainingJob, rtype: str) -> bool:
    """restartLimit gate (reference: pod.go:215-216): nil limit == unlimited."""
    limit = job.spec.replica_specs[rtype].restart_limit
    if limit is None:
        return True
    return job.status.restart_counts.get(rtype, 0) < limit


# ---------------------------------------------------------------------------
# Replica status counters
# ---------------------------------------------------------------------------

def count_replica_statuses(restart_count: int, pods: List[dict]) -> ReplicaStatus:
    """Pod phases -> per-role counters (reference: status.go:332-359).

    A Pending pod counts as Restarting whenever the role has ever restarted
    (restart_count > 0), else Scheduled if bound to a node, else Pending.
    Unknown counts as Failed.
    """
    rs = ReplicaStatus()
    for pod in pods:
        phase = (pod.get("status") or {}).get("phase", "Pending")
        if phase == "Pending":
            if restart_count > 0:
                rs.restarting += 1
            elif (pod.get("spec") or {}).get("nodeName"):
                rs.scheduled += 1
            else:
                rs.pending += 1
        elif phase == "Running":
            rs.active += 1
        elif phase == "Succeeded":
            rs.succeeded += 1
        else:  # Failed | Unknown
            rs.failed += 1
    return rs


# ---------------------------------------------------------------------------
# Conditions
# ---------------------------------------------------------------------------

def get_condition(status: TrainingJobStatus, ctype: str) -> Optional[Condition]:
    for cond in status.conditions:
        if cond.type == ctype:
            return cond
    return None


def is_job_completed(status: TrainingJobStatus) -> bool:
    """reference: status.go:35-59 — Succeed/Failed/Preempted/Timeout condition
    True means the job is finished and conditions freeze."""
    for ctype in (Phase.SUCCEEDED, Phase.FAILED, Phase.PREEMPTED, Phase.TIMEOUT):
        cond = get_condition(status, ctype)
        if cond is not None and cond.status == "True":
            return True
    return False


def set_condition(status: TrainingJobStatus, cond: Condition) -> None:
    """Append-only condition list; the previous last condition flips to False
    unless the new one merely updates its message (reference: status.go:60-75)."""
    if status.conditions:
        curr = status.conditions[-1]
        if (curr.type == cond.type and curr.status == cond.status
                and curr.reason == cond.reason):
            curr.message = cond.message
            return
        curr.status = "False"
    status.conditions.append(cond)


def update_job_conditions(job: AITrainingJob, phase: str, message: str,
                          now: float) -> None:
    """Set phase + append condition, unless the job already completed
    (reference: status.go:89-99)."""
    if is_job_completed(job.status):
        return
    ts = format_time(now)
    set_condition(job.status, Condition(
        type=phase, status="True", reason=PHASE_REASON.get(phase, ""),
        message=message, last_probe_time=ts, last_transition_time=ts))
    job.status.phase = phase


def is_failed_phase(phase: str) -> bool:
    """Terminal and not Succeed (reference: status.go:~365-380)."""
    return phase in ENDING_PHASES and phase != Phase.SUCCEEDED


# ---------------------------------------------------------------------------
# Job-level ending aggregation
# ---------------------------------------------------------------------------

@dataclass
class JobEndingDecision:
    terminate: bool = False
    phase: str = Phase.NONE
    message: str = ""


def aggregate_job_ending(
    job: AITrainingJob,
    replica_phases: Dict[str, str],
    message: str,
) -> JobEndingDecision:
    """Job-level CompletePolicy/FailPolicy over per-replica-type ending phases;
    Complete beats Fail (reference: status.go:144-174)."""
    completed = sum(1 for p in replica_phases.values() if p == Phase.SUCCEEDED)
    failed = 0
    ending_phase = Phase.NONE
    for p in replica_phases.values():
        if is_failed_phase(p):
            failed += 1
            ending_phase = p
    n = len(job.spec.replica_specs)
    spec = job.spec

    if spec.complete_policy == EndingPolicy.ANY and completed > 0:
        return JobEndingDecision(True, Phase.SUCCEEDED, f"job {job.name} completed")
    if spec.complete_policy == EndingPolicy.ALL and completed == n:
        return JobEndingDecision(True, Phase.SUCCEEDED, f"job {job.name} completed")
    if spec.fail_policy == EndingPolicy.ANY and failed > 0:
        return JobEndingDecision(True, ending_phase, message)
    if spec.fail_policy == EndingPolicy.ALL and failed == n:
        return JobEndingDecision(True, ending_phase, message)
    return JobEndingDecision(False)
