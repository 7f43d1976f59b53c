"""AITrainingJob API types.

Schema-compatible with the reference operator's CRD
(reference: pkg/apis/aitrainingjob/v1/types.go:29-142 and replica.go:9-63),
including its JSON-tag quirks:

* ``status.RestartCount`` — the Go field ``RestartCountes`` carries the
  malformed tag ``json:"RestartCount,,omitempty"`` (types.go:84), which Go's
  encoder parses as name ``RestartCount`` with ``omitempty`` active.
* ``status.RestartReplicaName`` — no JSON tag at all (types.go:86), so Go
  serializes it under the exact field name ``RestartReplicaName`` and it is
  NOT omitted when empty.
* The success phase is the string ``"Succeed"`` (types.go:108), not
  "Succeeded".

Pod templates are carried as plain dicts (corev1.PodTemplateSpec shape) —
they pass through to the API server untouched except for env injection.
"""
from __future__ import annotations

import copy
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


# --- Phases (reference: types.go:100-124) ---
class Phase:
    NONE = ""
    PENDING = "Pending"
    CREATING = "Creating"
    RUNNING = "Running"
    SUCCEEDED = "Succeed"  # NB: reference spells it "Succeed"
    FAILED = "Failed"
    TIMEOUT = "Timeout"
    RESTARTING = "Restarting"
    TERMINATING = "Terminating"
    PREEMPTED = "Preempted"
    NODE_FAIL = "NodeFail"


# Terminal phases (reference: constants.go:58-64).
ENDING_PHASES = (
    Phase.SUCCEEDED,
    Phase.FAILED,
    Phase.TIMEOUT,
    Phase.PREEMPTED,
    Phase.NODE_FAIL,
)

# Phases the controller actively reconciles (reference: controller.go:298-304).
ACTIVE_PHASES = (
    Phase.NONE,
    Phase.PENDING,
    Phase.CREATING,
    Phase.RUNNING,
    Phase.RESTARTING,
    Phase.TERMINATING,
)

# phase -> condition/event reason (reference: constants.go:65-77).
PHASE_REASON = {
    Phase.NONE: "",
    Phase.PENDING: "TrainingJobPending",
    Phase.CREATING: "TrainingJobCreating",
    Phase.RUNNING: "TrainingJobRunning",
    Phase.SUCCEEDED: "TrainingJobSucceed",
    Phase.FAILED: "TrainingJobFailed",
    Phase.TIMEOUT: "TrainingJobTimeout",
    Phase.RESTARTING: "TrainingJobRestarting",
    Phase.TERMINATING: "TrainingJobTerminating",
    Phase.PREEMPTED: "TrainingJobPreempted",
    Phase.NODE_FAIL: "TrainingJobNodeFail",
}


# --- Policies (reference: replica.go:24-34, 51-63; types.go:67-72) ---
class RestartPolicy:
    ALWAYS = "Always"
    ON_FAILURE = "OnFailure"
    ON_NODE_FAIL = "OnNodeFail"
    NEVER = "Never"
    EXIT_CODE = "ExitCode"
    ON_NODE_FAIL_WITH_EXIT_CODE = "OnNodeFailWithExitCode"

    ALL = (ALWAYS, ON_FAILURE, ON_NODE_FAIL, NEVER, EXIT_CODE,
           ON_NODE_FAIL_WITH_EXIT_CODE)


class RestartScope:
    ALL = "All"
    REPLICA = "Replica"
    POD = "Pod"

    VALUES = (ALL, REPLICA, POD)


class EndingPolicy:
    ALL = "All"
    RANK0 = "Rank0"
    ANY = "Any"
    NONE = "None"

    VALUES = (ALL, RANK0, ANY, NONE)


class CleanPodPolicy:
    ALL = "All"
    NONE = "None"

    VALUES = (ALL, NONE)


class EdlPolicy:
    AUTO = "Auto"
    MANUAL = "Manual"
    NEVER = "Never"

    VALUES = (AUTO, MANUAL, NEVER)


def _opt(d: Dict[str, Any], key: str, value: Any) -> None:
    """Set key only when value is truthy-meaningful (Go omitempty semantics)."""
    if value is None:
        return
    if value == "" or value == 0 or value is False or value == {} or value == []:
        return
    d[key] = value


@dataclass
class ReplicaSpec:
    """One role's replica description (reference: replica.go:9-21)."""

    min_replicas: Optional[int] = None
    max_replicas: Optional[int] = None
    replicas: Optional[int] = None
    restart_limit: Optional[int] = None
    template: Dict[str, Any] = field(default_factory=dict)
    restart_policy: str = ""
    restart_scope: str = ""
    fail_policy: str = ""
    complete_policy: str = ""
    edl_policy: str = ""

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ReplicaSpec":
        return cls(
            min_replicas=d.get("minReplicas"),
            max_replicas=d.get("maxReplicas"),
            replicas=d.get("replicas"),
            restart_limit=d.get("restartLimit"),
            template=copy.deepcopy(d.get("template") or {}),
            restart_policy=d.get("restartPolicy", ""),
            restart_scope=d.get("restartScope", ""),
            fail_policy=d.get("failPolicy", ""),
            complete_policy=d.get("completePolicy", ""),
            edl_policy=d.get("edlPolicy", ""),
        )

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {}
        _opt(d, "minReplicas", self.min_replicas)
        _opt(d, "maxReplicas", self.max_replicas)
        _opt(d, "replicas", self.replicas)
        _opt(d, "restartLimit", self.restart_limit)
        if self.template:
            d["template"] = copy.deepcopy(self.template)
        _opt(d, "restartPolicy", self.restart_policy)
        _opt(d, "restartScope", self.restart_scope)
        _opt(d, "failPolicy", self.fail_policy)
        _opt(d, "completePolicy", self.complete_policy)
        _opt(d, "edlPolicy", self.edl_policy)
        return d


@dataclass
class TrainingJobSpec:
    """reference: types.go:41-62."""

    restarting_exit_code: str = ""
    framework_type: str = ""
    fault_tolerant: bool = False
    priority: str = ""
    scheduler_name: str = ""
    time_limit: Optional[int] = None
    clean_pod_policy: Optional[str] = None
    fail_policy: str = ""
    complete_policy: str = ""
    replica_specs: Dict[str, ReplicaSpec] = field(default_factory=dict)

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "TrainingJobSpec":
        return cls(
            restarting_exit_code=d.get("restartingExitCode", ""),
            framework_type=d.get("frameworkType", ""),
            fault_tolerant=bool(d.get("faultTolerant", False)),
            priority=d.get("priority", ""),
            scheduler_name=d.get("schedulerName", ""),
            time_limit=d.get("timeLimit"),
            clean_pod_policy=d.get("cleanPodPolicy"),
            fail_policy=d.get("failPolicy", ""),
            complete_policy=d.get("completePolicy", ""),
            replica_specs={
                name: ReplicaSpec.from_dict(spec or {})
                for name, spec in (d.get("replicaSpecs") or {}).items()
            },
        )

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {}
        _opt(d, "restartingExitCode", self.restarting_exit_code)
        _opt(d, "frameworkType", self.framework_type)
        _opt(d, "faultTolerant", self.fault_tolerant)
        _opt(d, "priority", self.priority)
        _opt(d, "schedulerName", self.scheduler_name)
        _opt(d, "timeLimit", self.time_limit)
        if self.clean_pod_policy is not None:
            d["cleanPodPolicy"] = self.clean_pod_policy
        _opt(d, "failPolicy", self.fail_policy)
        _opt(d, "completePolicy", self.complete_policy)
        d["replicaSpecs"] = {
            name: spec.to_dict() for name, spec in self.replica_specs.items()
        }
        return d


@dataclass
class ReplicaStatus:
    """Per-role pod counters (reference: replica.go:36-49)."""

    pending: int = 0
    scheduled: int = 0
    active: int = 0
    succeeded: int = 0
    restarting: int = 0
    failed: int = 0

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ReplicaStatus":
        return cls(
            pending=d.get("pending", 0),
            scheduled=d.get("scheduled", 0),
            active=d.get("active", 0),
            succeeded=d.get("succeeded", 0),
            restarting=d.get("restarting", 0),
            failed=d.get("failed", 0),
        )

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {}
        _opt(d, "pending", self.pending)
        _opt(d, "scheduled", self.scheduled)
        _opt(d, "active", self.active)
        _opt(d, "succeeded", self.succeeded)
        _opt(d, "restarting", self.restarting)
        _opt(d, "failed", self.failed)
        return d


@dataclass
class Condition:
    """reference: types.go:128-142."""

    type: str = ""
    status: str = ""  # "True" | "False" | "Unknown"
    reason: str = ""
    message: str = ""
    last_probe_time: str = ""
    last_transition_time: str = ""

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "Condition":
        return cls(
            type=d.get("type", ""),
            status=d.get("status", ""),
            reason=d.get("reason", ""),
            message=d.get("message", ""),
            last_probe_time=d.get("lastProbeTime", ""),
            last_transition_time=d.get("lastTransitionTime", ""),
        )

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"type": self.type, "status": self.status}
        _opt(d, "reason", self.reason)
        _opt(d, "message", self.message)
        _opt(d, "lastProbeTime", self.last_probe_time)
        _opt(d, "lastTransitionTime", self.last_transition_time)
        return d


@dataclass
class TrainingJobStatus:
    """reference: types.go:76-95."""

    phase: str = Phase.NONE
    conditions: List[Condition] = field(default_factory=list)
    replica_statuses: Dict[str, ReplicaStatus] = field(default_factory=dict)
    restart_counts: Dict[str, int] = field(default_factory=dict)
    restart_replica_name: str = ""
    start_time: Optional[str] = None
    start_running_time: Optional[str] = None
    end_time: Optional[str] = None
    last_reconcile_time: Optional[str] = None

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "TrainingJobStatus":
        return cls(
            phase=d.get("phase", ""),
            conditions=[Condition.from_dict(c) for c in d.get("conditions") or []],
            replica_statuses={
                name: ReplicaStatus.from_dict(rs or {})
                for name, rs in (d.get("replicaStatuses") or {}).items()
            },
            restart_counts=dict(d.get("RestartCount") or {}),
            restart_replica_name=d.get("RestartReplicaName", ""),
            start_time=d.get("startTime"),
            start_running_time=d.get("startRunningTime"),
            end_time=d.get("endTime"),
            last_reconcile_time=d.get("lastReconcileTime"),
        )

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {
            "phase": self.phase,
            "conditions": [c.to_dict() for c in self.conditions],
            "replicaStatuses": {
                name: rs.to_dict() for name, rs in self.replica_statuses.items()
            },
            # Go tag json:"RestartCount,,omitempty" -> key "RestartCount"
            # (types.go:84); field without a tag serializes under its Go name
            # (types.go:86).
            "RestartReplicaName": self.restart_replica_name,
        }
        _opt(d, "RestartCount", dict(self.restart_counts))
        _opt(d, "startTime", self.start_time)
        _opt(d, "startRunningTime", self.start_running_time)
        _opt(d, "endTime", self.end_time)
        _opt(d, "lastReconcileTime", self.last_reconcile_time)
        return d


@dataclass
class AITrainingJob:
    """reference: types.go:29-39."""

    metadata: Dict[str, Any] = field(default_factory=dict)
    spec: TrainingJobSpec = field(default_factory=TrainingJobSpec)
    status: TrainingJobStatus = field(default_factory=TrainingJobStatus)

    # -- metadata conveniences -------------------------------------------
    @property
    def name(self) -> str:
        return self.metadata.get("name", "")

    @property
    def namespace(self) -> str:
        return self.metadata.get("namespace", "default")

    @property
    def uid(self) -> str:
        return self.metadata.get("uid", "")

    @property
    def annotations(self) -> Dict[str, str]:
        return self.metadata.setdefault("annotations", {})

    @property
    def key(self) -> str:
        return f"{self.namespace}/{self.name}"

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "AITrainingJob":
        return cls(
            metadata=copy.deepcopy(d.get("metadata") or {}),
            spec=TrainingJobSpec.from_dict(d.get("spec") or {}),
            status=TrainingJobStatus.from_dict(d.get("status") or {}),
        )

    def to_dict(self) -> Dict[str, Any]:
        from .constants import API_VERSION, CRD_KIND

        return {
            "apiVersion": API_VERSION,
            "kind": CRD_KIND,
            "metadata": copy.deepcopy(self.metadata),
            "spec": self.spec.to_dict(),
            "status": self.status.to_dict(),
        }

    def deep_copy(self) -> "AITrainingJob":
        return AITrainingJob.from_dict(self.to_dict())


def gen_general_name(job_name: str, rtype: str, index) -> str:
    """Pod / headless-service name ``{job}-{rtype}-{index}``
    (reference: pkg/controller/trainingjob.go:12-15)."""
    return f"{job_name}-{rtype}-{index}"
