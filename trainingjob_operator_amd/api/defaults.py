"""Defaulting for AITrainingJob, applied in the sync path before reconcile
(reference: pkg/apis/aitrainingjob/v1/defaults.go:15-53, applied at
pkg/controller/controller.go:297).

Defaults:
  cleanPodPolicy -> All; job failPolicy -> Any; job completePolicy -> All;
  per-replica: replicas -> 1, restartPolicy -> Never, restartScope -> All,
  failPolicy -> Any, completePolicy -> All.

MI355X extension (real semantics for fields the reference declared but never
read — SURVEY.md §C15): minReplicas/maxReplicas default to replicas when an
edlPolicy other than Never is set, so the elastic controller always has a
well-formed range.
"""
from .types import (
    AITrainingJob,
    CleanPodPolicy,
    EdlPolicy,
    EndingPolicy,
    ReplicaSpec,
    RestartPolicy,
    RestartScope,
)


def set_default_replica_spec(spec: ReplicaSpec) -> None:
    """reference: defaults.go:15-31."""
    if spec.replicas is None:
        spec.replicas = 1
    if spec.restart_policy == "":
        spec.restart_policy = RestartPolicy.NEVER
    if spec.restart_scope == "":
        spec.restart_scope = RestartScope.ALL
    if spec.fail_policy == "":
        spec.fail_policy = EndingPolicy.ANY
    if spec.complete_policy == "":
        spec.complete_policy = EndingPolicy.ALL
    # Elastic range defaults (new semantics; unused fields in the reference,
    # replica.go:10-11).
    if spec.edl_policy and spec.edl_policy != EdlPolicy.NEVER:
        if spec.min_replicas is None:
            spec.min_replicas = spec.replicas
        if spec.max_replicas is None:
            spec.max_replicas = max(spec.replicas, spec.min_replicas)


def set_defaults(job: AITrainingJob) -> AITrainingJob:
    """reference: defaults.go:34-53. Mutates and returns the job."""
    if job.spec.clean_pod_policy is None:
        job.spec.clean_pod_policy = CleanPodPolicy.ALL
    if job.spec.fail_policy == "":
        job.spec.fail_policy = EndingPolicy.ANY
    if job.spec.complete_policy == "":
        job.spec.complete_policy = EndingPolicy.ALL
    for spec in job.spec.replica_specs.values():
        set_default_replica_spec(spec)
    return job
