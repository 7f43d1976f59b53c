"""Real validation for AITrainingJob specs.

The reference ships only a dead validation stub that does not compile and is
imported nowhere (reference: pkg/apis/aitrainingjob/validation/validation.go:10-32,
and the controller carries ``// FIXME: need to validate trainingjob`` at
pkg/controller/trainingjob.go:21,33). This module implements the validation
the reference never did. Call after defaulting.
"""
from __future__ import annotations

import re
from typing import List

from .constants import AMD_GPU_RESOURCE, CONTAINER_PREFIX, MI355X_HBM_BYTES
from .types import (
    AITrainingJob,
    CleanPodPolicy,
    EdlPolicy,
    EndingPolicy,
    RestartPolicy,
    RestartScope,
)

_DNS1123 = re.compile(r"^[a-z0-9]([-a-z0-9]*[a-z0-9])?$")
# Pod names are {job}-{rtype}-{index}; keep headroom under the 63-char
# DNS-label limit for the per-pod headless service names.
_MAX_NAME = 40


class ValidationError(ValueError):
    def __init__(self, errors: List[str]):
        self.errors = errors
        super().__init__("; ".join(errors))


def _parse_exit_codes(s: str) -> List[int]:
    """Comma-separated retryable exit codes, e.g. "137,128"
    (reference: pkg/controller/controller.go:442-453)."""
    codes = []
    for part in s.split(","):
        part = part.strip()
        if part:
            codes.append(int(part))
    return codes


def validate(job: AITrainingJob) -> List[str]:
    """Return a list of error strings (empty == valid)."""
    errors: List[str] = []
    spec = job.spec

    if not job.name:
        errors.append("metadata.name: required")
    elif not _DNS1123.match(job.name):
        errors.append(f"metadata.name: {job.name!r} is not a DNS-1123 label")
    elif len(job.name) > _MAX_NAME:
        errors.append(
            f"metadata.name: {job.name!r} longer than {_MAX_NAME} chars "
            "(pod/service names must stay within the 63-char DNS label limit)"
        )

    if spec.restarting_exit_code:
        try:
            _parse_exit_codes(spec.restarting_exit_code)
        except ValueError:
            errors.append(
                f"spec.restartingExitCode: {spec.restarting_exit_code!r} is not "
                "a comma-separated integer list"
            )

    if spec.time_limit is not None and spec.time_limit <= 0:
        errors.append("spec.timeLimit: must be > 0 seconds")

    if spec.clean_pod_policy not in (None, *CleanPodPolicy.VALUES):
        errors.append(f"spec.cleanPodPolicy: invalid value {spec.clean_pod_policy!r}")
    for fname, val in (("failPolicy", spec.fail_policy),
                       ("completePolicy", spec.complete_policy)):
        if val and val not in EndingPolicy.VALUES:
            errors.append(f"spec.{fname}: invalid value {val!r}")

    if not spec.replica_specs:
        errors.append("spec.replicaSpecs: at least one replica type is required")

    for rtype, rs in spec.replica_specs.items():
        p = f"spec.replicaSpecs[{rtype}]"
        if not _DNS1123.match(rtype.lower()):
            errors.append(f"{p}: replica type name must be a DNS-1123 label")
        if rs.replicas is not None and rs.replicas < 0:
            errors.append(f"{p}.replicas: must be >= 0")
        if rs.restart_limit is not None and rs.restart_limit < 0:
            errors.append(f"{p}.restartLimit: must be >= 0")
        if rs.restart_policy and rs.restart_policy not in RestartPolicy.ALL:
            errors.append(f"{p}.restartPolicy: invalid value {rs.restart_policy!r}")
        if rs.restart_scope and rs.restart_scope not in RestartScope.VALUES:
            errors.append(f"{p}.restartScope: invalid value {rs.restart_scope!r}")
        for fname, val in (("failPolicy", rs.fail_policy),
                           ("completePolicy", rs.complete_policy)):
            if val and val not in EndingPolicy.VALUES:
                errors.append(f"{p}.{fname}: invalid value {val!r}")
        if rs.edl_policy and rs.edl_policy not in EdlPolicy.VALUES:
            errors.append(f"{p}.edlPolicy: invalid value {rs.edl_policy!r}")

        # Elastic range consistency (real semantics for reference's unused
        # fields, replica.go:10-11).
        lo, hi, n = rs.min_replicas, rs.max_replicas, rs.replicas
        if lo is not None and lo < 0:
            errors.append(f"{p}.minReplicas: must be >= 0")
        if lo is not None and hi is not None and lo > hi:
            errors.append(f"{p}: minReplicas ({lo}) > maxReplicas ({hi})")
        if n is not None and lo is not None and n < lo:
            errors.append(f"{p}: replicas ({n}) < minReplicas ({lo})")
        if n is not None and hi is not None and n > hi:
            errors.append(f"{p}: replicas ({n}) > maxReplicas ({hi})")

        # Pod template sanity: at least one aitj-* container participates
        # (reference treats non-prefixed containers as opaque,
        # constants.go:41-44; a job with none would never reach Succeed).
        containers = (rs.template.get("spec") or {}).get("containers") or []
        if rs.template and not any(
            (c.get("name") or "").startswith(CONTAINER_PREFIX) for c in containers
        ):
            errors.append(
                f"{p}.template: no container named '{CONTAINER_PREFIX}*' — "
                "only aitj-* containers participate in status tracking"
            )

    errors.extend(validate_hbm_sizing(job))
    return errors


def validate_or_raise(job: AITrainingJob) -> None:
    errs = validate(job)
    if errs:
        raise ValidationError(errs)


def validate_hbm_sizing(job: AITrainingJob) -> List[str]:
    """HBM-aware admission (SURVEY.md §2.3 'no GPU awareness'): when the job
    declares its model size (annotations ``elasticdeeplearning.ai/model`` or
    ``.../model-params``) and a replica requests amd.com/gpu, reject specs
    whose per-GPU DP training state cannot fit MI355X's 288 GB."""
    from . import sizing

    errors: List[str] = []
    n_params = sizing.declared_params(job)
    if n_params is None:
        return errors
    # a pp x tp sharded model (model-shards annotation) spreads its state
    # over N devices: admission checks the per-shard slice
    shards = sizing.declared_shards(job)
    est = sizing.estimate_training_bytes(n_params // shards)
    for rtype, rs in job.spec.replica_specs.items():
        gpus = gpus_requested(rs)
        if gpus <= 0:
            continue
        if shards > 1 and (rs.replicas or 0) * gpus < shards:
            errors.append(
                f"spec.replicaSpecs[{rtype}]: model-shards={shards} but "
                f"the job only provides {(rs.replicas or 0) * gpus} GPUs")
            continue
        if not sizing.fits_per_gpu(est, gpus):
            need = sizing.min_gpus_for(est)
            errors.append(
                f"spec.replicaSpecs[{rtype}]: model of {n_params:,} params "
                f"(/{shards} shards) needs ~{est.total_gb:.0f} GB of HBM "
                f"per replica but the pod requests {gpus} x 288 GB GPU(s); "
                f"request at least {need} amd.com/gpu (or shard further)")
    return errors


def gpus_requested(replica_spec) -> int:
    """amd.com/gpu count requested by one pod of this replica spec."""
    total = 0
    for c in (replica_spec.template.get("spec") or {}).get("containers") or []:
        limits = (c.get("resources") or {}).get("limits") or {}
        requests = (c.get("resources") or {}).get("requests") or {}
        total += int(limits.get(AMD_GPU_RESOURCE, requests.get(AMD_GPU_RESOURCE, 0)))
    return total


def hbm_fit_check(model_bytes_per_gpu: int, gpus: int) -> bool:
    """HBM-aware sizing: does the per-GPU shard fit in MI355X's 288 GB HBM3E?

    Used by the admission path to reject jobs whose declared model footprint
    cannot fit the requested GPU count (SURVEY.md §2.3 'no GPU awareness').
    """
    return gpus > 0 and model_bytes_per_gpu <= MI355X_HBM_BYTES
