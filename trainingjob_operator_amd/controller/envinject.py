"""Worker env/DNS contract rendering.

Reproduces the reference contract exactly (reference: pkg/controller/
pod.go:548-652, service.go:19-52):
  {RT}_INSTANCES / _INSTANCES_NUM / _PORTS / _PORTS_NUM / _HOSTS / _HOSTS_NUM
  TRAININGJOB_REPLICA_NAME / _REPLICA_INDEX / _REPLICA_RESTARTCOUNT /
  TRAININGJOB_SERVICE / _NAME / _NAMESPACE, and per-container
  TRAININGJOB_PORTS — only containers/ports named "aitj-*" participate.

MI355X-native extension (additive): MASTER_ADDR/MASTER_PORT/
WORLD_SIZE/RANK/LOCAL_RANK per replica role (world = that role's replicas,
master = index 0 of the role) so stock torch.distributed workers bootstrap
RCCL with zero launcher glue, plus the elastic rendezvous epoch and
min/max replicas.
"""
from __future__ import annotations

from typing import Dict, List

from ..api import constants as C
from ..api.types import AITrainingJob, ReplicaSpec, gen_general_name


def ports_from_spec(spec: ReplicaSpec) -> List[int]:
    """Ports that participate: containers named aitj-* -> ports named aitj-*
    (reference: service.go:19-43). Sorted for determinism (the reference
    iterates a Go map)."""
    ports = []
    for c in (spec.template.get("spec") or {}).get("containers") or []:
        if not (c.get("name") or "").startswith(C.CONTAINER_PREFIX):
            continue
        for p in c.get("ports") or []:
            if (p.get("name") or "").startswith(C.PORT_PREFIX):
                cp = p.get("containerPort")
                if cp is not None:
                    ports.append(int(cp))
    return sorted(set(ports))


def gpus_per_pod(spec: ReplicaSpec) -> int:
    """GPUs one pod of this replica requests (amd.com/gpu limits summed over
    aitj-* containers; ROCm device-plugin contract). 0 -> CPU-only pod."""
    total = 0
    for c in (spec.template.get("spec") or {}).get("containers") or []:
        if not (c.get("name") or "").startswith(C.CONTAINER_PREFIX):
            continue
        res = c.get("resources") or {}
        for kind in ("limits", "requests"):
            v = (res.get(kind) or {}).get(C.GPU_RESOURCE)
            if v is not None:
                try:
                    total += int(v)
                except (TypeError, ValueError):
                    pass
                break
    return total


def ports_from_container(container: dict) -> List[int]:
    """A single container's own aitj-* ports (reference: pod.go:644-648)."""
    out = []
    for p in container.get("ports") or []:
        if (p.get("name") or "").startswith(C.PORT_PREFIX):
            cp = p.get("containerPort")
            if cp is not None:
                out.append(int(cp))
    return sorted(set(out))


def render_env(job: AITrainingJob, rtype: str, index: int,
               restart_count: int, epoch: int = 0) -> List[Dict[str, str]]:
    """The env list appended to every init- and main container."""
    env: List[Dict[str, str]] = []
    ns = job.namespace
    for rt in sorted(job.spec.replica_specs):
        spec = job.spec.replica_specs[rt]
        replicas = spec.replicas or 0
        ports = ports_from_spec(spec)
        instances = [f"{gen_general_name(job.name, rt, i)}.{ns}"
                     for i in range(replicas)]
        hosts = [f"{inst}:{port}" for inst in instances for port in ports]
        up = rt.upper()
        env += [
            {"name": f"{up}_INSTANCES", "value": ",".join(instances)},
            {"name": f"{up}_INSTANCES_NUM", "value": str(len(instances))},
            {"name": f"{up}_PORTS", "value": ",".join(str(p) for p in ports)},
            {"name": f"{up}_PORTS_NUM", "value": str(len(ports))},
            {"name": f"{up}_HOSTS", "value": ",".join(hosts)},
            {"name": f"{up}_HOSTS_NUM", "value": str(len(hosts))},
        ]
    env += [
        {"name": C.ENV_REPLICA_NAME, "value": rtype},
        {"name": C.ENV_REPLICA_INDEX, "value": str(index)},
        {"name": C.ENV_REPLICA_RESTART_COUNT, "value": str(restart_count)},
        {"name": C.ENV_SERVICE,
         "value": f"{gen_general_name(job.name, rtype, index)}.{ns}"},
        {"name": C.ENV_JOB_NAME, "value": job.name},
        {"name": C.ENV_JOB_NAMESPACE, "value": ns},
    ]

    # --- MI355X RCCL rendezvous extension (role-local world) ---
    # Multi-GPU pods (G = amd.com/gpu per pod > 1) host G ranks launched
    # in-pod by torchrun: the operator injects the NODE-level contract
    # (NODE_RANK / NPROC_PER_NODE / base RANK = index*G, WORLD_SIZE =
    # replicas*G) and leaves per-process RANK/LOCAL_RANK assignment to the
    # in-pod launcher. Single-GPU (or CPU) pods keep the flat per-pod
    # contract with LOCAL_RANK=0 (one process per pod).
    spec = job.spec.replica_specs[rtype]
    ports = ports_from_spec(spec)
    master_port = ports[0] if ports else C.DEFAULT_MASTER_PORT
    gpus = gpus_per_pod(spec)
    per_pod = max(gpus, 1)
    env += [
        {"name": C.ENV_MASTER_ADDR,
         "value": f"{gen_general_name(job.name, rtype, 0)}.{ns}"},
        {"name": C.ENV_MASTER_PORT, "value": str(master_port)},
        {"name": C.ENV_WORLD_SIZE,
         "value": str((spec.replicas or 0) * per_pod)},
        {"name": C.ENV_RANK, "value": str(index * per_pod)},
        {"name": C.ENV_NODE_RANK, "value": str(index)},
        {"name": C.ENV_NPROC_PER_NODE, "value": str(per_pod)},
        {"name": C.ENV_LOCAL_WORLD_SIZE, "value": str(per_pod)},
        {"name": C.ENV_REND_EPOCH, "value": str(epoch)},
    ]
    if per_pod == 1:
        env.append({"name": C.ENV_LOCAL_RANK, "value": "0"})
    if spec.min_replicas is not None:
        env.append({"name": C.ENV_MIN_REPLICAS,
                    "value": str(spec.min_replicas)})
    if spec.max_replicas is not None:
        env.append({"name": C.ENV_MAX_REPLICAS,
                    "value": str(spec.max_replicas)})
    return env


def inject_env(pod_template: dict, job: AITrainingJob, rtype: str, index: int,
               restart_count: int, epoch: int = 0) -> None:
    """Append the contract env to every init/main container in-place
    (reference: pod.go:632-651); main containers additionally get their own
    TRAININGJOB_PORTS."""
    env = render_env(job, rtype, index, restart_count, epoch)
    pspec = pod_template.setdefault("spec", {})
    for c in pspec.get("initContainers") or []:
        c.setdefault("env", []).extend(env)
    for c in pspec.get("containers") or []:
        c.setdefault("env", []).extend(env)
        c["env"].append({
            "name": C.ENV_PORTS,
            "value": ",".join(str(p) for p in ports_from_container(c)),
        })
