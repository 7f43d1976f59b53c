"""In-memory fake Kubernetes API for tests.

The moral equivalent of the reference's generated-but-unused fake clientset
(reference: pkg/client/clientset/versioned/fake/) plus the
cluster-simulation helpers the reference never had: drive pod phases,
container statuses, node readiness, and scheduling from test code, and
assert on the recorded action log.
"""
from __future__ import annotations

import copy
import itertools
import queue
import threading
from typing import Dict, Iterator, List, Optional, Tuple

from ..api import constants as C
from ..utils.k8stime import format_time
from .client import ApiError, KubeApi
from .objects import matches_selector, meta


class FakeKubeApi(KubeApi):
    def __init__(self):
        self._lock = threading.RLock()
        self.pods: Dict[Tuple[str, str], dict] = {}
        self.services: Dict[Tuple[str, str], dict] = {}
        self.nodes: Dict[str, dict] = {}
        self.jobs: Dict[Tuple[str, str], dict] = {}
        self.leases: Dict[Tuple[str, str], dict] = {}
        self.events: List[dict] = []
        self.pod_logs: dict = {}
        self.crds: Dict[str, dict] = {}
        self.actions: List[tuple] = []
        self._uid = itertools.count(1)
        self._rv = itertools.count(1)
        self._watchers: List[queue.Queue] = []
        # simulation knobs
        self.auto_schedule = True   # bind new pods to a node immediately
        self.default_node = "node-0"
        self.add_node("node-0", ready=True)

    # -- internals --------------------------------------------------------
    def _record(self, *action):
        self.actions.append(action)

    def _stamp(self, obj: dict, namespace: str, kind: str):
        m = meta(obj)
        m["namespace"] = namespace
        m.setdefault("uid", f"uid-{next(self._uid)}")
        m["resourceVersion"] = str(next(self._rv))
        m.setdefault("creationTimestamp", format_time())
        obj["kind"] = kind

    def _emit(self, evt_type: str, kind: str, obj: dict):
        for q in list(self._watchers):
            q.put((evt_type, kind, copy.deepcopy(obj)))

    # -- pods -------------------------------------------------------------
    def create_pod(self, namespace, pod):
        with self._lock:
            pod = copy.deepcopy(pod)
            name = meta(pod)["name"]
            key = (namespace, name)
            if key in self.pods:
                raise ApiError(409, f"pod {name} exists")
            self._stamp(pod, namespace, "Pod")
            pod.setdefault("status", {})["phase"] = "Pending"
            if self.auto_schedule and self.default_node:
                pod["spec"]["nodeName"] = self.default_node
                pod["status"]["startTime"] = format_time()
            self.pods[key] = pod
            self._record("create", "pod", namespace, name)
            self._emit("ADDED", "pod", pod)
            return copy.deepcopy(pod)

    def get_pod(self, namespace, name):
        with self._lock:
            key = (namespace, name)
            if key not in self.pods:
                raise ApiError(404, f"pod {name}")
            return copy.deepcopy(self.pods[key])

    def list_pods(self, namespace=None, selector=None):
        with self._lock:
            out = []
            for (ns, _), pod in self.pods.items():
                if namespace and ns != namespace:
                    continue
                if selector and not matches_selector(pod, selector):
                    continue
                out.append(copy.deepcopy(pod))
            return out

    def patch_pod_metadata(self, namespace, name, metadata_patch):
        # merge-patch on metadata only (adoption patches ownerReferences)
        with self._lock:
            key = (namespace, name)
            if key not in self.pods:
                raise ApiError(404, f"pod {name}")
            pod = self.pods[key]
            for k2, v2 in metadata_patch.items():
                if isinstance(v2, dict):
                    pod["metadata"].setdefault(k2, {}).update(v2)
                else:
                    pod["metadata"][k2] = copy.deepcopy(v2)
            self._record("patch", "pod", namespace, name)
            self._emit("MODIFIED", "pod", pod)
            return copy.deepcopy(pod)

    def delete_pod(self, namespace, name, grace_period=None):
        with self._lock:
            key = (namespace, name)
            if key not in self.pods:
                raise ApiError(404, f"pod {name}")
            pod = self.pods.pop(key)
            self._record("delete", "pod", namespace, name, grace_period)
            self._emit("DELETED", "pod", pod)

    # -- services ---------------------------------------------------------
    def create_service(self, namespace, svc):
        with self._lock:
            svc = copy.deepcopy(svc)
            name = meta(svc)["name"]
            key = (namespace, name)
            if key in self.services:
                raise ApiError(409, f"service {name} exists")
            self._stamp(svc, namespace, "Service")
            self.services[key] = svc
            self._record("create", "service", namespace, name)
            self._emit("ADDED", "service", svc)
            return copy.deepcopy(svc)

    def list_services(self, namespace=None, selector=None):
        with self._lock:
            out = []
            for (ns, _), svc in self.services.items():
                if namespace and ns != namespace:
                    continue
                if selector and not matches_selector(svc, selector):
                    continue
                out.append(copy.deepcopy(svc))
            return out

    def delete_service(self, namespace, name):
        with self._lock:
            key = (namespace, name)
            if key not in self.services:
                raise ApiError(404, f"service {name}")
            svc = self.services.pop(key)
            self._record("delete", "service", namespace, name)
            self._emit("DELETED", "service", svc)

    # -- nodes ------------------------------------------------------------
    def list_nodes(self):
        with self._lock:
            return [copy.deepcopy(n) for n in self.nodes.values()]

    def get_node(self, name):
        with self._lock:
            if name not in self.nodes:
                raise ApiError(404, f"node {name}")
            return copy.deepcopy(self.nodes[name])

    def patch_node_status(self, name, status_patch):
        with self._lock:
            if name not in self.nodes:
                raise ApiError(404, f"node {name}")
            node = self.nodes[name]
            status = node.setdefault("status", {})
            for key, val in status_patch.items():
                if key == "conditions":
                    conds = {c["type"]: c
                             for c in status.get("conditions") or []}
                    for c in val:
                        conds[c["type"]] = c
                    status["conditions"] = list(conds.values())
                else:
                    status[key] = copy.deepcopy(val)
            self._record("patch", "node-status", name)

    def annotate_node(self, name, annotations):
        with self._lock:
            if name not in self.nodes:
                raise ApiError(404, f"node {name}")
            anns = self.nodes[name].setdefault("metadata", {}) \
                .setdefault("annotations", {})
            anns.update(annotations)

    # -- jobs -------------------------------------------------------------
    def ensure_crd(self, crd_manifest):
        with self._lock:
            self.crds[meta(crd_manifest).get("name", "crd")] = crd_manifest
            self._record("ensure", "crd", meta(crd_manifest).get("name"))

    def create_job(self, namespace, job_dict):
        with self._lock:
            job_dict = copy.deepcopy(job_dict)
            name = meta(job_dict)["name"]
            key = (namespace, name)
            if key in self.jobs:
                raise ApiError(409, f"job {name} exists")
            self._stamp(job_dict, namespace, C.CRD_KIND)
            self.jobs[key] = job_dict
            self._emit("ADDED", "job", job_dict)
            return copy.deepcopy(job_dict)

    def get_job(self, namespace, name):
        with self._lock:
            key = (namespace, name)
            if key not in self.jobs:
                raise ApiError(404, f"job {name}")
            return copy.deepcopy(self.jobs[key])

    def list_jobs(self, namespace=None):
        with self._lock:
            return [copy.deepcopy(j) for (ns, _), j in self.jobs.items()
                    if namespace is None or ns == namespace]

    def update_job(self, namespace, name, job):
        # real API servers IGNORE .status on a main-resource PUT when the
        # CRD declares a status subresource (ours does) -- emulate that so
        # tests catch controllers writing status through the wrong door
        with self._lock:
            key = (namespace, name)
            if key not in self.jobs:
                raise ApiError(404, f"job {name}")
            cur = self.jobs[key]
            new_rv = job.get("metadata", {}).get("resourceVersion")
            if new_rv and new_rv != cur["metadata"]["resourceVersion"]:
                raise ApiError(409, "resourceVersion conflict")
            job = copy.deepcopy(job)
            if "status" in cur:
                job["status"] = copy.deepcopy(cur["status"])
            else:
                job.pop("status", None)
            meta(job)["resourceVersion"] = str(next(self._rv))
            meta(job)["namespace"] = namespace
            meta(job).setdefault("uid", cur["metadata"]["uid"])
            self.jobs[key] = job
            self._record("update", "job", namespace, name)
            self._emit("MODIFIED", "job", job)
            return copy.deepcopy(job)

    def update_job_status(self, namespace, name, job):
        # /status subresource: applies ONLY .status (spec/metadata of the
        # stored object are preserved), like a real API server
        with self._lock:
            key = (namespace, name)
            if key not in self.jobs:
                raise ApiError(404, f"job {name}")
            cur = self.jobs[key]
            new_rv = job.get("metadata", {}).get("resourceVersion")
            if new_rv and new_rv != cur["metadata"]["resourceVersion"]:
                raise ApiError(409, "resourceVersion conflict")
            new = copy.deepcopy(cur)
            new["status"] = copy.deepcopy(job.get("status") or {})
            meta(new)["resourceVersion"] = str(next(self._rv))
            self.jobs[key] = new
            self._record("update_status", "job", namespace, name)
            self._emit("MODIFIED", "job", new)
            return copy.deepcopy(new)

    def delete_job(self, namespace, name):
        with self._lock:
            key = (namespace, name)
            if key not in self.jobs:
                raise ApiError(404, f"job {name}")
            j = self.jobs.pop(key)
            self._record("delete", "job", namespace, name)
            self._emit("DELETED", "job", j)

    # -- events / leases --------------------------------------------------
    def list_events(self, namespace, involved_name=None):
        with self._lock:
            out = []
            for e in self.events:
                if namespace and \
                        e.get("metadata", {}).get("namespace") != namespace:
                    continue
                if involved_name and e.get("involvedObject", {}) \
                        .get("name") != involved_name:
                    continue
                out.append(copy.deepcopy(e))
            return out

    def create_event(self, namespace, event):
        with self._lock:
            self.events.append(event)

    def get_lease(self, namespace, name):
        with self._lock:
            key = (namespace, name)
            if key not in self.leases:
                raise ApiError(404, f"lease {name}")
            return copy.deepcopy(self.leases[key])

    def create_lease(self, namespace, lease):
        with self._lock:
            key = (namespace, meta(lease)["name"])
            if key in self.leases:
                raise ApiError(409, "lease exists")
            self._stamp(lease, namespace, "Lease")
            self.leases[key] = copy.deepcopy(lease)
            return copy.deepcopy(lease)

    def update_lease(self, namespace, name, lease):
        with self._lock:
            key = (namespace, name)
            if key not in self.leases:
                raise ApiError(404, f"lease {name}")
            self.leases[key] = copy.deepcopy(lease)
            return copy.deepcopy(lease)

    # -- watches ----------------------------------------------------------
    def _watch_kind(self, kind: str, stop: threading.Event) -> Iterator[tuple]:
        q: queue.Queue = queue.Queue()
        self._watchers.append(q)
        try:
            while not stop.is_set():
                try:
                    evt_type, k, obj = q.get(timeout=0.2)
                except queue.Empty:
                    continue
                if k == kind:
                    yield evt_type, obj
        finally:
            self._watchers.remove(q)

    def watch_pods(self, namespace, stop):
        return self._watch_kind("pod", stop)

    def watch_services(self, namespace, stop):
        return self._watch_kind("service", stop)

    def watch_jobs(self, namespace, stop):
        return self._watch_kind("job", stop)

    # ======================================================================
    # Simulation helpers (test-side cluster behavior)
    # ======================================================================

    def add_node(self, name: str, ready: bool = True):
        with self._lock:
            self.nodes[name] = {
                "metadata": {"name": name},
                "status": {"conditions": [
                    {"type": "Ready",
                     "status": "True" if ready else "False"}]},
            }

    def set_node_ready(self, name: str, ready: bool):
        self.add_node(name, ready)

    def remove_node(self, name: str):
        with self._lock:
            self.nodes.pop(name, None)

    def _pod(self, namespace, name):
        key = (namespace, name)
        if key not in self.pods:
            raise KeyError(f"no pod {key}")
        return self.pods[key]

    def set_pod_phase(self, namespace: str, name: str, phase: str,
                      exit_code: Optional[int] = None,
                      waiting_reason: Optional[str] = None):
        """Drive a pod through its lifecycle, synthesizing container
        statuses for the aitj-* containers."""
        with self._lock:
            pod = self._pod(namespace, name)
            status = pod.setdefault("status", {})
            status["phase"] = phase
            cstatuses = []
            for c in (pod.get("spec") or {}).get("containers") or []:
                cname = c.get("name", "")
                if phase == "Running":
                    state = {"running": {"startedAt": format_time()}}
                elif phase == "Succeeded":
                    state = {"terminated": {"exitCode": 0}}
                elif phase == "Failed":
                    state = {"terminated": {
                        "exitCode": exit_code if exit_code is not None else 1,
                        "reason": "Error"}}
                elif waiting_reason:
                    state = {"waiting": {"reason": waiting_reason}}
                else:
                    state = {"waiting": {"reason": "ContainerCreating"}}
                cstatuses.append({"name": cname, "state": state})
            status["containerStatuses"] = cstatuses
            status.setdefault("startTime", format_time())
            pod["metadata"]["resourceVersion"] = str(next(self._rv))
            self._emit("MODIFIED", "pod", pod)

    def set_all_pods_phase(self, namespace: str, phase: str, **kw):
        for (ns, name) in list(self.pods):
            if ns == namespace:
                self.set_pod_phase(ns, name, phase, **kw)

    def set_pod_log(self, namespace: str, name: str, text: str):
        with self._lock:
            self._pod(namespace, name)  # existence check
            self.pod_logs[(namespace, name)] = text

    def read_pod_log(self, namespace, name, tail_lines=None):
        with self._lock:
            self._pod(namespace, name)
            text = self.pod_logs.get((namespace, name), "")
        if tail_lines is not None:
            return "\n".join(text.splitlines()[-tail_lines:])
        return text

    def set_pod_unschedulable(self, namespace: str, name: str,
                              message: str = "0/8 nodes are available",
                              since: Optional[float] = None):
        """Scheduler-style Unschedulable: Pending, unbound, with a
        PodScheduled=False condition stamped at `since` (default now)."""
        with self._lock:
            pod = self._pod(namespace, name)
            pod["spec"].pop("nodeName", None)
            status = pod.setdefault("status", {})
            status["phase"] = "Pending"
            conds = [c for c in status.get("conditions") or []
                     if c.get("type") != "PodScheduled"]
            conds.append({"type": "PodScheduled", "status": "False",
                          "reason": "Unschedulable", "message": message,
                          "lastTransitionTime": format_time(since)})
            status["conditions"] = conds
            pod["metadata"]["resourceVersion"] = str(next(self._rv))
            self._emit("MODIFIED", "pod", pod)

    def bind_pod(self, namespace: str, name: str, node: str):
        with self._lock:
            pod = self._pod(namespace, name)
            pod["spec"]["nodeName"] = node
            self._emit("MODIFIED", "pod", pod)

    def pod_names(self, namespace: Optional[str] = None) -> List[str]:
        with self._lock:
            return sorted(n for (ns, n) in self.pods
                          if namespace is None or ns == namespace)

    def service_names(self, namespace: Optional[str] = None) -> List[str]:
        with self._lock:
            return sorted(n for (ns, n) in self.services
                          if namespace is None or ns == namespace)
