"""Helpers over dict-shaped Kubernetes objects (corev1 wire format).

Pods/Services/Nodes stay as plain dicts end-to-end (what the REST API
speaks); these helpers centralize the accessor patterns the controller
needs.
"""
from __future__ import annotations

from typing import Dict, List, Optional

from ..api import constants as C


def meta(obj: dict) -> dict:
    return obj.setdefault("metadata", {})


def name_of(obj: dict) -> str:
    return meta(obj).get("name", "")


def namespace_of(obj: dict) -> str:
    return meta(obj).get("namespace", "default")


def labels_of(obj: dict) -> Dict[str, str]:
    return meta(obj).setdefault("labels", {})


def owner_refs(obj: dict) -> List[dict]:
    return meta(obj).get("ownerReferences") or []


def controller_ref(obj: dict) -> Optional[dict]:
    for ref in owner_refs(obj):
        if ref.get("controller"):
            return ref
    return None


def matches_selector(obj: dict, selector: Dict[str, str]) -> bool:
    lbls = labels_of(obj)
    return all(lbls.get(k) == v for k, v in selector.items())


def pod_phase(pod: dict) -> str:
    return (pod.get("status") or {}).get("phase", "Pending")


def pod_node(pod: dict) -> str:
    return (pod.get("spec") or {}).get("nodeName", "") or ""


def is_deleting(obj: dict) -> bool:
    return meta(obj).get("deletionTimestamp") is not None


def gen_owner_reference(job) -> dict:
    """reference: pkg/controller/controller.go:160-173."""
    return {
        "apiVersion": C.API_VERSION,
        "kind": C.CRD_KIND,
        "name": job.name,
        "uid": job.uid,
        "controller": True,
        "blockOwnerDeletion": True,
    }


def gen_labels(job_name: str) -> Dict[str, str]:
    """reference: pkg/controller/controller.go:175-180."""
    return {
        C.LABEL_GROUP_NAME: C.CRD_GROUP,
        C.LABEL_JOB_NAME: job_name.replace("/", "-"),
    }


def job_selector(job_name: str) -> Dict[str, str]:
    """Label selector for everything owned by a job
    (reference: controller.go:318-323)."""
    return gen_labels(job_name)


def selector_string(selector: Dict[str, str]) -> str:
    return ",".join(f"{k}={v}" for k, v in sorted(selector.items()))
