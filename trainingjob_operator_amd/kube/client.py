"""Kubernetes API access layer.

``KubeApi`` is the minimal typed surface the controller needs (the moral
equivalent of the reference's generated clientset + kubeflow-common control
objects). Two implementations:

  * ``RealKubeApi`` — direct REST against an API server (requests; the
    ``kubernetes`` Python package is not available in this environment, and
    this subset is small enough that a hand-rolled client is simpler and
    dependency-free).
  * ``FakeKubeApi`` (fake.py) — in-memory cluster for tests.

Watch support streams JSON lines from ``?watch=true`` endpoints.
"""
from __future__ import annotations

import json
import os
import threading
from typing import Any, Dict, Iterator, List, Optional

from ..api import constants as C


class ApiError(Exception):
    def __init__(self, status: int, reason: str = ""):
        self.status = status
        self.reason = reason
        super().__init__(f"kube api error {status}: {reason}")

    @property
    def not_found(self) -> bool:
        return self.status == 404

    @property
    def conflict(self) -> bool:
        return self.status == 409

    @property
    def already_exists(self) -> bool:
        return self.status == 409


class KubeApi:
    """Interface; all objects are dicts in wire shape."""

    # pods ---------------------------------------------------------------
    def create_pod(self, namespace: str, pod: dict) -> dict: ...
    def get_pod(self, namespace: str, name: str) -> dict: ...
    def list_pods(self, namespace: Optional[str] = None,
                  selector: Optional[Dict[str, str]] = None) -> List[dict]: ...
    def delete_pod(self, namespace: str, name: str,
                   grace_period: Optional[int] = None) -> None: ...
    def patch_pod_metadata(self, namespace: str, name: str,
                           metadata_patch: dict) -> dict: ...
    def read_pod_log(self, namespace: str, name: str,
                     tail_lines: Optional[int] = None) -> str: ...

    # services -----------------------------------------------------------
    def create_service(self, namespace: str, svc: dict) -> dict: ...
    def list_services(self, namespace: Optional[str] = None,
                      selector: Optional[Dict[str, str]] = None) -> List[dict]: ...
    def delete_service(self, namespace: str, name: str) -> None: ...

    # nodes --------------------------------------------------------------
    def list_nodes(self) -> List[dict]: ...
    def get_node(self, name: str) -> dict: ...
    def patch_node_status(self, name: str, status_patch: dict) -> None: ...
    def annotate_node(self, name: str, annotations: Dict[str, str]) -> None: ...

    # custom resources (aitrainingjobs) ----------------------------------
    def ensure_crd(self, crd_manifest: dict) -> None: ...
    def get_job(self, namespace: str, name: str) -> dict: ...
    def list_jobs(self, namespace: Optional[str] = None) -> List[dict]: ...
    def update_job(self, namespace: str, name: str, job: dict) -> dict: ...
    def update_job_status(self, namespace: str, name: str, job: dict) -> dict: ...
    def delete_job(self, namespace: str, name: str) -> None: ...

    # events / leases ----------------------------------------------------
    def create_event(self, namespace: str, event: dict) -> None: ...
    def list_events(self, namespace: str,
                    involved_name: Optional[str] = None) -> List[dict]: ...
    def get_lease(self, namespace: str, name: str) -> dict: ...
    def create_lease(self, namespace: str, lease: dict) -> dict: ...
    def update_lease(self, namespace: str, name: str, lease: dict) -> dict: ...

    # watches (yield (event_type, object) tuples) ------------------------
    def watch_pods(self, namespace: Optional[str],
                   stop: threading.Event) -> Iterator[tuple]: ...
    def watch_services(self, namespace: Optional[str],
                       stop: threading.Event) -> Iterator[tuple]: ...
    def watch_jobs(self, namespace: Optional[str],
                   stop: threading.Event) -> Iterator[tuple]: ...


# ---------------------------------------------------------------------------
# Real REST client
# ---------------------------------------------------------------------------

_SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


class RealKubeApi(KubeApi):
    def __init__(self, base_url: Optional[str] = None,
                 token: Optional[str] = None,
                 ca_cert: Optional[str] = None,
                 verify: bool = True):
        import requests  # lazy; offline wheelhouse provides it
        self._requests = requests
        if base_url is None:
            host = os.environ.get("KUBERNETES_SERVICE_HOST")
            port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
            if host:
                base_url = f"https://{host}:{port}"
                token_file = os.path.join(_SA_DIR, "token")
                if token is None and os.path.exists(token_file):
                    token = open(token_file).read().strip()
                ca_file = os.path.join(_SA_DIR, "ca.crt")
                if ca_cert is None and os.path.exists(ca_file):
                    ca_cert = ca_file
            else:
                base_url = "http://127.0.0.1:8001"  # kubectl proxy
        self.base_url = base_url.rstrip("/")
        self.session = requests.Session()
        if token:
            self.session.headers["Authorization"] = f"Bearer {token}"
        self.session.verify = ca_cert if ca_cert else verify

    # -- plumbing ---------------------------------------------------------
    def _req(self, method: str, path: str, body: Optional[dict] = None,
             params: Optional[dict] = None, timeout: float = 30.0) -> Any:
        url = self.base_url + path
        headers = {}
        if method == "PATCH":
            headers["Content-Type"] = "application/merge-patch+json"
        r = self.session.request(method, url, json=body, params=params,
                                 timeout=timeout, headers=headers)
        if r.status_code >= 400:
            raise ApiError(r.status_code, r.text[:500])
        if r.text:
            return r.json()
        return None

    @staticmethod
    def _sel(params: dict, selector: Optional[Dict[str, str]]):
        if selector:
            params["labelSelector"] = ",".join(
                f"{k}={v}" for k, v in sorted(selector.items()))

    def _ns_path(self, kind: str, namespace: Optional[str]) -> str:
        if namespace:
            return f"/api/v1/namespaces/{namespace}/{kind}"
        return f"/api/v1/{kind}"

    def _job_path(self, namespace: Optional[str]) -> str:
        base = f"/apis/{C.CRD_GROUP}/{C.CRD_VERSION}"
        if namespace:
            return f"{base}/namespaces/{namespace}/{C.CRD_PLURAL}"
        return f"{base}/{C.CRD_PLURAL}"

    # -- pods -------------------------------------------------------------
    def create_pod(self, namespace, pod):
        return self._req("POST", self._ns_path("pods", namespace), pod)

    def get_pod(self, namespace, name):
        return self._req("GET", self._ns_path("pods", namespace) + "/" + name)

    def list_pods(self, namespace=None, selector=None):
        params: dict = {}
        self._sel(params, selector)
        return self._req("GET", self._ns_path("pods", namespace),
                         params=params)["items"]

    def delete_pod(self, namespace, name, grace_period=None):
        body = {}
        if grace_period is not None:
            body["gracePeriodSeconds"] = grace_period
        self._req("DELETE", self._ns_path("pods", namespace) + "/" + name,
                  body or None)

    def patch_pod_metadata(self, namespace, name, metadata_patch):
        url = (self.base_url + self._ns_path("pods", namespace)
               + "/" + name)
        r = self.session.patch(
            url, json={"metadata": metadata_patch},
            headers={"Content-Type": "application/merge-patch+json"},
            timeout=30.0)
        if r.status_code >= 400:
            raise ApiError(r.status_code, r.text[:500])
        return r.json()

    def read_pod_log(self, namespace, name, tail_lines=None):
        params = {}
        if tail_lines is not None:
            params["tailLines"] = str(tail_lines)
        url = (self.base_url + self._ns_path("pods", namespace)
               + f"/{name}/log")
        r = self.session.get(url, params=params, timeout=30.0)
        if r.status_code >= 400:
            raise ApiError(r.status_code, r.text[:500])
        return r.text

    # -- services ---------------------------------------------------------
    def create_service(self, namespace, svc):
        return self._req("POST", self._ns_path("services", namespace), svc)

    def list_services(self, namespace=None, selector=None):
        params: dict = {}
        self._sel(params, selector)
        return self._req("GET", self._ns_path("services", namespace),
                         params=params)["items"]

    def delete_service(self, namespace, name):
        self._req("DELETE", self._ns_path("services", namespace) + "/" + name)

    # -- nodes ------------------------------------------------------------
    def list_nodes(self):
        return self._req("GET", "/api/v1/nodes")["items"]

    def get_node(self, name):
        return self._req("GET", f"/api/v1/nodes/{name}")

    def patch_node_status(self, name, status_patch):
        self._req("PATCH", f"/api/v1/nodes/{name}/status",
                  {"status": status_patch})

    def annotate_node(self, name, annotations):
        self._req("PATCH", f"/api/v1/nodes/{name}",
                  {"metadata": {"annotations": annotations}})

    # -- custom resources -------------------------------------------------
    def ensure_crd(self, crd_manifest):
        path = "/apis/apiextensions.k8s.io/v1/customresourcedefinitions"
        try:
            self._req("POST", path, crd_manifest)
        except ApiError as e:
            if not e.already_exists:
                raise

    def get_job(self, namespace, name):
        return self._req("GET", self._job_path(namespace) + "/" + name)

    def list_jobs(self, namespace=None):
        return self._req("GET", self._job_path(namespace))["items"]

    def update_job(self, namespace, name, job):
        return self._req("PUT", self._job_path(namespace) + "/" + name, job)

    def update_job_status(self, namespace, name, job):
        return self._req("PUT",
                         self._job_path(namespace) + f"/{name}/status", job)

    def delete_job(self, namespace, name):
        self._req("DELETE", self._job_path(namespace) + "/" + name)

    # -- events / leases --------------------------------------------------
    def create_event(self, namespace, event):
        try:
            self._req("POST", self._ns_path("events", namespace), event)
        except ApiError:
            pass  # events are best-effort

    def list_events(self, namespace, involved_name=None):
        params = {}
        if involved_name:
            params["fieldSelector"] = \
                f"involvedObject.name={involved_name}"
        return self._req("GET", self._ns_path("events", namespace),
                         params=params)["items"]

    def _lease_path(self, namespace):
        return (f"/apis/coordination.k8s.io/v1/namespaces/{namespace}/leases")

    def get_lease(self, namespace, name):
        return self._req("GET", self._lease_path(namespace) + "/" + name)

    def create_lease(self, namespace, lease):
        return self._req("POST", self._lease_path(namespace), lease)

    def update_lease(self, namespace, name, lease):
        return self._req("PUT", self._lease_path(namespace) + "/" + name,
                         lease)

    # -- watches ----------------------------------------------------------
    def _watch(self, path: str, stop: threading.Event,
               params: Optional[dict] = None) -> Iterator[tuple]:
        """Streaming watch with client-go re-list semantics:

        * tracks resourceVersion across reconnects;
        * BOOKMARK events only advance the rv (requested via
          allowWatchBookmarks);
        * 410 Gone — an ERROR event with code 410, or an HTTP 410 on
          reconnect — means the rv EXPIRED from etcd: drop the rv (the
          next connect starts from "now") and yield a synthetic
          ("RELIST", {}) so the consumer re-lists; events between expiry
          and reconnect are otherwise silently lost.
        The controller is level-triggered (full LIST per sync + periodic
        resync), so a missed event only delays work; RELIST removes even
        that delay."""
        params = dict(params or {})
        params["watch"] = "true"
        params["allowWatchBookmarks"] = "true"
        while not stop.is_set():
            try:
                r = self.session.get(self.base_url + path, params=params,
                                     stream=True, timeout=(10, 300))
                if r.status_code == 410:
                    params.pop("resourceVersion", None)
                    yield "RELIST", {}
                    continue
                for line in r.iter_lines():
                    if stop.is_set():
                        return
                    if not line:
                        continue
                    evt = json.loads(line)
                    etype = evt.get("type", "")
                    obj = evt.get("object", {})
                    if etype == "ERROR":
                        if obj.get("code") == 410:
                            params.pop("resourceVersion", None)
                            yield "RELIST", {}
                        break   # reconnect either way
                    rv = obj.get("metadata", {}).get("resourceVersion")
                    if rv:
                        params["resourceVersion"] = rv
                    if etype == "BOOKMARK":
                        continue
                    yield etype, obj
            except Exception:
                if stop.is_set():
                    return
                stop.wait(2.0)  # reconnect backoff

    def watch_pods(self, namespace, stop):
        return self._watch(self._ns_path("pods", namespace), stop)

    def watch_services(self, namespace, stop):
        return self._watch(self._ns_path("services", namespace), stop)

    def watch_jobs(self, namespace, stop):
        return self._watch(self._job_path(namespace), stop)
