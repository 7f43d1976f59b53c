"""Minimal Kubernetes-API-shaped HTTP server backed by FakeKubeApi.

Lets tests drive RealKubeApi over a real socket (loopback integration for
the REST client: paths, selectors, status subresource, watch streaming).
"""
from __future__ import annotations

import json
import re
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from urllib.parse import parse_qs, urlparse

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.kube.client import ApiError
from trainingjob_operator_amd.kube.fake import FakeKubeApi

POD_RE = re.compile(r"^/api/v1/namespaces/([^/]+)/pods(?:/([^/]+))?$")
SVC_RE = re.compile(r"^/api/v1/namespaces/([^/]+)/services(?:/([^/]+))?$")
NODE_RE = re.compile(r"^/api/v1/nodes(?:/([^/]+))?(/status)?$")
JOB_RE = re.compile(
    rf"^/apis/{C.CRD_GROUP}/{C.CRD_VERSION}/namespaces/([^/]+)/"
    rf"{C.CRD_PLURAL}(?:/([^/]+))?(/status)?$")
LEASE_RE = re.compile(
    r"^/apis/coordination.k8s.io/v1/namespaces/([^/]+)/leases(?:/([^/]+))?$")
EVENT_RE = re.compile(r"^/api/v1/namespaces/([^/]+)/events$")


def _selector(query) -> dict:
    sel = {}
    for part in (query.get("labelSelector", [""])[0] or "").split(","):
        if "=" in part:
            k, v = part.split("=", 1)
            sel[k] = v
    return sel or None


class _Handler(BaseHTTPRequestHandler):
    fake: FakeKubeApi = None

    def log_message(self, *a):
        pass

    def _send(self, code: int, body=None):
        data = json.dumps(body or {}).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)

    def _body(self):
        n = int(self.headers.get("Content-Length", 0) or 0)
        if not n:
            return {}
        return json.loads(self.rfile.read(n))

    def _watch(self, kind):
        self.send_response(200)
        self.send_header("Content-Type", "application/json")
        self.send_header("Transfer-Encoding", "chunked")
        self.end_headers()
        stop = threading.Event()

        def emit(evt_type, obj):
            line = json.dumps({"type": evt_type, "object": obj}) + "\n"
            data = line.encode()
            self.wfile.write(f"{len(data):x}\r\n".encode() + data
                             + b"\r\n")
            self.wfile.flush()

        # resourceVersion-expiry simulation: one ERROR/410 event, like a
        # real API server whose etcd compacted the requested rv
        if getattr(self.fake, "expire_next_watch", False):
            self.fake.expire_next_watch = False
            try:
                emit("ERROR", {"kind": "Status", "code": 410,
                               "reason": "Expired"})
            except (BrokenPipeError, ConnectionResetError):
                pass
            return
        try:
            for evt_type, obj in self.fake._watch_kind(kind, stop):
                line = json.dumps({"type": evt_type, "object": obj}) + "\n"
                data = line.encode()
                self.wfile.write(f"{len(data):x}\r\n".encode() + data
                                 + b"\r\n")
                self.wfile.flush()
        except (BrokenPipeError, ConnectionResetError):
            pass
        finally:
            stop.set()

    def do_GET(self):
        u = urlparse(self.path)
        q = parse_qs(u.query)
        watching = q.get("watch", ["false"])[0] == "true"
        try:
            if u.path.endswith("/log") and "/pods/" in u.path:
                parts = u.path.split("/")
                ns, name = parts[4], parts[6]
                tail = q.get("tailLines", [None])[0]
                text = self.fake.read_pod_log(
                    ns, name, int(tail) if tail else None)
                self.send_response(200)
                self.send_header("Content-Type", "text/plain")
                self.end_headers()
                self.wfile.write(text.encode())
                return None
            if (m := POD_RE.match(u.path)):
                ns, name = m.groups()
                if watching:
                    return self._watch("pod")
                if name:
                    return self._send(200, self.fake.get_pod(ns, name))
                return self._send(200, {"items": self.fake.list_pods(
                    ns, _selector(q))})
            if (m := SVC_RE.match(u.path)):
                ns, name = m.groups()
                if name:
                    raise ApiError(404, "no single-service GET in fixture")
                return self._send(200, {"items": self.fake.list_services(
                    ns, _selector(q))})
            if (m := NODE_RE.match(u.path)):
                name, _ = m.groups()
                if name:
                    return self._send(200, self.fake.get_node(name))
                return self._send(200, {"items": self.fake.list_nodes()})
            if (m := JOB_RE.match(u.path)):
                ns, name, _ = m.groups()
                if watching:
                    return self._watch("job")
                if name:
                    return self._send(200, self.fake.get_job(ns, name))
                return self._send(200, {"items": self.fake.list_jobs(ns)})
            if "/events" in u.path and u.path.startswith("/api/v1/"):
                ns = u.path.split("/")[4]
                fs = q.get("fieldSelector", [""])[0]
                name = fs.split("=", 1)[1] if "=" in fs else None
                return self._send(200, {"items": self.fake.list_events(
                    ns, involved_name=name)})
            if (m := LEASE_RE.match(u.path)):
                ns, name = m.groups()
                return self._send(200, self.fake.get_lease(ns, name))
            return self._send(404, {"message": f"no route {u.path}"})
        except ApiError as e:
            return self._send(e.status, {"message": e.reason})

    def do_POST(self):
        u = urlparse(self.path)
        body = self._body()
        try:
            if (m := POD_RE.match(u.path)):
                return self._send(201, self.fake.create_pod(m.group(1), body))
            if (m := SVC_RE.match(u.path)):
                return self._send(201,
                                  self.fake.create_service(m.group(1), body))
            if (m := JOB_RE.match(u.path)):
                return self._send(201, self.fake.create_job(m.group(1), body))
            if (m := LEASE_RE.match(u.path)):
                return self._send(201,
                                  self.fake.create_lease(m.group(1), body))
            if (m := EVENT_RE.match(u.path)):
                self.fake.create_event(m.group(1), body)
                return self._send(201, body)
            if u.path == ("/apis/apiextensions.k8s.io/v1/"
                          "customresourcedefinitions"):
                self.fake.ensure_crd(body)
                return self._send(201, body)
            return self._send(404, {"message": f"no route {u.path}"})
        except ApiError as e:
            return self._send(e.status, {"message": e.reason})

    def do_PUT(self):
        u = urlparse(self.path)
        body = self._body()
        try:
            if (m := JOB_RE.match(u.path)):
                ns, name, _status = m.groups()
                if _status:
                    return self._send(
                        200, self.fake.update_job_status(ns, name, body))
                return self._send(200, self.fake.update_job(ns, name, body))
            if "/events" in u.path and u.path.startswith("/api/v1/"):
                ns = u.path.split("/")[4]
                fs = q.get("fieldSelector", [""])[0]
                name = fs.split("=", 1)[1] if "=" in fs else None
                return self._send(200, {"items": self.fake.list_events(
                    ns, involved_name=name)})
            if (m := LEASE_RE.match(u.path)):
                return self._send(200, self.fake.update_lease(
                    m.group(1), m.group(2), body))
            return self._send(404, {"message": f"no route {u.path}"})
        except ApiError as e:
            return self._send(e.status, {"message": e.reason})

    def do_PATCH(self):
        u = urlparse(self.path)
        body = self._body()
        try:
            if (m := NODE_RE.match(u.path)):
                name, status = m.groups()
                if status:
                    self.fake.patch_node_status(name,
                                                body.get("status", {}))
                else:
                    self.fake.annotate_node(
                        name, body.get("metadata", {})
                        .get("annotations", {}))
                return self._send(200, self.fake.get_node(name))
            return self._send(404, {"message": f"no route {u.path}"})
        except ApiError as e:
            return self._send(e.status, {"message": e.reason})

    def do_DELETE(self):
        u = urlparse(self.path)
        try:
            if (m := POD_RE.match(u.path)) and m.group(2):
                body = self._body()
                self.fake.delete_pod(m.group(1), m.group(2),
                                     body.get("gracePeriodSeconds"))
                return self._send(200, {})
            if (m := SVC_RE.match(u.path)) and m.group(2):
                self.fake.delete_service(m.group(1), m.group(2))
                return self._send(200, {})
            if (m := JOB_RE.match(u.path)) and m.group(2):
                self.fake.delete_job(m.group(1), m.group(2))
                return self._send(200, {})
            return self._send(404, {"message": f"no route {u.path}"})
        except ApiError as e:
            return self._send(e.status, {"message": e.reason})


class MockKubeServer:
    def __init__(self, fake: FakeKubeApi):
        handler = type("H", (_Handler,), {"fake": fake})
        self.httpd = ThreadingHTTPServer(("127.0.0.1", 0), handler)
        self.thread = threading.Thread(target=self.httpd.serve_forever,
                                       daemon=True)

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self.httpd.server_port}"

    def start(self):
        self.thread.start()
        return self

    def stop(self):
        self.httpd.shutdown()
