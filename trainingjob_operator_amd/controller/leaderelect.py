"""Lease-based leader election (modern equivalent of the reference's
endpoints lock in kube-system — reference: cmd/app/server.go:85-106; same
crash-on-loss semantics, lease/renew/retry defaults from options.go:39-49).
"""
from __future__ import annotations

import logging
import socket
import threading
import uuid
from typing import Callable, Optional

from ..kube.client import ApiError, KubeApi
from ..utils.k8stime import format_time, parse_time

log = logging.getLogger(__name__)


class LeaderElector:
    def __init__(self, api: KubeApi, namespace: str, name: str,
                 identity: Optional[str] = None,
                 lease_duration: float = 15.0, renew_deadline: float = 10.0,
                 retry_period: float = 3.0):
        self.api = api
        self.namespace = namespace
        self.name = name
        self.identity = identity or f"{socket.gethostname()}_{uuid.uuid4().hex[:8]}"
        self.lease_duration = lease_duration
        self.renew_deadline = renew_deadline
        self.retry_period = retry_period
        self.is_leader = False

    def _lease_body(self, now: float, acquire: bool, current=None) -> dict:
        spec = {
            "holderIdentity": self.identity,
            "leaseDurationSeconds": int(self.lease_duration),
            "renewTime": format_time(now),
        }
        if acquire:
            spec["acquireTime"] = format_time(now)
            transitions = 0
            if current:
                transitions = current.get("spec", {}) \
                    .get("leaseTransitions", 0) + 1
            spec["leaseTransitions"] = transitions
        elif current:
            spec["acquireTime"] = current.get("spec", {}).get("acquireTime")
            spec["leaseTransitions"] = current.get("spec", {}) \
                .get("leaseTransitions", 0)
        return {"apiVersion": "coordination.k8s.io/v1", "kind": "Lease",
                "metadata": {"name": self.name,
                             "namespace": self.namespace},
                "spec": spec}

    def try_acquire_or_renew(self, now: float) -> bool:
        try:
            lease = self.api.get_lease(self.namespace, self.name)
        except ApiError as e:
            if not e.not_found:
                raise
            try:
                self.api.create_lease(self.namespace,
                                      self._lease_body(now, acquire=True))
                self.is_leader = True
                return True
            except ApiError:
                return False
        spec = lease.get("spec", {})
        holder = spec.get("holderIdentity", "")
        renew = parse_time(spec.get("renewTime")) or 0.0
        duration = spec.get("leaseDurationSeconds", self.lease_duration)
        if holder == self.identity:
            body = self._lease_body(now, acquire=False, current=lease)
            body["metadata"] = lease["metadata"]
            self.api.update_lease(self.namespace, self.name, body)
            self.is_leader = True
            return True
        if now - renew < duration:
            self.is_leader = False
            return False  # someone else holds a live lease
        # expired: take over
        body = self._lease_body(now, acquire=True, current=lease)
        body["metadata"] = lease["metadata"]
        try:
            self.api.update_lease(self.namespace, self.name, body)
            self.is_leader = True
            return True
        except ApiError:
            return False

    def run(self, on_started_leading: Callable[[], None],
            stop: threading.Event,
            on_lost: Optional[Callable[[], None]] = None) -> None:
        """Block until leadership, call on_started_leading (in a thread),
        then renew; on loss call on_lost (default: crash, like the
        reference's klog.Fatalf at server.go:102)."""
        import time
        while not stop.is_set():
            if self.try_acquire_or_renew(time.time()):
                break
            stop.wait(self.retry_period)
        if stop.is_set():
            return
        log.info("became leader as %s", self.identity)
        t = threading.Thread(target=on_started_leading, daemon=True)
        t.start()
        import time
        while not stop.is_set():
            stop.wait(self.retry_period)
            if stop.is_set():
                return
            deadline = time.time() + self.renew_deadline
            renewed = False
            while time.time() < deadline and not stop.is_set():
                try:
                    if self.try_acquire_or_renew(time.time()):
                        renewed = True
                        break
                except Exception:
                    log.exception("lease renew error")
                stop.wait(1.0)
            if not renewed:
                self.is_leader = False
                log.error("lost leader election lease")
                if on_lost:
                    on_lost()
                else:
                    raise SystemExit("leaderelection lost")
                return
