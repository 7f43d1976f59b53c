"""In-flight create/delete expectations (reference: the client-go
ControllerExpectations used at controller.go:390-404 and pod.go:489-494).

The reference reads pods through an informer cache, so after issuing a
create it must suppress syncs until the watch event arrives or the cache
would show a gap and the controller would create duplicates. This
controller reads LIVE lists, so the race window is smaller, but the same
guard still matters for *ambiguous* API failures (a create that timed out
may or may not have landed) and for deletes with grace periods.

Observations are driven from three places:
  * the run loop's watch events (production, like the reference),
  * the fresh LIST at the top of every sync (``observe_list`` — a live
    list is at least as current as any informer cache), and
  * immediate cancellation when an API call fails cleanly.

Expectations expire after TTL seconds (client-go: 5 min) so a lost event
can never deadlock a job.
"""
from __future__ import annotations

import threading
import time
from typing import Dict, Set


class ControllerExpectations:
    TTL = 300.0

    def __init__(self) -> None:
        self._lock = threading.Lock()
        # key -> {"creates": set of names, "deletes": set of names, "ts": t}
        self._exp: Dict[str, dict] = {}

    def _entry(self, key: str) -> dict:
        return self._exp.setdefault(
            key, {"creates": set(), "deletes": set(), "ts": time.time()})

    # -- raise ---------------------------------------------------------
    def expect_creation(self, key: str, name: str) -> None:
        with self._lock:
            e = self._entry(key)
            e["creates"].add(name)
            e["ts"] = time.time()

    def expect_deletion(self, key: str, name: str) -> None:
        with self._lock:
            e = self._entry(key)
            e["deletes"].add(name)
            e["ts"] = time.time()

    # -- observe -------------------------------------------------------
    def creation_observed(self, key: str, name: str) -> None:
        with self._lock:
            e = self._exp.get(key)
            if e:
                e["creates"].discard(name)

    def deletion_observed(self, key: str, name: str) -> None:
        with self._lock:
            e = self._exp.get(key)
            if e:
                e["deletes"].discard(name)

    def observe_list(self, key: str, present_names: Set[str]) -> None:
        """Reconcile against a FRESH list: expected creations that are
        present and expected deletions that are gone have happened."""
        with self._lock:
            e = self._exp.get(key)
            if not e:
                return
            e["creates"] -= present_names
            e["deletes"] &= present_names

    # -- query ---------------------------------------------------------
    def satisfied(self, key: str, now: float = None) -> bool:
        now = time.time() if now is None else now
        with self._lock:
            e = self._exp.get(key)
            if not e:
                return True
            if not e["creates"] and not e["deletes"]:
                del self._exp[key]
                return True
            if now - e["ts"] > self.TTL:   # expired: never deadlock
                del self._exp[key]
                return True
            return False

    def forget(self, key: str) -> None:
        with self._lock:
            self._exp.pop(key, None)
