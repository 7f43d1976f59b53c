"""Rate-limited work queue with the client-go semantics the reference's
controller relies on (reference: pkg/controller/controller.go:236-268,406-422):
dedup of queued keys, no concurrent processing of one key, re-queue of keys
marked dirty while processing, exponential per-item backoff, delayed adds.
"""
from __future__ import annotations

import heapq
import threading
import time
from typing import Dict, Optional, Set


class RateLimitedQueue:
    def __init__(self, base_delay: float = 0.005, max_delay: float = 60.0):
        self._cond = threading.Condition()
        self._queue: list = []          # FIFO of ready keys
        self._queued: Set[str] = set()
        self._processing: Set[str] = set()
        self._dirty: Set[str] = set()
        self._delayed: list = []        # heap of (ready_time, key)
        self._failures: Dict[str, int] = {}
        self._base = base_delay
        self._max = max_delay
        self._shutdown = False

    # -- producers --------------------------------------------------------
    def add(self, key: str) -> None:
        with self._cond:
            if key in self._processing:
                self._dirty.add(key)
                return
            if key not in self._queued:
                self._queued.add(key)
                self._queue.append(key)
                self._cond.notify()

    def add_after(self, key: str, delay: float) -> None:
        if delay <= 0:
            return self.add(key)
        with self._cond:
            heapq.heappush(self._delayed, (time.monotonic() + delay, key))
            self._cond.notify()

    def add_rate_limited(self, key: str) -> None:
        with self._cond:
            n = self._failures.get(key, 0)
            self._failures[key] = n + 1
        self.add_after(key, min(self._base * (2 ** n), self._max))

    def forget(self, key: str) -> None:
        with self._cond:
            self._failures.pop(key, None)

    # -- consumers ---------------------------------------------------------
    def get(self, timeout: Optional[float] = None) -> Optional[str]:
        deadline = None if timeout is None else time.monotonic() + timeout
        with self._cond:
            while True:
                self._drain_delayed()
                if self._queue:
                    key = self._queue.pop(0)
                    self._queued.discard(key)
                    self._processing.add(key)
                    return key
                if self._shutdown:
                    return None
                wait = 0.05
                if self._delayed:
                    wait = min(wait, max(0.0,
                               self._delayed[0][0] - time.monotonic()))
                if deadline is not None:
                    remaining = deadline - time.monotonic()
                    if remaining <= 0:
                        return None
                    wait = min(wait, remaining)
                self._cond.wait(wait if wait > 0 else 0.001)

    def done(self, key: str) -> None:
        with self._cond:
            self._processing.discard(key)
            if key in self._dirty:
                self._dirty.discard(key)
                if key not in self._queued:
                    self._queued.add(key)
                    self._queue.append(key)
                    self._cond.notify()

    def _drain_delayed(self) -> None:
        now = time.monotonic()
        while self._delayed and self._delayed[0][0] <= now:
            _, key = heapq.heappop(self._delayed)
            if key in self._processing:
                self._dirty.add(key)
            elif key not in self._queued:
                self._queued.add(key)
                self._queue.append(key)

    def shut_down(self) -> None:
        with self._cond:
            self._shutdown = True
            self._cond.notify_all()

    def __len__(self) -> int:
        with self._cond:
            return len(self._queue) + len(self._delayed)
