"""Deduplicating work queue with rate-limited retries — the client-go
workqueue analogue the virtual-kubelet PodController uses (wired at reference
main.go:180-190; the reference runs it with a single worker, main.go:263 —
this one runs N)."""

from __future__ import annotations

import heapq
import threading
import time
from typing import Dict, Optional, Set


class WorkQueue:
    def __init__(self, base_delay_s: float = 0.05, max_delay_s: float = 30.0):
        self._cond = threading.Condition()
        self._queue: list = []  # heap of (ready_at, seq, key)
        self._queued: Set[str] = set()
        self._processing: Set[str] = set()
        self._dirty: Set[str] = set()
        self._failures: Dict[str, int] = {}
        self._seq = 0
        self._shutdown = False
        self.base_delay_s = base_delay_s
        self.max_delay_s = max_delay_s

    def add(self, key: str, delay_s: float = 0.0) -> None:
        with self._cond:
            if self._shutdown:
                return
            if key in self._processing:
                self._dirty.add(key)
                return
            if key in self._queued:
                return
            self._queued.add(key)
            self._seq += 1
            heapq.heappush(self._queue, (time.monotonic() + delay_s, self._seq, key))
            self._cond.notify()

    def add_rate_limited(self, key: str) -> None:
        with self._cond:
            failures = self._failures.get(key, 0)
            self._failures[key] = failures + 1
        delay = min(self.base_delay_s * (2 ** min(failures, 16)), self.max_delay_s)
        self.add(key, delay_s=delay)

    def forget(self, key: str) -> None:
        with self._cond:
            self._failures.pop(key, None)

    def get(self, timeout_s: Optional[float] = None) -> Optional[str]:
        deadline = None if timeout_s is None else time.monotonic() + timeout_s
        with self._cond:
            while True:
                if self._shutdown:
                    return None
                now = time.monotonic()
                if self._queue and self._queue[0][0] <= now:
                    _, _, key = heapq.heappop(self._queue)
                    self._queued.discard(key)
                    self._processing.add(key)
                    return key
                wait: Optional[float]
                if self._queue:
                    wait = self._queue[0][0] - now
                else:
                    wait = None
                if deadline is not None:
                    remaining = deadline - now
                    if remaining <= 0:
                        return None
                    wait = remaining if wait is None else min(wait, remaining)
                self._cond.wait(wait)

    def done(self, key: str) -> None:
        with self._cond:
            self._processing.discard(key)
            if key in self._dirty:
                self._dirty.discard(key)
                self._queued.add(key)
                self._seq += 1
                heapq.heappush(self._queue, (time.monotonic(), self._seq, key))
                self._cond.notify()

    def shutdown(self) -> None:
        with self._cond:
            self._shutdown = True
            self._cond.notify_all()

    def __len__(self) -> int:
        with self._cond:
            return len(self._queued) + len(self._processing)
