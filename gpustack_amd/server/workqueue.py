"""Rate-limited work queue with exponential backoff (reference:
gpustack/server/workqueue.py:94,130 — the k8s-style queue the reconcilers
drain: coalescing adds, per-item exponential backoff on failure, delayed
re-adds).

Controllers use it so a persistently failing item (bad model spec, dead
provider) retries at 10s * 2^n capped instead of hot-looping, while new
events for the same item coalesce into one pending entry."""
from __future__ import annotations

import heapq
import threading
import time


class ExponentialBackoff:
    def __init__(self, base: float = 1.0, cap: float = 300.0):
        self.base = base
        self.cap = cap
        self.failures: dict = {}

    def next_delay(self, item) -> float:
        n = self.failures.get(item, 0)
        self.failures[item] = n + 1
        return min(self.base * (2 ** n), self.cap)

    def forget(self, item) -> None:
        self.failures.pop(item, None)


class WorkQueue:
    """add() coalesces; get() blocks until an item is due; done() must be
    called after processing — with requeue=True the item comes back after
    its (growing) backoff delay."""

    def __init__(self, base_delay: float = 1.0, max_delay: float = 300.0):
        self._lock = threading.Condition()
        self._pending: set = set()        # queued or delayed
        self._processing: set = set()
        self._dirty: set = set()          # re-added while processing
        self._heap: list[tuple[float, int, object]] = []  # (due, seq, item)
        self._seq = 0
        self.backoff = ExponentialBackoff(base_delay, max_delay)
        self._shutdown = False

    def add(self, item, delay: float = 0.0) -> None:
        with self._lock:
            if item in self._processing:
                self._dirty.add(item)  # coalesce: re-add when done() runs
                return
            if item in self._pending:
                return  # coalesced
            self._pending.add(item)
            heapq.heappush(self._heap, (time.monotonic() + delay,
                                        self._seq, item))
            self._seq += 1
            self._lock.notify()

    def get(self, timeout: float | None = None):
        """Next due item, or None on timeout/shutdown."""
        deadline = None if timeout is None else time.monotonic() + timeout
        with self._lock:
            while True:
                if self._shutdown:
                    return None
                now = time.monotonic()
                if self._heap and self._heap[0][0] <= now:
                    _, _, item = heapq.heappop(self._heap)
                    self._pending.discard(item)
                    self._processing.add(item)
                    return item
                wait = self._heap[0][0] - now if self._heap else None
                if deadline is not None:
                    remain = deadline - now
                    if remain <= 0:
                        return None
                    wait = remain if wait is None else min(wait, remain)
                self._lock.wait(wait)

    def done(self, item, requeue: bool = False) -> None:
        readd = False
        with self._lock:
            self._processing.discard(item)
            if item in self._dirty:
                self._dirty.discard(item)
                readd = True
        if requeue:
            self.add(item, delay=self.backoff.next_delay(item))
            return
        self.backoff.forget(item)
        if readd:
            self.add(item)

    def shutdown(self) -> None:
        with self._lock:
            self._shutdown = True
            self._lock.notify_all()

    def __len__(self) -> int:
        with self._lock:
            return len(self._pending) + len(self._processing)
