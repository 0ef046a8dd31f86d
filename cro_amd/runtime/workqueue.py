"""Rate-limited deduplicating workqueue (client-go workqueue semantics).

The concurrency backbone of every controller: events enqueue request keys,
N workers drain them.  Guarantees the reference's controllers rely on:

* a key queued multiple times before being picked up is processed once;
* a key re-added while being processed is re-queued after Done (dirty set);
* ``add_rate_limited`` applies per-key exponential backoff, reset by
  ``forget`` on a successful reconcile;
* ``add_after`` schedules a delayed requeue (the reconciler's RequeueAfter).

Unlike the reference (fixed 30 s RequeueAfter while waiting on hardware,
composableresource_controller.go:236,298), our controllers lean on
``add_rate_limited`` with a millisecond-scale base so hardware readiness is
observed at event speed — that is the attach-latency headroom BASELINE.md
calls for.
"""

from __future__ import annotations

import heapq
import threading
import time
from typing import Dict, Hashable, List, Optional, Tuple


class RateLimitedQueue:
    def __init__(self, base_delay: float = 0.005, max_delay: float = 2.0):
        self._base = base_delay
        self._max = max_delay
        self._cond = threading.Condition()
        self._queue: List[Hashable] = []  # FIFO of ready keys
        self._queued: set = set()  # keys in _queue
        self._processing: set = set()
        self._dirty: set = set()  # re-added while processing
        self._failures: Dict[Hashable, int] = {}
        self._timers: List[Tuple[float, int, Hashable]] = []  # heap by deadline
        self._timer_seq = 0
        self._shutdown = False

    # -- producers ---------------------------------------------------------

    def add(self, key: Hashable) -> None:
        with self._cond:
            if self._shutdown:
                return
            if key in self._processing:
                self._dirty.add(key)
                return
            if key not in self._queued:
                self._queue.append(key)
                self._queued.add(key)
                self._cond.notify()

    def add_after(self, key: Hashable, delay: float) -> None:
        if delay <= 0:
            self.add(key)
            return
        with self._cond:
            if self._shutdown:
                return
            self._timer_seq += 1
            heapq.heappush(self._timers, (time.monotonic() + delay, self._timer_seq, key))
            self._cond.notify()

    def add_rate_limited(self, key: Hashable) -> None:
        with self._cond:
            n = self._failures.get(key, 0)
            self._failures[key] = n + 1
        # clamp the exponent: the failure count is unbounded and 2**n is an
        # arbitrary-precision int that overflows float math past ~2**1024
        self.add_after(key, min(self._base * (2 ** min(n, 32)), self._max))

    def forget(self, key: Hashable) -> None:
        with self._cond:
            self._failures.pop(key, None)

    def num_failures(self, key: Hashable) -> int:
        with self._cond:
            return self._failures.get(key, 0)

    # -- consumers ---------------------------------------------------------

    def get(self, timeout: Optional[float] = None) -> Optional[Hashable]:
        """Block until a key is ready (or timeout/shutdown → None)."""
        deadline = None if timeout is None else time.monotonic() + timeout
        with self._cond:
            while True:
                self._promote_timers()
                if self._queue:
                    key = self._queue.pop(0)
                    self._queued.discard(key)
                    self._processing.add(key)
                    return key
                if self._shutdown:
                    return None
                now = time.monotonic()
                if deadline is not None and now >= deadline:
                    return None
                waits = []
                if self._timers:
                    waits.append(self._timers[0][0] - now)
                if deadline is not None:
                    waits.append(deadline - now)
                self._cond.wait(max(min(waits), 0.0) if waits else None)

    def done(self, key: Hashable) -> None:
        with self._cond:
            self._processing.discard(key)
            if key in self._dirty:
                self._dirty.discard(key)
                if key not in self._queued:
                    self._queue.append(key)
                    self._queued.add(key)
                    self._cond.notify()

    def shutdown(self) -> None:
        with self._cond:
            self._shutdown = True
            self._cond.notify_all()

    def __len__(self) -> int:
        with self._cond:
            return len(self._queue) + len(self._timers)

    # -- internals ---------------------------------------------------------

    def _promote_timers(self) -> None:
        now = time.monotonic()
        while self._timers and self._timers[0][0] <= now:
            _, _, key = heapq.heappop(self._timers)
            if key not in self._processing and key not in self._queued:
                self._queue.append(key)
                self._queued.add(key)
            elif key in self._processing:
                self._dirty.add(key)

