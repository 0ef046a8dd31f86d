"""Controller: watch sources → predicates → workqueue → reconcile workers.

Equivalent of controller-runtime's controller + builder
(composabilityrequest_controller.go:681-690, composableresource_controller.go:457-461)
with one deliberate departure: ``max_concurrent_reconciles`` defaults to 8 for
per-device fan-out (the reference leaves the default of 1, which serializes an
8-GPU attach — SURVEY.md §6).

A reconciler returns :class:`Result` (requeue_after seconds) or raises; a
raised error is written back by the reconciler itself (requeueOnErr parity)
and the key is re-queued with exponential backoff.
"""

from __future__ import annotations

import logging
import threading
from dataclasses import dataclass
from typing import Callable, List, Optional

from .store import WatchEvent
from .workqueue import RateLimitedQueue

log = logging.getLogger(__name__)

# Request keys are object names (both CRDs are cluster-scoped).
Request = str


@dataclass
class Result:
    requeue_after: Optional[float] = None
    requeue: bool = False


class Reconciler:
    def reconcile(self, req: Request) -> Result:  # pragma: no cover - interface
        raise NotImplementedError


@dataclass
class Source:
    """One watched kind feeding this controller's queue."""

    kind: str
    # predicate(event) -> bool; default: everything
    predicate: Optional[Callable[[WatchEvent], bool]] = None
    # mapper(event) -> list of request keys; default: [object name]
    mapper: Optional[Callable[[WatchEvent], List[Request]]] = None


class Controller:
    def __init__(
        self,
        name: str,
        reconciler: Reconciler,
        sources: List[Source],
        max_concurrent_reconciles: int = 8,
        base_backoff: float = 0.005,
        max_backoff: float = 2.0,
    ):
        self.name = name
        self.reconciler = reconciler
        self.sources = sources
        self.workers = max_concurrent_reconciles
        self.queue = RateLimitedQueue(base_delay=base_backoff, max_delay=max_backoff)
        self._threads: List[threading.Thread] = []
        self._stop = threading.Event()
        self._metrics = None  # set by manager

    # -- event intake ------------------------------------------------------

    def start(self, store) -> None:
        self.start_watch(store)
        self.start_workers()

    def start_watch(self, store) -> None:
        """Phase 1: subscribe + informer-cache replay.  The manager runs this
        for EVERY controller before any worker starts (manager.py) — workers
        of one controller must not advance state machines before sibling
        controllers are watching, or cross-controller events are lost."""
        kinds = sorted({s.kind for s in self.sources})
        events = store.watch(kinds)
        self._store = store
        self._events = events

        # replay (controller-runtime parity): every object already in the
        # store is delivered as a synthetic ADDED event so a restarted
        # operator resumes mid-state-machine without an external nudge.
        # Subscribing BEFORE the list means an object created in the gap is
        # seen twice, never missed — the workqueue dedupes.
        from .store import WatchEvent

        for kind in kinds:
            for obj in store.list(kind):
                self._dispatch(WatchEvent("ADDED", obj))

        def pump():
            while not self._stop.is_set():
                try:
                    ev = events.get(timeout=0.1)
                except Exception:
                    continue
                self._dispatch(ev)

        t = threading.Thread(target=pump, name=f"{self.name}-watch", daemon=True)
        t.start()
        self._threads.append(t)

    def start_workers(self) -> None:
        for i in range(self.workers):
            t = threading.Thread(target=self._worker, name=f"{self.name}-worker-{i}", daemon=True)
            t.start()
            self._threads.append(t)

    def _dispatch(self, ev: WatchEvent) -> None:
        for s in self.sources:
            if s.kind != ev.object.kind:
                continue
            if s.predicate is not None and not s.predicate(ev):
                continue
            reqs = s.mapper(ev) if s.mapper else [ev.object.metadata.name]
            for r in reqs:
                self.queue.add(r)

    # -- workers -----------------------------------------------------------

    def _worker(self) -> None:
        while True:
            key = self.queue.get(timeout=0.2)
            if key is None:
                if self._stop.is_set():
                    return
                continue
            try:
                self._process(key)
            finally:
                self.queue.done(key)

    def _process(self, key: Request) -> None:
        try:
            result = self.reconciler.reconcile(key)
        except Exception as exc:  # requeue with backoff (controller-runtime parity)
            log.debug("%s: reconcile %s failed: %s", self.name, key, exc)
            if self._metrics:
                self._metrics.reconcile_total.labels(self.name, "error").inc()
            self.queue.add_rate_limited(key)
            return
        if self._metrics:
            self._metrics.reconcile_total.labels(self.name, "success").inc()
        self.queue.forget(key)
        if result and result.requeue_after is not None:
            self.queue.add_after(key, result.requeue_after)
        elif result and result.requeue:
            self.queue.add(key)

    def stop(self) -> None:
        self._stop.set()
        self.queue.shutdown()
        for t in self._threads:
            t.join(timeout=2)
        # unsubscribe the watch queue: a store outliving this controller
        # (stack rebuilt on the same store) must not fan out to dead queues
        store = getattr(self, "_store", None)
        if store is not None and hasattr(store, "stop_watch"):
            store.stop_watch(self._events)
