"""Manager: wires controllers, runnables, admission, metrics, health.

Equivalent of ctrl.NewManager + SetupWithManager wiring (cmd/main.go:137-201):
owns the store/client, starts every controller's watch pump + workers, runs
periodic runnables (the upstream syncer's 1-min ticker analog,
upstreamsyncer_controller.go:52-77), registers admission validators (the
in-process form of the validating webhook), and exposes Prometheus metrics
via prometheus_client when a port is given.
"""

from __future__ import annotations

import threading
import time
from typing import Callable, List, Optional, Tuple

from ..metrics import Metrics
from .client import Client
from .controller import Controller
from .store import InMemoryStore


class Manager:
    def __init__(
        self,
        store: Optional[InMemoryStore] = None,
        metrics_port: Optional[int] = None,
        client=None,
    ):
        """``client`` overrides the in-memory store with any object
        implementing the Client surface plus ``watch``/``list`` (e.g.
        runtime.remote.RemoteClient) — controllers then run against a remote
        API server; admission lives server-side in that mode."""
        if client is not None:
            self.store = client  # controllers watch/list through it
            self.client = client
        else:
            self.store = store or InMemoryStore()
            self.client = Client(self.store)
        self.metrics = Metrics()
        self._metrics_port = metrics_port
        self._controllers: List[Controller] = []
        # (period_seconds, fn) — fn is called every period until stop
        self._runnables: List[Tuple[float, Callable[[], None]]] = []
        self._threads: List[threading.Thread] = []
        self._stop = threading.Event()
        self._started = False

    def add_controller(self, c: Controller) -> None:
        c._metrics = self.metrics
        self._controllers.append(c)

    def add_runnable(self, period: float, fn: Callable[[], None]) -> None:
        self._runnables.append((period, fn))

    def register_admission(self, kind: str, fn) -> None:
        if not hasattr(self.store, "register_admission"):
            import logging

            logging.getLogger(__name__).info(
                "remote-client mode: admission for %s is enforced server-side", kind
            )
            return
        self.store.register_admission(kind, fn)

    def start(self) -> None:
        if self._started:
            return
        self._started = True
        if self._metrics_port is not None:
            import prometheus_client

            prometheus_client.start_http_server(self._metrics_port)
        # two-phase start: all watches subscribed (with cache replay) before
        # any worker runs, so no controller can advance a state machine past
        # events a sibling controller has not yet subscribed to
        for c in self._controllers:
            c.start_watch(self.store)
        for c in self._controllers:
            c.start_workers()
        for period, fn in self._runnables:
            t = threading.Thread(
                target=self._tick, args=(period, fn), name="runnable", daemon=True
            )
            t.start()
            self._threads.append(t)

    def _tick(self, period: float, fn: Callable[[], None]) -> None:
        while not self._stop.wait(period):
            try:
                fn()
            except Exception:  # runnables log their own errors; never die
                import logging

                logging.getLogger(__name__).exception("runnable failed")

    def stop(self) -> None:
        self._stop.set()
        for c in self._controllers:
            c.stop()
        for t in self._threads:
            t.join(timeout=2)
        recorder = getattr(self, "recorder", None)
        if recorder is not None:  # drain buffered events before exit
            recorder.flush(timeout=2.0)

    # -- helpers for tests/benches ----------------------------------------

    def wait_for(self, predicate: Callable[[], bool], timeout: float = 10.0, poll: float = 0.002) -> bool:
        """Busy-wait helper; returns True when predicate holds within timeout."""
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if predicate():
                return True
            time.sleep(poll)
        return predicate()
