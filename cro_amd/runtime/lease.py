"""Lease-based leader election (cmd/main.go:137-155 parity).

The reference manager elects a leader through a coordination.k8s.io Lease
with ID ``c5744f42.hpsys.ibm.ie.com``; standby replicas block until the
lease is free and take over when the holder stops renewing.  This module
implements the same protocol over the operator's Client surface, so the
election works identically against the embedded store (single node) and
the remote API server (split deployment, replicas on different nodes) —
unlike a local flock, which only excludes processes sharing a filesystem.

Protocol (client-go leaderelection semantics):

* acquire: create the Lease with ``holderIdentity=self``; if it exists and
  the holder's ``renewTime + leaseDurationSeconds`` has passed, take over
  with an optimistic-concurrency update (``leaseTransitions += 1``).
* renew: update ``renewTime`` every ``retry_period`` while leading; if a
  renewal cannot land within ``renew_deadline``, leadership is lost and
  ``on_stopped_leading`` fires (the caller must stop reconciling).
* conflicts: every write carries the observed resourceVersion, so two
  candidates racing for an expired lease serialize through the store's
  ConflictError — exactly one wins.
"""

from __future__ import annotations

import logging
import math
import threading
import time
import uuid
from typing import Callable, Optional

from ..api.v1alpha1.types import Lease
from .errors import AlreadyExistsError, ApiError, ConflictError

log = logging.getLogger(__name__)

LEADER_ELECTION_ID = "c5744f42.hpsys.ibm.ie.com"  # cmd/main.go:148


def _now_rfc3339(t: Optional[float] = None) -> str:
    if t is None:
        t = time.time()
    frac = int(round((t % 1) * 1e6))
    return time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime(t)) + f".{frac:06d}Z"


def _parse_rfc3339(s: str) -> float:
    import calendar

    base, _, frac = s.rstrip("Z").partition(".")
    t = calendar.timegm(time.strptime(base, "%Y-%m-%dT%H:%M:%S"))
    return t + (float("0." + frac) if frac else 0.0)


class LeaderElector:
    """Acquire/renew the election Lease; fire callbacks on transitions.

    ``run(stop)`` blocks until leadership is acquired, then renews until
    ``stop`` is set or a renewal misses ``renew_deadline``. ``start()``
    runs the same loop on a daemon thread and returns an Event that is set
    while this instance leads.
    """

    def __init__(
        self,
        client,
        identity: Optional[str] = None,
        name: str = LEADER_ELECTION_ID,
        lease_duration: float = 15.0,
        renew_deadline: float = 10.0,
        retry_period: float = 2.0,
        on_started_leading: Optional[Callable[[], None]] = None,
        on_stopped_leading: Optional[Callable[[], None]] = None,
    ):
        if renew_deadline >= lease_duration:
            raise ValueError("renew_deadline must be < lease_duration")
        self.client = client
        self.identity = identity or f"{uuid.uuid4()}"
        self.name = name
        self.lease_duration = lease_duration
        self.renew_deadline = renew_deadline
        self.retry_period = retry_period
        self.on_started_leading = on_started_leading
        self.on_stopped_leading = on_stopped_leading
        self.is_leader = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._stop = threading.Event()

    # -- one acquire-or-renew attempt ---------------------------------------

    def _try_acquire_or_renew(self) -> bool:
        now = time.time()
        try:
            lease = self.client.try_get(Lease, self.name)
        except ApiError:
            return False
        if lease is None:
            lease = Lease()
            lease.metadata.name = self.name
            lease.spec.holderIdentity = self.identity
            lease.spec.leaseDurationSeconds = int(math.ceil(self.lease_duration))
            lease.spec.acquireTime = _now_rfc3339(now)
            lease.spec.renewTime = _now_rfc3339(now)
            try:
                self.client.create(lease)
                return True
            except (AlreadyExistsError, ConflictError, ApiError):
                return False

        if lease.spec.holderIdentity == self.identity:
            lease.spec.renewTime = _now_rfc3339(now)
        else:
            renew = (
                _parse_rfc3339(lease.spec.renewTime)
                if lease.spec.renewTime
                else 0.0
            )
            if renew + lease.spec.leaseDurationSeconds > now:
                return False  # current holder is live
            # expired — take over
            lease.spec.holderIdentity = self.identity
            lease.spec.leaseDurationSeconds = int(math.ceil(self.lease_duration))
            lease.spec.acquireTime = _now_rfc3339(now)
            lease.spec.renewTime = _now_rfc3339(now)
            lease.spec.leaseTransitions += 1
        try:
            self.client.update(lease)  # carries observed resourceVersion
            return True
        except (ConflictError, ApiError):
            return False

    # -- main loop -----------------------------------------------------------

    def run(self, stop: Optional[threading.Event] = None) -> None:
        stop = stop or self._stop
        while not stop.is_set():
            if self._try_acquire_or_renew():
                break
            stop.wait(self.retry_period)
        if stop.is_set():
            return
        log.info("leader election: %s acquired %s", self.identity, self.name)
        self.is_leader.set()
        if self.on_started_leading:
            self.on_started_leading()
        try:
            last_renew = time.monotonic()
            while not stop.is_set():
                stop.wait(self.retry_period)
                if stop.is_set():
                    break
                if self._try_acquire_or_renew():
                    last_renew = time.monotonic()
                elif time.monotonic() - last_renew > self.renew_deadline:
                    log.warning(
                        "leader election: %s lost %s (renew deadline)",
                        self.identity, self.name,
                    )
                    break
        finally:
            self.is_leader.clear()
            if self.on_stopped_leading:
                self.on_stopped_leading()
            self._release(stop)

    def _release(self, stop: threading.Event) -> None:
        """Voluntary release on clean shutdown: zero the holder so a standby
        can take over immediately instead of waiting out lease_duration."""
        if not stop.is_set():
            return  # deadline-loss path: we no longer own it; leave as-is
        try:
            lease = self.client.try_get(Lease, self.name)
            if lease is not None and lease.spec.holderIdentity == self.identity:
                lease.spec.holderIdentity = ""
                lease.spec.renewTime = None
                self.client.update(lease)
        except Exception:
            pass

    def start(self) -> threading.Event:
        self._thread = threading.Thread(
            target=self.run, name=f"leader-elect-{self.identity[:8]}", daemon=True
        )
        self._thread.start()
        return self.is_leader

    def stop(self, timeout: float = 5.0) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout)
