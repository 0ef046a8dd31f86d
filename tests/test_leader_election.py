"""Lease-based leader election (cmd/main.go:137-155 parity).

Two operator candidates contend for the coordination Lease through the
same Client surface controllers use; failover is proven by killing the
leader (crash = stops renewing) and watching the standby take over.
"""

import threading
import time

import pytest

from cro_amd.api.v1alpha1.types import Lease
from cro_amd.controllers import build_manager
from cro_amd.fabric.adapter import Adapter
from cro_amd.fabric.mock import MockFabric
from cro_amd.runtime.lease import LEADER_ELECTION_ID, LeaderElector

FAST = dict(lease_duration=0.6, renew_deadline=0.4, retry_period=0.05)


@pytest.fixture
def client():
    mgr = build_manager(Adapter("DRA", MockFabric()), None, enable_webhook=False)
    yield mgr.client


def wait_for(pred, timeout=5.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if pred():
            return True
        time.sleep(0.02)
    return False


def test_single_elector_acquires(client):
    a = LeaderElector(client, identity="a", **FAST)
    leading = a.start()
    assert leading.wait(3)
    lease = client.get(Lease, LEADER_ELECTION_ID)
    assert lease.spec.holderIdentity == "a"
    assert lease.spec.renewTime
    a.stop()


def test_standby_blocks_while_leader_renews(client):
    a = LeaderElector(client, identity="a", **FAST)
    b = LeaderElector(client, identity="b", **FAST)
    assert a.start().wait(3)
    b.start()
    # b must not grab leadership while a renews (several lease periods)
    time.sleep(1.5)
    assert a.is_leader.is_set()
    assert not b.is_leader.is_set()
    assert client.get(Lease, LEADER_ELECTION_ID).spec.holderIdentity == "a"
    a.stop()
    b.stop()


def test_takeover_after_leader_crash(client):
    # "crash": acquire once and never renew (no clean release runs)
    a = LeaderElector(client, identity="a", **FAST)
    assert a._try_acquire_or_renew()
    assert client.get(Lease, LEADER_ELECTION_ID).spec.holderIdentity == "a"

    b = LeaderElector(client, identity="b", **FAST)
    b.start()
    assert b.is_leader.wait(5)  # takes over after lease_duration expires
    lease = client.get(Lease, LEADER_ELECTION_ID)
    assert lease.spec.holderIdentity == "b"
    assert lease.spec.leaseTransitions == 1
    b.stop()


def test_clean_shutdown_releases_immediately(client):
    a = LeaderElector(client, identity="a", **FAST)
    b = LeaderElector(client, identity="b", **FAST)
    assert a.start().wait(3)
    b.start()
    t0 = time.monotonic()
    a.stop()  # voluntary release zeroes the holder
    assert b.is_leader.wait(5)
    # takeover must beat the full expiry wait (release, not expiry)
    assert time.monotonic() - t0 < FAST["lease_duration"] + 1.0
    b.stop()


def test_no_double_leadership(client):
    """At no sampled instant do both candidates believe they lead."""
    a = LeaderElector(client, identity="a", **FAST)
    b = LeaderElector(client, identity="b", **FAST)
    a.start()
    b.start()
    overlap = []
    stop = threading.Event()

    def sample():
        while not stop.is_set():
            if a.is_leader.is_set() and b.is_leader.is_set():
                overlap.append(time.monotonic())
            time.sleep(0.005)

    t = threading.Thread(target=sample, daemon=True)
    t.start()
    time.sleep(1.0)
    # force churn: whoever leads shuts down, the other takes over
    (a if a.is_leader.is_set() else b).stop()
    time.sleep(1.0)
    stop.set()
    t.join(1)
    assert not overlap
    a.stop()
    b.stop()


def test_callbacks_fire(client):
    events = []
    a = LeaderElector(
        client, identity="a", **FAST,
        on_started_leading=lambda: events.append("started"),
        on_stopped_leading=lambda: events.append("stopped"),
    )
    assert a.start().wait(3)
    a.stop()
    assert wait_for(lambda: events == ["started", "stopped"])


def test_lost_renewal_fires_stopped(client):
    """A leader whose renewals stop landing (API unreachable) must demote
    itself within renew_deadline — split-brain prevention."""
    stopped = threading.Event()
    a = LeaderElector(
        client, identity="a", **FAST,
        on_stopped_leading=stopped.set,
    )
    assert a.start().wait(3)

    # sever the API: every write now fails
    real_update = client.update

    def failing_update(obj):
        from cro_amd.runtime.errors import ApiError

        raise ApiError("injected outage")

    client.update = failing_update
    try:
        assert stopped.wait(5)
        assert not a.is_leader.is_set()
    finally:
        client.update = real_update
        a.stop()
