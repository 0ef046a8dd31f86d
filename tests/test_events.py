"""Event subsystem: recorder semantics (dedup, retention, never-throws)
and lifecycle emission — beyond the reference, which constructs no
EventRecorder anywhere (verified: no record.Event/Eventf call sites in
internal/controller)."""

import pytest

from cro_amd.api.v1alpha1.types import ComposabilityRequest, Event
from cro_amd.runtime.client import Client
from cro_amd.runtime.events import EventRecorder, NullRecorder
from cro_amd.runtime.store import InMemoryStore
from tests.conftest import drive, make_node, make_request


@pytest.fixture
def client():
    return Client(InMemoryStore())


# -- recorder semantics ------------------------------------------------------


def test_dedup_bumps_count(client):
    rec = EventRecorder(client)
    for _ in range(3):
        rec.normal(("ComposableResource", "gpu-1"), "Online", "device online")
    events = client.list(Event)
    assert len(events) == 1
    assert events[0].count == 3
    assert events[0].first_seen <= events[0].last_seen
    assert events[0].involved_kind == "ComposableResource"
    assert events[0].involved_name == "gpu-1"


def test_distinct_messages_are_distinct_events(client):
    rec = EventRecorder(client)
    rec.normal(("X", "a"), "Online", "m1")
    rec.warning(("X", "a"), "Online", "m2")
    assert len(client.list(Event)) == 2
    types = {e.type for e in client.list(Event)}
    assert types == {"Normal", "Warning"}


def test_retention_evicts_oldest(client):
    rec = EventRecorder(client, max_events=10)
    rec.EVICT_EVERY = 5  # keep the test cheap and deterministic
    for i in range(25):
        rec.normal(("X", f"obj-{i}"), "R", f"msg {i}")
    events = client.list(Event)
    # eviction is amortized: bounded by max + EVICT_EVERY overshoot
    assert len(events) <= 10 + 5
    names = {e.involved_name for e in events}
    assert "obj-0" not in names  # oldest gone
    assert "obj-24" in names  # newest kept


def test_async_mode_drains_off_thread(client):
    rec = EventRecorder(client, asynchronous=True)
    for i in range(20):
        rec.normal(("X", f"o{i}"), "R", "m")
    assert rec.flush(timeout=10)
    assert len(client.list(Event)) == 20
    # dedup still applies through the buffer
    for _ in range(3):
        rec.normal(("X", "o0"), "R", "m")
    assert rec.flush(timeout=10)
    ev = [e for e in client.list(Event) if e.involved_name == "o0"][0]
    assert ev.count == 4


def test_async_overflow_drops_oldest(client):
    class SlowClient:
        """Stand-in that blocks writes so the buffer can overflow."""

        def __init__(self, inner):
            self.inner = inner
            self.gate = __import__("threading").Event()

        def try_get(self, *a):
            self.gate.wait(10)
            return self.inner.try_get(*a)

        def __getattr__(self, name):
            return getattr(self.inner, name)

    slow = SlowClient(client)
    rec = EventRecorder(slow, asynchronous=True, buffer_size=8)
    for i in range(50):
        rec.normal(("X", f"o{i}"), "R", "m")
    slow.gate.set()
    assert rec.flush(timeout=10)
    stored = {e.involved_name for e in client.list(Event)}
    assert len(stored) <= 9  # bounded: 8 buffered + ≤1 in flight
    assert "o49" in stored  # newest survived the overflow


def test_recorder_never_raises():
    class Broken:
        def try_get(self, *a):
            raise RuntimeError("api down")

    rec = EventRecorder(Broken())
    rec.normal(("X", "a"), "R", "m")  # must not raise


def test_null_recorder_is_noop(client):
    NullRecorder().normal(("X", "a"), "R", "m")
    assert client.list(Event) == []


# -- lifecycle emission ------------------------------------------------------


def _reasons(client, name=None):
    evs = client.list(Event)
    if name:
        evs = [e for e in evs if e.involved_name == name]
    return {e.reason for e in evs}


def test_attach_lifecycle_emits_events(mock_world):
    w = mock_world
    make_node(w.client, "node0")
    w.client.create(make_request("req1", size=1, target_node="node0"))
    w.ops.set_driver("node0", True)
    drive(w.request_rec, "req1")
    req = w.client.get(ComposabilityRequest, "req1")
    child = next(iter(req.status.resources))
    drive(w.resource_rec, child)
    drive(w.request_rec, child)  # dual-kind sync
    drive(w.request_rec, "req1")

    assert {"NodesAllocated", "Running"} <= _reasons(w.client, "req1")
    child_reasons = _reasons(w.client, child)
    assert {"AttachStarted", "FabricAttached", "Online"} <= child_reasons

    # detach path
    w.client.delete(ComposabilityRequest, "req1")
    for _ in range(10):
        drive(w.request_rec, "req1")
        drive(w.resource_rec, child)
        drive(w.request_rec, child)
    child_reasons = _reasons(w.client, child)
    assert {"DetachStarted", "Detached"} <= child_reasons


def test_reconcile_error_emits_warning(mock_world):
    w = mock_world
    make_node(w.client, "node0")
    w.client.create(make_request("req1", size=1, target_node="node0"))
    w.ops.set_driver("node0", True)
    w.fabric.config.fail_attach = 99
    drive(w.request_rec, "req1")
    req = w.client.get(ComposabilityRequest, "req1")
    child = next(iter(req.status.resources))
    for _ in range(3):
        try:
            w.resource_rec.reconcile(child)
        except Exception:
            pass
    evs = [e for e in w.client.list(Event) if e.involved_name == child]
    warnings = [e for e in evs if e.type == "Warning"]
    assert warnings and warnings[0].reason == "ReconcileError"


# -- API + CLI surface -------------------------------------------------------


def test_events_served_and_croctl(mock_world):
    from fastapi.testclient import TestClient as HttpClient

    from cro_amd.cmd.croctl import main as croctl
    from cro_amd.server.api import build_app

    w = mock_world
    make_node(w.client, "node0")
    w.client.create(make_request("req1", size=1, target_node="node0"))
    w.ops.set_driver("node0", True)
    drive(w.request_rec, "req1")

    http = HttpClient(build_app(w.client))
    resp = http.get("/apis/cro.hpsys.ibm.ie.com/v1alpha1/events")
    assert resp.status_code == 200
    items = resp.json()["items"]
    assert any(e["reason"] == "NodesAllocated" for e in items)

    rc = croctl(["events"], client=http)
    assert rc == 0
    rc = croctl(["events", "--for", "ComposabilityRequest/req1"], client=http)
    assert rc == 0
    rc = croctl(["events", "--for", "ComposabilityRequest/ghost"], client=http)
    assert rc == 0
