"""Durable standalone store: atomic snapshots + reload = the etcd analog.
A restarted operator resumes every state machine from disk instead of
forgetting the fleet ("the CRD status IS the checkpoint" —
true in cluster mode via etcd; this makes it true standalone)."""

import os

import pytest

from cro_amd.api.v1alpha1.types import ComposabilityRequest, ComposableResource
from cro_amd.runtime.client import Client
from cro_amd.runtime.store import InMemoryStore
from tests.conftest import make_request, make_resource


def test_snapshot_roundtrip(tmp_path):
    path = str(tmp_path / "state.json")
    store = InMemoryStore(persist_path=path)
    client = Client(store)
    req = client.create(make_request("r1", size=2, target_node="node0"))
    res = client.create(make_resource("gpu-1"))
    res.status.state = "Online"
    res.status.device_id = "GPU-abc"
    client.update_status(res)
    store.close()

    reloaded = InMemoryStore(persist_path=path)
    c2 = Client(reloaded)
    r = c2.get(ComposabilityRequest, "r1")
    assert r.spec.resource.size == 2
    assert r.metadata.uid == req.metadata.uid  # identity survives
    d = c2.get(ComposableResource, "gpu-1")
    assert d.status.state == "Online" and d.status.device_id == "GPU-abc"

    # RV monotonicity continues across the restart (no RV reuse)
    d.spec.model = "mi308x"
    updated = c2.update(d)
    assert int(updated.metadata.resourceVersion) > int(d.metadata.resourceVersion)
    reloaded.close()


def test_seq_continuity_expires_old_tokens(tmp_path):
    path = str(tmp_path / "state.json")
    store = InMemoryStore(persist_path=path)
    client = Client(store)
    client.create(make_request("r1", target_node="n"))
    token = store.current_seq()
    client.create(make_request("r2", model="m2", target_node="n"))
    store.close()

    reloaded = InMemoryStore(persist_path=path)
    # pre-restart token < seq but the log is empty → Expired → re-list
    assert reloaded.events_since(token) is None
    assert reloaded.events_since(reloaded.current_seq()) == []
    reloaded.close()


def test_corrupt_snapshot_starts_empty(tmp_path):
    path = tmp_path / "state.json"
    path.write_text("{not json")
    store = InMemoryStore(persist_path=str(path))
    assert store.list("ComposabilityRequest") == []
    # and it can persist fresh state over the corrupt file
    Client(store).create(make_request("r1", target_node="n"))
    store.close()
    reloaded = InMemoryStore(persist_path=str(path))
    assert len(reloaded.list("ComposabilityRequest")) == 1
    reloaded.close()


def test_debounced_background_persistence(tmp_path):
    import time

    path = str(tmp_path / "state.json")
    store = InMemoryStore(persist_path=path, persist_debounce=0.01)
    Client(store).create(make_request("r1", target_node="n"))
    deadline = time.monotonic() + 5
    while time.monotonic() < deadline and not os.path.exists(path):
        time.sleep(0.02)
    assert os.path.exists(path)  # written without close()
    store.close()


def test_restart_resumes_running_fleet(tmp_path):
    """Full managed-stack restart from disk: the reloaded operator sees the
    mid-flight objects and drives them to completion."""
    from cro_amd.bench_harness import build_local_stack

    path = str(tmp_path / "state.json")
    store = InMemoryStore(persist_path=path)
    stack = build_local_stack(node_name="n0", use_gpu=False, syncer_period=None)
    # rebuild the manager on the durable store
    from cro_amd.controllers import build_manager
    from cro_amd.fabric.adapter import Adapter
    from cro_amd.nodeops.amdgpu import MockNodeOps

    fabric = stack.fabric
    mgr = build_manager(Adapter("DRA", fabric), None, store=store)
    ops = MockNodeOps(client=mgr.client)
    mgr.resource_reconciler.node_ops = ops

    base_add = fabric.add_resource

    def add(resource):
        did, cdi = base_add(resource)
        ops.fabric_composed(resource.spec.target_node, did)
        return did, cdi

    # base_add already bridges the ORIGINAL stack's ops; rebind cleanly
    fabric.add_resource = add

    from tests.conftest import make_node

    make_node(mgr.client, "n0")
    ops.set_driver("n0", True)
    mgr.start()
    mgr.client.create(make_request("p1", size=1, target_node="n0"))
    assert mgr.wait_for(
        lambda: (r := mgr.client.try_get(ComposabilityRequest, "p1")) is not None
        and r.status.state == "Running",
        timeout=15,
    )
    mgr.stop()
    store.close()

    # "operator restart": fresh manager over the reloaded store
    store2 = InMemoryStore(persist_path=path)
    mgr2 = build_manager(Adapter("DRA", fabric), None, store=store2)
    ops2 = MockNodeOps(client=mgr2.client)
    mgr2.resource_reconciler.node_ops = ops2
    for did in fabric.attached_to("n0"):
        ops2.fabric_composed("n0", did)

    def add2(resource):
        did, cdi = base_add(resource)
        ops2.fabric_composed(resource.spec.target_node, did)
        return did, cdi

    fabric.add_resource = add2
    ops2.set_driver("n0", True)
    mgr2.start()
    r = mgr2.client.get(ComposabilityRequest, "p1")
    assert r.status.state == "Running"  # fleet survived the restart
    # and teardown runs to completion on the resumed state machines
    mgr2.client.delete(ComposabilityRequest, "p1")
    assert mgr2.wait_for(
        lambda: mgr2.client.try_get(ComposabilityRequest, "p1") is None, timeout=15
    )
    assert mgr2.wait_for(lambda: fabric.attached_to("n0") == [], timeout=15)
    mgr2.stop()
    store2.close()


def test_property_machine_reload_matches(tmp_path):
    """Persistence fuzz: after arbitrary interleavings from the store
    state machine, close + reload must reproduce the exact object set."""
    import hypothesis.strategies as st
    from hypothesis import given, settings

    from tests.test_store_properties import NAMES, make_obj

    @settings(max_examples=25, deadline=None)
    @given(st.lists(st.tuples(st.sampled_from(["create", "update", "status", "delete"]),
                              st.sampled_from(NAMES)), max_size=30))
    def run(ops):
        path = str(tmp_path / f"fuzz-{hash(tuple(ops)) & 0xffff}.json")
        if os.path.exists(path):
            os.unlink(path)
        store = InMemoryStore(persist_path=path)
        for op, name in ops:
            try:
                if op == "create":
                    store.create(make_obj(name))
                elif op == "update":
                    cur = store.get("ComposableResource", name)
                    cur.spec.model = "m1"
                    store.update(cur)
                elif op == "status":
                    cur = store.get("ComposableResource", name)
                    cur.status.state = "Online"
                    store.update_status(cur)
                else:
                    store.delete("ComposableResource", name)
            except Exception:
                pass
        expected = {
            o.metadata.name: (o.metadata.resourceVersion, o.status.state)
            for o in store.list("ComposableResource")
        }
        store.close()
        reloaded = InMemoryStore(persist_path=path)
        got = {
            o.metadata.name: (o.metadata.resourceVersion, o.status.state)
            for o in reloaded.list("ComposableResource")
        }
        reloaded.close()
        assert got == expected, (got, expected)

    run()
