"""Per-state tests of the ComposabilityRequest 6-state machine."""

import pytest

from cro_amd.api.v1alpha1.types import (
    ComposabilityRequest,
    ComposableResource,
    NodeSpecRequirements,
    ScalarResourceStatus,
)
from cro_amd.controllers.composabilityrequest import (
    DELETE_DEVICE_ANNOTATION,
    LAST_USED_TIME_ANNOTATION,
)
from cro_amd.controllers.composableresource import FINALIZER, MANAGED_BY_LABEL
from tests.conftest import make_node, make_request, make_resource


def test_none_to_node_allocating(mock_world):
    mock_world.client.create(make_request("r1"))
    mock_world.request_rec.reconcile("r1")
    got = mock_world.client.get(ComposabilityRequest, "r1")
    assert got.status.state == "NodeAllocating"
    assert FINALIZER in got.metadata.finalizers
    assert got.status.scalarResource == got.spec.resource


def test_allocate_samenode_pinned(mock_world):
    make_node(mock_world.client, "node0")
    mock_world.client.create(make_request("r1", size=3, target_node="node0"))
    mock_world.request_rec.reconcile("r1")
    mock_world.request_rec.reconcile("r1")
    got = mock_world.client.get(ComposabilityRequest, "r1")
    assert got.status.state == "Updating"
    assert len(got.status.resources) == 3
    assert all(v.node_name == "node0" for v in got.status.resources.values())


def test_missing_target_node_garbage_collects(mock_world):
    # GC runs before state handling: a request pinned to a non-existent node
    # is deleted outright (composabilityrequest_controller.go:147-167)
    mock_world.client.create(make_request("r1", size=1, target_node="ghost"))
    # GC runs before state handling; with no finalizer yet the delete is final
    mock_world.request_rec.reconcile("r1")
    assert mock_world.client.try_get(ComposabilityRequest, "r1") is None


def test_allocate_samenode_node_vanishes_mid_allocation(mock_world):
    # the NodeAllocating "target node does not existed" error path needs the
    # node to pass GC but fail during allocation — race seam covered via a
    # node that exists at GC time and is deleted by a hook before allocation
    make_node(mock_world.client, "node0")
    mock_world.client.create(make_request("r1", size=1, target_node="node0"))
    mock_world.request_rec.reconcile("r1")
    from cro_amd.api.v1alpha1.types import Node
    import cro_amd.controllers.composabilityrequest as crq

    orig = crq.node_exists
    calls = {"n": 0}

    def flaky(client, name):
        calls["n"] += 1
        if calls["n"] == 1:
            return True  # GC check passes
        return False  # allocation check fails

    crq.node_exists = flaky
    try:
        with pytest.raises(ValueError):
            mock_world.request_rec.reconcile("r1")
    finally:
        crq.node_exists = orig
    got = mock_world.client.get(ComposabilityRequest, "r1")
    assert "does not existed" in got.status.error


def test_allocate_samenode_free_picks_unoccupied_node(mock_world):
    make_node(mock_world.client, "node0")
    make_node(mock_world.client, "node1")
    # r0 occupies node0
    mock_world.client.create(make_request("r0", size=1, target_node="node0"))
    mock_world.client.create(make_request("r1", size=2))
    mock_world.request_rec.reconcile("r1")
    mock_world.request_rec.reconcile("r1")
    got = mock_world.client.get(ComposabilityRequest, "r1")
    assert {v.node_name for v in got.status.resources.values()} == {"node1"}


def test_allocate_differentnode_spreads(mock_world):
    for i in range(3):
        make_node(mock_world.client, f"node{i}")
    mock_world.client.create(make_request("r1", size=3, policy="differentnode"))
    mock_world.request_rec.reconcile("r1")
    mock_world.request_rec.reconcile("r1")
    got = mock_world.client.get(ComposabilityRequest, "r1")
    nodes = [v.node_name for v in got.status.resources.values()]
    assert len(set(nodes)) == 3


def test_allocate_differentnode_insufficient_nodes(mock_world):
    make_node(mock_world.client, "node0")
    mock_world.client.create(make_request("r1", size=2, policy="differentnode"))
    mock_world.request_rec.reconcile("r1")
    with pytest.raises(ValueError):
        mock_world.request_rec.reconcile("r1")
    assert "insufficient" in mock_world.client.get(ComposabilityRequest, "r1").status.error


def test_other_spec_capacity_filters_nodes(mock_world):
    make_node(mock_world.client, "small", milli_cpu=1000)
    make_node(mock_world.client, "big", milli_cpu=64000)
    other = NodeSpecRequirements(milli_cpu=32000)
    mock_world.client.create(make_request("r1", size=1, other_spec=other))
    mock_world.request_rec.reconcile("r1")
    mock_world.request_rec.reconcile("r1")
    got = mock_world.client.get(ComposabilityRequest, "r1")
    assert {v.node_name for v in got.status.resources.values()} == {"big"}


def test_updating_creates_children_and_reaches_running(mock_world):
    make_node(mock_world.client, "node0")
    mock_world.client.create(make_request("r1", size=2, target_node="node0"))
    mock_world.request_rec.reconcile("r1")  # → NodeAllocating
    mock_world.request_rec.reconcile("r1")  # → Updating (names allocated)
    mock_world.request_rec.reconcile("r1")  # creates children
    children = mock_world.client.list(ComposableResource, {MANAGED_BY_LABEL: "r1"})
    assert len(children) == 2
    assert all(c.spec.target_node == "node0" for c in children)
    # drive children to Online (resource controller)
    for c in children:
        mock_world.resource_rec.reconcile(c.metadata.name)
        mock_world.resource_rec.reconcile(c.metadata.name)
    # sync child status into parent (dual-kind watch analog)
    for c in children:
        mock_world.request_rec.reconcile(c.metadata.name)
    mock_world.request_rec.reconcile("r1")
    got = mock_world.client.get(ComposabilityRequest, "r1")
    assert got.status.state == "Running"
    assert all(v.state == "Online" for v in got.status.resources.values())


def test_resource_change_syncs_into_parent(mock_world):
    make_node(mock_world.client, "node0")
    mock_world.client.create(make_request("r1", size=1, target_node="node0"))
    req = mock_world.client.get(ComposabilityRequest, "r1")
    req.status.state = "Updating"
    req.status.resources = {"gpu-c1": ScalarResourceStatus(node_name="node0")}
    req.status.scalarResource = req.spec.resource
    mock_world.client.update_status(req)
    child = make_resource("gpu-c1", managed_by="r1")
    mock_world.client.create(child)
    got_child = mock_world.client.get(ComposableResource, "gpu-c1")
    got_child.status.state = "Attaching"
    got_child.status.device_id = "GPU-123"
    mock_world.client.update_status(got_child)
    mock_world.request_rec.reconcile("gpu-c1")  # dual-kind path
    got = mock_world.client.get(ComposabilityRequest, "r1")
    assert got.status.resources["gpu-c1"].state == "Attaching"
    assert got.status.resources["gpu-c1"].device_id == "GPU-123"


def test_scale_down_uses_priority_buckets(mock_world):
    make_node(mock_world.client, "node0")
    mock_world.client.create(make_request("r1", size=4, target_node="node0"))
    req = mock_world.client.get(ComposabilityRequest, "r1")
    req.status.state = "NodeAllocating"
    req.status.scalarResource = req.spec.resource
    names = ["gpu-a", "gpu-b", "gpu-c", "gpu-d"]
    req.status.resources = {n: ScalarResourceStatus(node_name="node0") for n in names}
    mock_world.client.update_status(req)
    states = {
        "gpu-a": ("Online", {}),  # bucket 3
        "gpu-b": ("Attaching", {}),  # bucket 2 (has device id)
        "gpu-c": ("Online", {DELETE_DEVICE_ANNOTATION: "true"}),  # bucket 1
        "gpu-d": ("", {}),  # bucket 0 — deleted first
    }
    for n in names:
        r = make_resource(n, managed_by="r1")
        st, ann = states[n]
        for k, v in ann.items():
            r.metadata.annotations[k] = v
        mock_world.client.create(r)
        got = mock_world.client.get(ComposableResource, n)
        got.status.state = st
        if st in ("Attaching", "Online"):
            got.status.device_id = f"GPU-{n}"
        mock_world.client.update_status(got)
    # shrink to 2: buckets 0 (gpu-d) then 1 (gpu-c) evicted
    req = mock_world.client.get(ComposabilityRequest, "r1")
    req.spec.resource.size = 2
    mock_world.client.update(req)
    mock_world.request_rec.reconcile("r1")
    got = mock_world.client.get(ComposabilityRequest, "r1")
    assert set(got.status.resources) == {"gpu-a", "gpu-b"}


def test_scale_down_lru_within_bucket(mock_world):
    make_node(mock_world.client, "node0")
    mock_world.client.create(make_request("r1", size=2, target_node="node0"))
    req = mock_world.client.get(ComposabilityRequest, "r1")
    req.status.state = "NodeAllocating"
    req.status.scalarResource = req.spec.resource
    req.status.resources = {
        "gpu-old": ScalarResourceStatus(node_name="node0"),
        "gpu-new": ScalarResourceStatus(node_name="node0"),
    }
    mock_world.client.update_status(req)
    for n, ts in (("gpu-old", "2020-01-01T00:00:00Z"), ("gpu-new", "2030-01-01T00:00:00Z")):
        r = make_resource(n, managed_by="r1")
        r.metadata.annotations[LAST_USED_TIME_ANNOTATION] = ts
        mock_world.client.create(r)
        got = mock_world.client.get(ComposableResource, n)
        got.status.state = "Online"
        got.status.device_id = f"GPU-{n}"
        mock_world.client.update_status(got)
    req = mock_world.client.get(ComposabilityRequest, "r1")
    req.spec.resource.size = 1
    mock_world.client.update(req)
    mock_world.request_rec.reconcile("r1")
    got = mock_world.client.get(ComposabilityRequest, "r1")
    assert set(got.status.resources) == {"gpu-new"}  # LRU evicted


def test_running_spec_drift_returns_to_node_allocating(mock_world):
    make_node(mock_world.client, "node0")
    mock_world.client.create(make_request("r1", size=1, target_node="node0"))
    req = mock_world.client.get(ComposabilityRequest, "r1")
    req.status.state = "Running"
    req.status.scalarResource = req.spec.resource
    mock_world.client.update_status(req)
    req = mock_world.client.get(ComposabilityRequest, "r1")
    req.spec.resource.size = 4
    mock_world.client.update(req)
    mock_world.request_rec.reconcile("r1")
    assert mock_world.client.get(ComposabilityRequest, "r1").status.state == "NodeAllocating"


def test_cleaning_deletes_children_then_deleting(mock_world):
    make_node(mock_world.client, "node0")
    mock_world.client.create(make_request("r1", size=1, target_node="node0"))
    req = mock_world.client.get(ComposabilityRequest, "r1")
    req.metadata.finalizers = [FINALIZER]
    req = mock_world.client.update(req)
    req.status.state = "Cleaning"
    mock_world.client.update_status(req)
    mock_world.client.create(make_resource("gpu-x", managed_by="r1"))
    res = mock_world.request_rec.reconcile("r1")
    assert res.requeue_after is not None  # children still there
    # child deletion is immediate (no finalizer on it)
    assert mock_world.client.try_get(ComposableResource, "gpu-x") is None
    mock_world.request_rec.reconcile("r1")
    assert mock_world.client.get(ComposabilityRequest, "r1").status.state == "Deleting"
    mock_world.request_rec.reconcile("r1")  # finalizer removed; object stays
    assert mock_world.client.try_get(ComposabilityRequest, "r1") is not None
    mock_world.client.delete(ComposabilityRequest, "r1")
    assert mock_world.client.try_get(ComposabilityRequest, "r1") is None


def test_deletion_full_flow(mock_world):
    make_node(mock_world.client, "node0")
    mock_world.client.create(make_request("r1", size=1, target_node="node0"))
    for _ in range(3):
        mock_world.request_rec.reconcile("r1")
    mock_world.client.delete(ComposabilityRequest, "r1")
    # Updating state sees deletionTimestamp → Cleaning → Deleting → gone
    for _ in range(4):
        if mock_world.client.try_get(ComposabilityRequest, "r1") is None:
            break
        mock_world.request_rec.reconcile("r1")
    assert mock_world.client.try_get(ComposabilityRequest, "r1") is None
    assert mock_world.client.list(ComposableResource, {MANAGED_BY_LABEL: "r1"}) == []


def test_gc_on_target_node_deleted(mock_world):
    make_node(mock_world.client, "node0")
    mock_world.client.create(make_request("r1", size=1, target_node="node0"))
    mock_world.request_rec.reconcile("r1")
    from cro_amd.api.v1alpha1.types import Node

    mock_world.client.delete(Node, "node0")
    mock_world.request_rec.reconcile("r1")
    got = mock_world.client.get(ComposabilityRequest, "r1")
    assert got.metadata.deletionTimestamp is not None


def test_samenode_reallocation_keeps_implicit_node(mock_world):
    """Re-entering NodeAllocating with stale status entries but no
    surviving child CRs must keep the implicitly chosen node (resolved
    from status), never allocate children with an empty target."""
    from cro_amd.api.v1alpha1.types import ComposabilityRequest, ComposableResource
    from tests.conftest import drive, make_node, make_request

    w = mock_world
    make_node(w.client, "node0")
    make_node(w.client, "node1")
    w.ops.set_driver("node0", True)
    w.client.create(make_request("r1", size=1))  # samenode, no target
    drive(w.request_rec, "r1")
    req = w.client.get(ComposabilityRequest, "r1")
    chosen = next(iter(req.status.resources.values())).node_name
    assert chosen  # a node was picked

    # children vanish out-of-band (GC-like), status entries remain
    for child in w.client.list(ComposableResource):
        child.metadata.finalizers = []
        w.client.update(child)
        try:
            w.client.delete(ComposableResource, child.metadata.name)
        except Exception:
            pass
    req = w.client.get(ComposabilityRequest, "r1")
    req.status.state = "NodeAllocating"
    w.client.update_status(req)
    drive(w.request_rec, "r1")

    req = w.client.get(ComposabilityRequest, "r1")
    targets = {e.node_name for e in req.status.resources.values()}
    assert targets == {chosen}, targets  # same node, never ""
