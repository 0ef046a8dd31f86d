"""DRA surface tests (ResourceSlice publication, slice-based visibility) and
the syncer running as a manager runnable (ticker path)."""

import time

import pytest

from cro_amd.api.v1alpha1.types import ComposableResource, Node, ResourceSlice
from cro_amd.nodeops.amdgpu import AmdNodeOps
from cro_amd.nodeops.execs import MockNodeExec
from cro_amd.runtime.client import Client
from cro_amd.runtime.store import InMemoryStore
from tests.conftest import make_node, make_request
from tests.test_nodeops import NODE, kfd_fixture


def test_resource_slice_publication():
    ex = MockNodeExec()
    ids = kfd_fixture(ex, 2)
    client = Client(InMemoryStore())
    ops = AmdNodeOps(ex, client=client, destructive=False)
    ops.refresh_after_attach(NODE)
    slices = client.list(ResourceSlice)
    assert len(slices) == 1
    sl = slices[0]
    assert sl.metadata.name == f"{NODE}-gpu-pool"
    assert sl.spec.node_name == NODE
    assert {d.uuid for d in sl.spec.devices} == set(ids)
    d0 = next(d for d in sl.spec.devices if d.uuid == ids[0])
    assert d0.attributes["pci-bdf"] == "0000:03:00.0"
    assert d0.attributes["vram-bytes"] == "309237645312"
    assert d0.attributes["xgmi-peers"] == "2"

    # drain one device → republished slice drops it
    ops.drain(NODE, ids[0])
    ops.refresh_after_detach(NODE)
    sl = client.list(ResourceSlice)[0]
    assert {d.uuid for d in sl.spec.devices} == {ids[1]}


def test_is_visible_dra_uses_slice():
    ex = MockNodeExec()
    ids = kfd_fixture(ex, 1)
    client = Client(InMemoryStore())
    ops = AmdNodeOps(ex, client=client, destructive=False)
    # no slice published yet → not visible through DRA even though sysfs
    # enumerates the device
    assert not ops.is_visible_dra(NODE, ids[0])
    ops.refresh_after_attach(NODE)
    assert ops.is_visible_dra(NODE, ids[0])
    assert not ops.is_visible_dra(NODE, "GPU-ghost")


def test_syncer_runs_as_manager_runnable():
    """The ticker path (upstreamsyncer_controller.go:52-77 analog): drift is
    repaired without manual sync() calls."""
    from cro_amd.bench_harness import build_local_stack

    stack = build_local_stack(node_name="node0", use_gpu=False)
    # re-wire with a fast syncer before starting
    from cro_amd.controllers.upstreamsyncer import UpstreamSyncer

    syncer = UpstreamSyncer(stack.mgr.client, stack.mgr.resource_reconciler.adapter,
                            stack.ops, grace_period=0.05)
    stack.mgr.add_runnable(0.05, syncer.sync)
    stack.mgr.start()
    try:
        did = next(iter(stack.fabric._pool))
        stack.fabric.force_attach(did, "node0")  # out-of-band drift
        stack.ops.visible.setdefault("node0", set()).add(did)
        # ticker tracks, grace expires, detach CR created and drives the
        # physical detach end to end
        assert stack.mgr.wait_for(
            lambda: stack.fabric.attached_to("node0") == [], timeout=15
        )
        assert stack.mgr.wait_for(
            lambda: stack.mgr.client.list(ComposableResource) == [], timeout=10
        )
    finally:
        stack.mgr.stop()


def test_differentnode_keep_and_evict_existing_children(mock_world):
    """NodeAllocating keep/evict over existing children: duplicates on one
    node are evicted under differentnode policy (:261-272)."""
    from cro_amd.api.v1alpha1.types import ComposabilityRequest, ScalarResourceStatus
    from tests.conftest import make_resource

    for i in range(3):
        make_node(mock_world.client, f"node{i}")
    mock_world.client.create(make_request("r1", size=2, policy="differentnode"))
    req = mock_world.client.get(ComposabilityRequest, "r1")
    req.status.state = "NodeAllocating"
    req.status.scalarResource = req.spec.resource
    req.status.resources = {
        "gpu-a": ScalarResourceStatus(node_name="node0"),
        "gpu-b": ScalarResourceStatus(node_name="node0"),  # duplicate node
    }
    mock_world.client.update_status(req)
    for name in ("gpu-a", "gpu-b"):
        r = make_resource(name, managed_by="r1", target_node="node0")
        mock_world.client.create(r)
        got = mock_world.client.get(ComposableResource, name)
        got.status.state = "Online"
        got.status.device_id = f"GPU-{name}"
        mock_world.client.update_status(got)
    mock_world.request_rec.reconcile("r1")
    got = mock_world.client.get(ComposabilityRequest, "r1")
    nodes = {v.node_name for v in got.status.resources.values()}
    assert len(got.status.resources) == 2
    assert len(nodes) == 2  # second device re-allocated to a distinct node


def test_samenode_evicts_child_on_wrong_node(mock_world):
    from cro_amd.api.v1alpha1.types import ComposabilityRequest, ScalarResourceStatus
    from tests.conftest import make_resource

    make_node(mock_world.client, "node0")
    make_node(mock_world.client, "node1")
    mock_world.client.create(make_request("r1", size=2))
    req = mock_world.client.get(ComposabilityRequest, "r1")
    req.status.state = "NodeAllocating"
    req.status.scalarResource = req.spec.resource
    req.status.resources = {
        "gpu-a": ScalarResourceStatus(node_name="node0"),
        "gpu-b": ScalarResourceStatus(node_name="node1"),  # violates samenode
    }
    mock_world.client.update_status(req)
    for name, node in (("gpu-a", "node0"), ("gpu-b", "node1")):
        r = make_resource(name, managed_by="r1", target_node=node)
        mock_world.client.create(r)
    mock_world.request_rec.reconcile("r1")
    got = mock_world.client.get(ComposabilityRequest, "r1")
    nodes = {v.node_name for v in got.status.resources.values()}
    assert nodes == {"node0"}  # gpu-b evicted, replacement on node0
    assert len(got.status.resources) == 2


def test_status_update_conflict(client):
    from cro_amd.runtime.errors import ConflictError

    created = client.create(make_request("r1"))
    stale = created.model_copy(deep=True)
    created.status.state = "NodeAllocating"
    client.update_status(created)
    stale.status.state = "Bogus"
    with pytest.raises(ConflictError):
        client.update_status(stale)
