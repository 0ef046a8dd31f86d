"""Per-state tests of the ComposableResource 5-state machine, driving
reconcile() by hand (the reference's pattern: single-step deterministic
state-machine testing, composableresource_controller_test.go)."""

import pytest

from cro_amd.api.v1alpha1.types import ComposableResource, DeviceTaintRule
from cro_amd.controllers.composableresource import (
    FINALIZER,
    READY_TO_DETACH_CDI_LABEL,
    READY_TO_DETACH_LABEL,
)
from cro_amd.fabric.base import FabricError
from cro_amd.fabric.mock import MockFabricConfig
from cro_amd.nodeops.amdgpu import GPULoadsPresent
from tests.conftest import drive, make_node, make_resource


def seed(world, name="gpu-1", **kw):
    make_node(world.client, kw.pop("node", "node0"))
    return world.client.create(make_resource(name, **kw))


# -- None state -------------------------------------------------------------


def test_none_adds_finalizer_and_moves_to_attaching(mock_world):
    seed(mock_world)
    mock_world.resource_rec.reconcile("gpu-1")
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert FINALIZER in got.metadata.finalizers
    assert got.status.state == "Attaching"
    assert got.status.error == ""


def test_none_imports_ready_to_detach_labels(mock_world):
    make_node(mock_world.client, "node0")
    r = make_resource("gpu-1", labels={
        READY_TO_DETACH_LABEL: "GPU-dead",
        READY_TO_DETACH_CDI_LABEL: "amd.com/gpu=GPU-dead",
    })
    mock_world.client.create(r)
    mock_world.resource_rec.reconcile("gpu-1")
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert got.status.device_id == "GPU-dead"
    assert got.status.cdi_device_id == "amd.com/gpu=GPU-dead"
    assert got.status.state == "Attaching"


# -- Attaching --------------------------------------------------------------


def test_attaching_happy_path_to_online(mock_world):
    seed(mock_world)
    mock_world.resource_rec.reconcile("gpu-1")  # None → Attaching
    mock_world.resource_rec.reconcile("gpu-1")  # Attaching → Online
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert got.status.state == "Online"
    assert got.status.device_id.startswith("GPU-")
    assert got.status.cdi_device_id.startswith("amd.com/gpu=")
    # CDI spec written via node ops
    assert got.status.device_id in mock_world.ops.cdi_written["node0"]


def test_attaching_driver_missing_errors(mock_world):
    seed(mock_world)
    mock_world.ops.set_driver("node0", False)
    mock_world.resource_rec.reconcile("gpu-1")
    with pytest.raises(Exception):
        mock_world.resource_rec.reconcile("gpu-1")
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert "amdgpu" in got.status.error


def test_attaching_async_fabric_waits_then_lands(mock_world):
    mock_world.fabric.config = MockFabricConfig(asynchronous=True, attach_latency=0.05)
    seed(mock_world)
    mock_world.resource_rec.reconcile("gpu-1")
    res = mock_world.resource_rec.reconcile("gpu-1")  # WaitingDeviceAttaching
    assert res.requeue_after is not None
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert got.status.state == "Attaching"
    assert got.status.device_id == ""
    import time

    time.sleep(0.06)
    mock_world.resource_rec.reconcile("gpu-1")
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert got.status.state == "Online"


def test_attaching_visibility_pending_requeues_short(mock_world):
    mock_world.ops.attach_visible_delay = 0.2
    seed(mock_world)
    mock_world.resource_rec.reconcile("gpu-1")
    res = mock_world.resource_rec.reconcile("gpu-1")
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert got.status.state == "Attaching"
    assert got.status.device_id != ""  # fabric side done, node not yet visible
    assert res.requeue_after is not None
    assert res.requeue_after < 1.0  # sub-second, not the reference's 30 s


def test_attaching_fabric_error_recorded(mock_world):
    mock_world.fabric.config = MockFabricConfig(fail_attach=1)
    seed(mock_world)
    mock_world.resource_rec.reconcile("gpu-1")
    with pytest.raises(FabricError):
        mock_world.resource_rec.reconcile("gpu-1")
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert "injected attach failure" in got.status.error
    # retry succeeds
    mock_world.resource_rec.reconcile("gpu-1")
    assert mock_world.client.get(ComposableResource, "gpu-1").status.state == "Online"


def test_attaching_deletion_without_device_goes_deleting(mock_world):
    seed(mock_world)
    mock_world.resource_rec.reconcile("gpu-1")
    mock_world.client.delete(ComposableResource, "gpu-1")
    mock_world.resource_rec.reconcile("gpu-1")
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert got.status.state == "Deleting"


def test_attaching_probe_failure_blocks_online(mock_world):
    seed(mock_world)

    def failing_probe(node, device_id):
        return {"ok": False, "msg": "mfma mismatch"}

    mock_world.resource_rec.node_ops.health_probe = failing_probe
    mock_world.resource_rec.reconcile("gpu-1")
    with pytest.raises(FabricError):
        mock_world.resource_rec.reconcile("gpu-1")
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert got.status.state == "Attaching"
    assert "health probe failed" in got.status.error


# -- Online -----------------------------------------------------------------


def to_online(world, name="gpu-1", **kw):
    seed(world, name, **kw)
    world.resource_rec.reconcile(name)
    world.resource_rec.reconcile(name)
    got = world.client.get(ComposableResource, name)
    assert got.status.state == "Online"
    return got


def test_online_health_check_records_error(mock_world):
    got = to_online(mock_world)
    mock_world.fabric.config.unhealthy_devices.add(got.status.device_id)
    mock_world.resource_rec.reconcile("gpu-1")
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert "Critical" in got.status.error
    assert got.status.state == "Online"  # stays online, error surfaced
    mock_world.fabric.config.unhealthy_devices.clear()
    mock_world.resource_rec.reconcile("gpu-1")
    assert mock_world.client.get(ComposableResource, "gpu-1").status.error == ""


def test_online_deletion_moves_to_detaching(mock_world):
    to_online(mock_world)
    mock_world.client.delete(ComposableResource, "gpu-1")
    mock_world.resource_rec.reconcile("gpu-1")
    assert mock_world.client.get(ComposableResource, "gpu-1").status.state == "Detaching"


def test_online_ready_to_detach_label_triggers_delete(mock_world):
    make_node(mock_world.client, "node0")
    r = make_resource("gpu-1", labels={READY_TO_DETACH_LABEL: "GPU-x"})
    mock_world.client.create(r)
    mock_world.resource_rec.reconcile("gpu-1")  # → Attaching w/ device id
    got = mock_world.client.get(ComposableResource, "gpu-1")
    got.status.state = "Online"
    mock_world.client.update_status(got)
    mock_world.resource_rec.reconcile("gpu-1")
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert got.metadata.deletionTimestamp is not None  # delete issued


# -- Detaching --------------------------------------------------------------


def to_detaching(world, name="gpu-1", **kw):
    to_online(world, name, **kw)
    world.client.delete(ComposableResource, name)
    world.resource_rec.reconcile(name)
    return world.client.get(ComposableResource, name)


def test_detaching_full_path(mock_world):
    got = to_detaching(mock_world)
    device_id = got.status.device_id
    mock_world.resource_rec.reconcile("gpu-1")  # Detaching → Deleting
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert got.status.state == "Deleting"
    assert got.status.device_id == ""
    assert got.status.cdi_device_id == ""
    # drained from node ops, CDI removed, fabric detached
    assert device_id not in mock_world.ops.visible.get("node0", set())
    assert device_id not in mock_world.ops.cdi_written.get("node0", set())
    assert device_id not in mock_world.fabric.attached_to("node0")
    mock_world.resource_rec.reconcile("gpu-1")  # Deleting → finalizer removed
    assert mock_world.client.try_get(ComposableResource, "gpu-1") is None


def test_detaching_blocked_by_loads(mock_world):
    got = to_detaching(mock_world)
    mock_world.ops.add_load("node0", got.status.device_id)
    with pytest.raises(GPULoadsPresent):
        mock_world.resource_rec.reconcile("gpu-1")
    assert mock_world.client.get(ComposableResource, "gpu-1").status.state == "Detaching"
    mock_world.ops.clear_loads("node0")
    mock_world.resource_rec.reconcile("gpu-1")
    assert mock_world.client.get(ComposableResource, "gpu-1").status.state == "Deleting"


def test_detaching_force_detach_bypasses_loads(mock_world):
    got = to_detaching(mock_world, force_detach=True)
    mock_world.ops.add_load("node0", got.status.device_id)
    mock_world.resource_rec.reconcile("gpu-1")
    assert mock_world.client.get(ComposableResource, "gpu-1").status.state == "Deleting"


def test_detaching_creates_and_removes_taint_dra(mock_world):
    to_detaching(mock_world)
    # spy: taint must exist mid-detach; MockNodeOps drains synchronously so
    # check that a full detach leaves no taint rule behind
    mock_world.resource_rec.reconcile("gpu-1")
    assert mock_world.client.list(DeviceTaintRule) == []


def test_detaching_async_fabric_waits(mock_world):
    mock_world.fabric.config = MockFabricConfig(asynchronous=True, detach_latency=0.05)
    got = to_detaching(mock_world)
    res = mock_world.resource_rec.reconcile("gpu-1")
    assert res.requeue_after is not None
    assert mock_world.client.get(ComposableResource, "gpu-1").status.state == "Detaching"
    import time

    time.sleep(0.06)
    mock_world.resource_rec.reconcile("gpu-1")
    assert mock_world.client.get(ComposableResource, "gpu-1").status.state == "Deleting"


def test_detaching_device_plugin_checks_whole_node(mock_world):
    mock_world.adapter.device_resource_type = "DEVICE_PLUGIN"
    to_detaching(mock_world)
    mock_world.ops.add_load("node0", "GPU-other-device")  # other device busy
    with pytest.raises(GPULoadsPresent):
        mock_world.resource_rec.reconcile("gpu-1")  # whole-node check trips


# -- garbage collection -----------------------------------------------------


def test_gc_on_node_deletion(mock_world):
    to_online(mock_world)
    from cro_amd.api.v1alpha1.types import Node

    mock_world.client.delete(Node, "node0")
    mock_world.resource_rec.reconcile("gpu-1")
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert got.status.state == "Deleting"
    assert got.metadata.deletionTimestamp is not None
    assert "not found" in got.status.error
    mock_world.resource_rec.reconcile("gpu-1")  # Deleting → finalizer off → gone
    assert mock_world.client.try_get(ComposableResource, "gpu-1") is None


def test_fabric_wait_resumes_at_max_after_restart(mock_world):
    """the exponential async-fabric wait must survive an
    operator restart — status.fabric_wait_started is persisted on the first
    Waiting, and a fresh reconciler (empty in-memory counters) resumes at
    fabric_wait_max instead of re-ramping from the base interval."""
    from cro_amd.controllers.composableresource import (
        ComposableResourceReconciler,
        ReconcileConfig,
    )
    from cro_amd.fabric import WaitingDeviceAttaching

    seed(mock_world)
    # fabric stuck composing: every add_resource says "still attaching"
    def waiting_add(resource):
        raise WaitingDeviceAttaching("composing")

    mock_world.fabric.add_resource = waiting_add
    rec = mock_world.resource_rec
    rec.reconcile("gpu-1")  # None → Attaching
    r1 = rec.reconcile("gpu-1")  # first Waiting
    assert r1.requeue_after == rec.config.fabric_wait_base
    # short waits stay write-free (the ramp is not skewed by self-requeues)
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert got.status.fabric_wait_started == ""

    r2 = rec.reconcile("gpu-1")
    assert r2.requeue_after == rec.config.fabric_wait_base * 2  # ramping

    # keep polling until the ramp caps → the marker is persisted
    for _ in range(10):
        rec.reconcile("gpu-1")
        if mock_world.client.get(ComposableResource, "gpu-1").status.fabric_wait_started:
            break
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert got.status.fabric_wait_started  # persisted at the cap

    # "restart": fresh reconciler over the same store, no in-memory state
    rec2 = ComposableResourceReconciler(
        mock_world.client, mock_world.adapter, mock_world.ops, ReconcileConfig()
    )
    r3 = rec2.reconcile("gpu-1")
    assert r3.requeue_after == rec2.config.fabric_wait_max  # no re-ramp

    # fabric completes → wait state cleared
    from tests.conftest import mock_world as _unused  # noqa: F401

    def ok_add(resource):
        mock_world.ops.fabric_composed(resource.spec.target_node, "GPU-done")
        return "GPU-done", "amd.com/gpu=GPU-done"

    mock_world.fabric.add_resource = ok_add
    for _ in range(5):
        rec2.reconcile("gpu-1")
    got = mock_world.client.get(ComposableResource, "gpu-1")
    assert got.status.state == "Online"
    assert got.status.fabric_wait_started == ""
