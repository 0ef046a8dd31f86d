"""Upstream syncer drift repair: grace-period tracking and detach-CR
creation (upstreamsyncer_controller_test.go:646-855 analog — the sync
function is driven directly, not through the ticker)."""

from cro_amd.api.v1alpha1.types import ComposableResource
from cro_amd.controllers.composableresource import (
    READY_TO_DETACH_CDI_LABEL,
    READY_TO_DETACH_LABEL,
)
from cro_amd.controllers.upstreamsyncer import UpstreamSyncer
from tests.conftest import make_node, make_resource


def make_syncer(world, grace=0.0):
    return UpstreamSyncer(world.client, world.adapter, world.ops, grace_period=grace)


def attach_one(world, node="node0"):
    """Compose a device out-of-band (fabric drift)."""
    did = next(iter(world.fabric._pool))
    world.fabric.force_attach(did, node)
    return did


def test_tracked_device_not_immediately_detached(mock_world):
    make_node(mock_world.client, "node0")
    syncer = make_syncer(mock_world, grace=100.0)
    attach_one(mock_world)
    syncer.sync()
    assert len(syncer.missing_devices) == 1
    assert mock_world.client.list(ComposableResource) == []


def test_detach_cr_created_after_grace(mock_world):
    make_node(mock_world.client, "node0")
    syncer = make_syncer(mock_world, grace=0.0)
    did = attach_one(mock_world)
    syncer.sync()  # starts tracking
    import time

    time.sleep(0.01)
    syncer.sync()  # grace (0) exceeded → creates detach CR
    crs = mock_world.client.list(ComposableResource)
    assert len(crs) == 1
    cr = crs[0]
    assert cr.metadata.labels[READY_TO_DETACH_LABEL] == did
    assert READY_TO_DETACH_CDI_LABEL in cr.metadata.labels
    assert cr.spec.target_node == "node0"
    assert did not in syncer.missing_devices


def test_local_cr_appearing_stops_tracking(mock_world):
    make_node(mock_world.client, "node0")
    syncer = make_syncer(mock_world, grace=100.0)
    did = attach_one(mock_world)
    syncer.sync()
    assert did in syncer.missing_devices
    r = make_resource("gpu-1")
    mock_world.client.create(r)
    got = mock_world.client.get(ComposableResource, "gpu-1")
    got.status.device_id = did
    mock_world.client.update_status(got)
    syncer.sync()
    assert did not in syncer.missing_devices


def test_device_vanishing_upstream_stops_tracking(mock_world):
    make_node(mock_world.client, "node0")
    syncer = make_syncer(mock_world, grace=100.0)
    did = attach_one(mock_world)
    syncer.sync()
    assert did in syncer.missing_devices
    mock_world.fabric._pool[did].attached_node = ""  # fabric let it go
    syncer.sync()
    assert did not in syncer.missing_devices


def test_detach_cr_drives_full_physical_detach(mock_world):
    """End-to-end: syncer CR → resource controller walks it through
    bookkeeping and physically detaches the device."""
    make_node(mock_world.client, "node0")
    syncer = make_syncer(mock_world, grace=0.0)
    did = attach_one(mock_world)
    mock_world.ops.visible.setdefault("node0", set()).add(did)  # node sees it
    syncer.sync()
    import time

    time.sleep(0.01)
    syncer.sync()
    cr = mock_world.client.list(ComposableResource)[0]
    name = cr.metadata.name
    rec = mock_world.resource_rec
    rec.reconcile(name)  # None → Attaching (+ device id from label)
    rec.reconcile(name)  # Attaching → Online (already visible)
    rec.reconcile(name)  # Online + label → delete issued
    rec.reconcile(name)  # Online + deletionTimestamp → Detaching
    rec.reconcile(name)  # Detaching → Deleting (drained + fabric detach)
    rec.reconcile(name)  # finalizer off → gone
    assert mock_world.client.list(ComposableResource) == []
    assert mock_world.fabric.attached_to("node0") == []


def test_abandoned_async_compose_repaired(mock_world):
    """A CR deleted while its async compose is in flight leaks the device
    on the fabric (nobody records the identity); once the compose lands,
    get_resources must surface it so the syncer repairs the leak."""
    import time

    from cro_amd.api.v1alpha1.types import ComposabilityRequest
    from cro_amd.fabric.mock import MockFabricConfig
    from tests.conftest import drive, make_request

    w = mock_world
    make_node(w.client, "node0")
    w.fabric.config.asynchronous = True
    w.fabric.config.attach_latency = 0.05
    w.ops.set_driver("node0", True)

    w.client.create(make_request("r1", size=1, target_node="node0"))
    drive(w.request_rec, "r1", n=3)
    req = w.client.get(ComposabilityRequest, "r1")
    child = next(iter(req.status.resources))
    w.resource_rec.reconcile(child)  # Attaching entry
    w.resource_rec.reconcile(child)  # starts the compose → Waiting
    assert w.fabric.attached_to("node0")  # fabric holds it mid-compose

    # abandon: delete the request, drain everything before the compose lands
    w.client.delete(ComposabilityRequest, "r1")
    for _ in range(10):
        drive(w.request_rec, "r1", n=2)
        drive(w.resource_rec, child, n=2)
        drive(w.request_rec, child, n=2)
    assert w.client.list(ComposableResource) == []
    assert w.fabric.attached_to("node0")  # leaked

    time.sleep(0.06)  # compose lands fabric-side
    # on hardware the composed device is now physically enumerable on the
    # node; the mock node-ops needs that told explicitly
    w.ops.fabric_composed("node0", w.fabric.attached_to("node0")[0])
    syncer = make_syncer(w, grace=0.0)
    syncer.sync()
    time.sleep(0.01)
    syncer.sync()  # grace exceeded → detach CR
    crs = w.client.list(ComposableResource)
    assert len(crs) == 1
    for _ in range(10):
        drive(w.resource_rec, crs[0].metadata.name, n=3)
        drive(w.request_rec, crs[0].metadata.name, n=2)
    assert w.fabric.attached_to("node0") == []  # repaired
