"""Crash-resume matrix: a fresh manager over the same store must converge
from every intermediate state (the CR status IS the checkpoint —
SURVEY.md §5.4).  Each case hand-drives the machines to a waypoint with no
manager running, then starts one and asserts terminal convergence."""

import pytest

from cro_amd.api.v1alpha1.types import ComposabilityRequest, ComposableResource
from cro_amd.bench_harness import build_local_stack
from cro_amd.controllers import build_manager
from tests.conftest import make_request


def fresh_stack():
    return build_local_stack(node_name="node0", use_gpu=False)


def restart(stack):
    mgr2 = build_manager(
        stack.mgr.resource_reconciler.adapter, None, store=stack.mgr.store
    )
    mgr2.resource_reconciler.node_ops = stack.ops
    mgr2.start()
    return mgr2


def drive_request_to(stack, state: str, deleting: bool = False):
    """Hand-advance (no manager threads) until the request hits `state`."""
    stack.mgr.client.create(make_request("r1", size=1, target_node="node0"))
    req_rec = stack.mgr.request_reconciler
    res_rec = stack.mgr.resource_reconciler
    for _ in range(20):
        req = stack.mgr.client.get(ComposabilityRequest, "r1")
        if req.status.state == state and not deleting:
            return
        if req.status.state == state and req.metadata.deletionTimestamp:
            return
        req_rec.reconcile("r1")
        for child in stack.mgr.client.list(
            ComposableResource, {"app.kubernetes.io/managed-by": "r1"}
        ):
            res_rec.reconcile(child.metadata.name)
            req_rec.reconcile(child.metadata.name)  # dual-kind sync
        if deleting and stack.mgr.client.get(ComposabilityRequest, "r1").metadata.deletionTimestamp is None:
            if stack.mgr.client.get(ComposabilityRequest, "r1").status.state == state:
                stack.mgr.client.delete(ComposabilityRequest, "r1")
                return
    raise AssertionError(
        f"could not reach {state}: {stack.mgr.client.get(ComposabilityRequest, 'r1').status}"
    )


@pytest.mark.parametrize("waypoint", ["NodeAllocating", "Updating", "Running"])
def test_resume_converges_to_running(waypoint):
    stack = fresh_stack()
    drive_request_to(stack, waypoint)
    mgr2 = restart(stack)
    try:
        assert mgr2.wait_for(
            lambda: mgr2.client.get(ComposabilityRequest, "r1").status.state == "Running",
            timeout=15,
        ), mgr2.client.get(ComposabilityRequest, "r1").status
    finally:
        mgr2.stop()


@pytest.mark.parametrize("waypoint", ["Updating", "Running"])
def test_resume_completes_teardown(waypoint):
    stack = fresh_stack()
    drive_request_to(stack, waypoint, deleting=True)
    mgr2 = restart(stack)
    try:
        assert mgr2.wait_for(
            lambda: mgr2.client.try_get(ComposabilityRequest, "r1") is None, timeout=15
        )
        assert mgr2.wait_for(lambda: stack.fabric.attached_to("node0") == [], timeout=15)
    finally:
        mgr2.stop()


def test_resume_mid_attach_child():
    """Child stuck in Attaching WITH a fabric-assigned device id: resume must
    finish attach rather than re-requesting a new device."""
    stack = fresh_stack()
    stack.mgr.client.create(make_request("r1", size=1, target_node="node0"))
    req_rec = stack.mgr.request_reconciler
    res_rec = stack.mgr.resource_reconciler
    for _ in range(3):
        req_rec.reconcile("r1")
    child = stack.mgr.client.list(
        ComposableResource, {"app.kubernetes.io/managed-by": "r1"}
    )[0]
    res_rec.reconcile(child.metadata.name)  # None → Attaching
    # partial attach: fabric assigned the device, visibility not yet reached
    did, cdi = stack.fabric.add_resource(
        stack.mgr.client.get(ComposableResource, child.metadata.name)
    )
    got = stack.mgr.client.get(ComposableResource, child.metadata.name)
    got.status.device_id = did
    got.status.cdi_device_id = cdi
    stack.mgr.client.update_status(got)

    mgr2 = restart(stack)
    try:
        assert mgr2.wait_for(
            lambda: mgr2.client.get(ComposabilityRequest, "r1").status.state == "Running",
            timeout=15,
        )
        # exactly one device attached — the resumed attach adopted the
        # already-assigned device instead of leaking a second one
        assert len(stack.fabric.attached_to("node0")) == 1
        assert stack.fabric.attached_to("node0") == [did]
    finally:
        mgr2.stop()
