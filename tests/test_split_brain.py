"""Split-brain safety: two full operator instances (leader election
misconfigured away) reconciling the SAME store must stay eventually
consistent — optimistic concurrency arbitrates writes, double-composed
fabric devices are repaired by the syncer, and teardown drains fully.

The reference relies on leader election alone; this proves the state
machines themselves survive a dual-writer accident."""

import time

from cro_amd.api.v1alpha1.types import ComposabilityRequest, ComposableResource, Node
from cro_amd.controllers import build_manager
from cro_amd.fabric.adapter import Adapter
from cro_amd.fabric.mock import MockFabric
from cro_amd.nodeops.amdgpu import MockNodeOps
from cro_amd.runtime.store import InMemoryStore
from tests.conftest import make_request


def test_dual_operator_churn_converges():
    store = InMemoryStore()
    fabric = MockFabric(models={"mi355x": 8})

    managers = []
    for i in range(2):
        mgr = build_manager(
            Adapter("DRA", fabric),
            None,
            store=store,
            enable_webhook=(i == 0),  # admission registers once
            syncer_period=0.3,
            syncer_grace=0.5,
        )
        ops = MockNodeOps(client=mgr.client)
        mgr.resource_reconciler.node_ops = ops
        mgr.syncer.node_ops = ops
        managers.append((mgr, ops))

    # both operators' node-ops must observe composition (they share the node)
    orig_add = fabric.add_resource

    def add(resource):
        did, cdi = orig_add(resource)
        for _, ops in managers:
            ops.fabric_composed(resource.spec.target_node, did)
        return did, cdi

    fabric.add_resource = add

    client = managers[0][0].client
    node = Node()
    node.metadata.name = "node0"
    client.create(node)
    for _, ops in managers:
        ops.set_driver("node0", True)

    for mgr, _ in managers:
        mgr.start()
    try:
        # churn: create/delete cycles with both operators racing
        for cycle in range(10):
            client.create(make_request(f"sb-{cycle}", size=2, target_node="node0"))
            deadline = time.monotonic() + 20
            while time.monotonic() < deadline:
                r = client.try_get(ComposabilityRequest, f"sb-{cycle}")
                if r is not None and r.status.state == "Running":
                    break
                time.sleep(0.01)
            r = client.try_get(ComposabilityRequest, f"sb-{cycle}")
            assert r is not None and r.status.state == "Running", (
                cycle,
                r.status if r else None,
            )
            client.delete(ComposabilityRequest, f"sb-{cycle}")
            deadline = time.monotonic() + 20
            while time.monotonic() < deadline:
                if client.try_get(ComposabilityRequest, f"sb-{cycle}") is None:
                    break
                time.sleep(0.01)
            assert client.try_get(ComposabilityRequest, f"sb-{cycle}") is None, cycle

        # quiescence: no leaked CRs; any double-composed device repaired
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            if (
                not client.list(ComposableResource)
                and fabric.attached_to("node0") == []
            ):
                break
            time.sleep(0.05)
        assert client.list(ComposableResource) == []
        assert fabric.attached_to("node0") == []
    finally:
        for mgr, _ in managers:
            mgr.stop()
