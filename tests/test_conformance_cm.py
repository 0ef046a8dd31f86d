"""Conformance: the full manager driving the FTI CM backend over HTTP —
the asynchronous resize protocol (CM+DRA matrix half of the reference's
suite, composableresource_controller_test.go:1008ff).

The fake CM models the fabric-side async lifecycle: a resize request marks a
device composing; a later GET shows it ADD_COMPLETE, which the client adopts.
"""

import json

import httpx
import pytest

from cro_amd.api.v1alpha1.types import ComposabilityRequest, Node
from cro_amd.controllers import build_manager
from cro_amd.fabric.adapter import Adapter
from cro_amd.fabric.fti.cm import FTICMClient
from cro_amd.fabric.fti.token import CachedToken
from cro_amd.nodeops.amdgpu import MockNodeOps
from tests.conftest import make_request
from tests.fakes import FakeFTIServer
from tests.test_fabric_fti import CREDS, MACHINE_UUID, seed_chain


@pytest.fixture
def cm_stack():
    server = FakeFTIServer()
    state = {"devices": [], "counter": 0, "pending": 0}

    def refresh_machine():
        server.cm_machines[MACHINE_UUID] = server.cm_machine(
            devices=[server.cm_device(d) for d in state["devices"]],
            device_count=len(state["devices"]),
        )

    refresh_machine()

    orig_handler = server.handler

    def handler(request: httpx.Request) -> httpx.Response:
        if request.url.path.endswith("/actions/resize"):
            body = json.loads(request.content)
            if "increase_resource_count" in body:
                state["pending"] += 1
                state["scaleups"] = state.get("scaleups", 0) + 1
            else:
                for d in body["remove_resources"]["devices"]:
                    if d in state["devices"]:
                        state["devices"].remove(d)
                refresh_machine()
            return httpx.Response(202, json={})
        if "cluster_manager" in request.url.path and state["pending"]:
            # async compose "lands" by the time of the next machine GET
            while state["pending"]:
                state["counter"] += 1
                state["devices"].append(f"GPU-cm-{state['counter']}")
                state["pending"] -= 1
            refresh_machine()
        return orig_handler(request)

    transport = httpx.MockTransport(handler)

    mgr = build_manager(Adapter("DRA", None), None)
    provider = FTICMClient(
        mgr.client,
        endpoint="fabric.example",
        tenant_id="tenant-1",
        cluster_id="cluster-1",
        token=CachedToken("fabric.example", credentials=CREDS, transport=transport),
        transport=transport,
    )
    mgr.resource_reconciler.adapter = Adapter("DRA", provider)
    ops = MockNodeOps(client=mgr.client)
    mgr.resource_reconciler.node_ops = ops
    seed_chain(mgr.client)
    ops.set_driver("node0", True)

    orig_add = provider.add_resource

    def add_resource(resource):
        did, cdi = orig_add(resource)
        ops.fabric_composed(resource.spec.target_node, did)
        return did, cdi

    provider.add_resource = add_resource
    mgr.start()

    class Stack:
        pass

    s = Stack()
    s.mgr, s.server, s.ops, s.state = mgr, server, ops, state
    yield s
    mgr.stop()


def test_cm_async_attach_lifecycle(cm_stack):
    mgr = cm_stack.mgr
    mgr.client.create(make_request("r1", size=2, target_node="node0"))
    assert mgr.wait_for(
        lambda: (req := mgr.client.try_get(ComposabilityRequest, "r1")) is not None
        and req.status.state == "Running",
        timeout=20,
    )
    req = mgr.client.get(ComposabilityRequest, "r1")
    ids = {v.device_id for v in req.status.resources.values()}
    assert len(ids) == 2 and all(i.startswith("GPU-cm-") for i in ids)
    # the CM path went through at least one WaitingDeviceAttaching cycle
    assert cm_stack.state.get("scaleups", 0) >= 2

    mgr.client.delete(ComposabilityRequest, "r1")
    assert mgr.wait_for(
        lambda: mgr.client.try_get(ComposabilityRequest, "r1") is None, timeout=20
    )
    assert cm_stack.state["devices"] == []  # fabric-side devices released
