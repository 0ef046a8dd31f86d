"""Conformance: the full manager driving the FTI FM backend over HTTP.

This is the reference's primary test shape (controller + envtest + fake
fabric TLS server, composableresource_controller_test.go:6028ff): requests
flow through watch-driven reconciles, the fabric is an HTTP fake speaking
the real FM protocol, and the node path is the mock."""

import pytest

from cro_amd.api.v1alpha1.types import ComposabilityRequest, Node
from cro_amd.controllers import build_manager
from cro_amd.fabric.adapter import Adapter
from cro_amd.fabric.fti.fm import FTIFMClient
from cro_amd.fabric.fti.token import CachedToken
from cro_amd.nodeops.amdgpu import MockNodeOps
from tests.conftest import make_request
from tests.fakes import FakeFTIServer

MACHINE_UUID = "99999999-aaaa-bbbb-cccc-dddddddddddd"


@pytest.fixture
def fm_stack():
    server = FakeFTIServer()
    # FM owns a machine whose scale-up always succeeds with a fresh serial
    counter = {"n": 0}
    attached = []

    orig_handler = server.handler

    def handler(request):
        if request.method == "PATCH" and "fabric_manager" in request.url.path:
            counter["n"] += 1
            serial = f"GPU-fm-{counter['n']}"
            attached.append(serial)
            server.fm_scaleup_response = server.fm_machine(
                resources=[server.fm_resource(serial)]
            )
            # keep GET machine info consistent for detach existence checks
            server.fm_machines[MACHINE_UUID] = server.fm_machine(
                resources=[server.fm_resource(s) for s in attached]
            )
        if request.method == "DELETE" and "fabric_manager" in request.url.path:
            import json as _json

            body = _json.loads(request.content)
            res_uuid = body["tenants"]["machines"][0]["resources"][0]["res_specs"][0][
                "res_uuid"
            ]
            serial = res_uuid.replace("res-", "")
            if serial in attached:
                attached.remove(serial)
            server.fm_machines[MACHINE_UUID] = server.fm_machine(
                resources=[server.fm_resource(s) for s in attached]
            )
        return orig_handler(request)

    import httpx

    transport = httpx.MockTransport(handler)
    creds = lambda: {"username": "u", "password": "p", "client_id": "c", "client_secret": "s", "realm": "r"}  # noqa: E731

    server.fm_machines[MACHINE_UUID] = server.fm_machine(resources=[])

    mgr = build_manager(Adapter("DRA", None), None)
    provider = FTIFMClient(
        mgr.client,
        endpoint="fabric.example",
        tenant_id="tenant-1",
        cluster_id="",
        token=CachedToken("fabric.example", credentials=creds, transport=transport),
        transport=transport,
    )
    mgr.resource_reconciler.adapter = Adapter("DRA", provider)
    ops = MockNodeOps(client=mgr.client)
    mgr.resource_reconciler.node_ops = ops

    node = Node()
    node.metadata.name = "node0"
    node.status.provider_id = f"fsas-cdi://{MACHINE_UUID}"
    mgr.client.create(node)
    ops.set_driver("node0", True)

    # bridge: FM attach makes the device fabric-composed on the mock node
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
    s.mgr, s.server, s.ops, s.attached = mgr, server, ops, attached
    yield s
    mgr.stop()


def test_fm_end_to_end_lifecycle(fm_stack):
    mgr = fm_stack.mgr
    mgr.client.create(make_request("r1", size=2, target_node="node0"))
    assert mgr.wait_for(
        lambda: (
            (req := mgr.client.try_get(ComposabilityRequest, "r1")) is not None
            and req.status.state == "Running"
        ),
        timeout=15,
    )
    req = mgr.client.get(ComposabilityRequest, "r1")
    ids = {v.device_id for v in req.status.resources.values()}
    assert len(ids) == 2 and all(i.startswith("GPU-fm-") for i in ids)
    assert all(
        v.cdi_device_id.startswith("res-GPU-fm-") for v in req.status.resources.values()
    )
    assert len(fm_stack.attached) == 2

    mgr.client.delete(ComposabilityRequest, "r1")
    assert mgr.wait_for(
        lambda: mgr.client.try_get(ComposabilityRequest, "r1") is None, timeout=15
    )
    assert fm_stack.attached == []  # FM DELETEs issued for both devices
    assert fm_stack.server.token_calls == 1  # token cache held across all calls
