"""Conformance: the full manager driving the NEC CDIM backend — layout-apply
connect/disconnect with the eesv→eeio link walk, provisional GPU UUIDs, and
async in-progress polling, end to end."""

import pytest

from cro_amd.api.v1alpha1.types import ComposabilityRequest, Node
from cro_amd.controllers import build_manager
from cro_amd.fabric.adapter import Adapter
from cro_amd.fabric.nec import NECClient
from cro_amd.nodeops.amdgpu import MockNodeOps
from tests.conftest import make_request
from tests.fakes import FakeNECServer

NODE_ID = "nec-node-001"
PROVISIONAL = "GPU-aaaaaaaa-bbbb-cccc-dddd-eeeeeeeeeeee"


@pytest.fixture
def nec_stack(monkeypatch):
    monkeypatch.setenv("NEC_PROVISIONAL_GPU_UUID", PROVISIONAL)
    server = FakeNECServer()

    host = FakeNECServer.adapter(
        "host-adapter", "sourceFabricAdapter", "eesv",
        links=[{"type": "destinationFabricAdapter", "deviceID": "io-adapter"}],
    )
    io = FakeNECServer.adapter("io-adapter", "destinationFabricAdapter", "eeio")
    gpu = FakeNECServer.gpu("nec-gpu-1")
    server.resources = [host, io, gpu]
    server.nodes = [{"id": NODE_ID, "name": "node0", "resources": [host, io]}]

    # layout-apply mutates connectivity: connect links the gpu to eeio,
    # disconnect unlinks it (so re-attach finds it free again)
    orig_handler = server.handler

    def handler(request):
        resp = orig_handler(request)
        if request.url.path.endswith("/layout-apply") and request.method == "POST" \
                and resp.status_code == 200:
            import json as _json

            proc = _json.loads(request.content)["procedures"][0]
            if proc["operation"] == "connect":
                gpu["device"]["links"] = [
                    {"type": "eeio", "deviceID": proc["sourceDeviceID"]},
                    {"type": "destinationFabricAdapter", "deviceID": proc["sourceDeviceID"]},
                ]
            else:
                gpu["device"]["links"] = []
        return resp

    import httpx

    transport = httpx.MockTransport(handler)

    mgr = build_manager(Adapter("DRA", None), None)
    provider = NECClient(
        mgr.client, ip="10.0.0.1", layout_apply_port="8000",
        configuration_manager_port="8001", transport=transport, poll_interval=0.01,
    )
    mgr.resource_reconciler.adapter = Adapter("DRA", provider)
    ops = MockNodeOps(client=mgr.client)
    mgr.resource_reconciler.node_ops = ops

    node = Node()
    node.metadata.name = "node0"
    node.status.provider_id = NODE_ID
    mgr.client.create(node)
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
    s.mgr, s.server, s.gpu = mgr, server, gpu
    yield s
    mgr.stop()


def test_nec_end_to_end_lifecycle(nec_stack):
    mgr = nec_stack.mgr
    mgr.client.create(make_request("r1", size=1, target_node="node0", model="mi355x"))
    assert mgr.wait_for(
        lambda: (req := mgr.client.try_get(ComposabilityRequest, "r1")) is not None
        and req.status.state == "Running",
        timeout=15,
    ), mgr.client.get(ComposabilityRequest, "r1").status
    req = mgr.client.get(ComposabilityRequest, "r1")
    entry = next(iter(req.status.resources.values()))
    assert entry.device_id == PROVISIONAL  # CDIM exposes no GPU UUID
    assert entry.cdi_device_id == "nec-gpu-1"  # manager device id for detach
    connects = [c for c in nec_stack.server.layout_calls
                if c["procedures"][0]["operation"] == "connect"]
    assert len(connects) == 1

    mgr.client.delete(ComposabilityRequest, "r1")
    assert mgr.wait_for(
        lambda: mgr.client.try_get(ComposabilityRequest, "r1") is None, timeout=15
    )
    disconnects = [c for c in nec_stack.server.layout_calls
                   if c["procedures"][0]["operation"] == "disconnect"]
    assert len(disconnects) == 1
    assert nec_stack.gpu["device"]["links"] == []  # fabric-side released


def test_nec_in_progress_polling_through_manager(nec_stack):
    nec_stack.server.apply_status_script = ["IN_PROGRESS", "IN_PROGRESS", "COMPLETED"]
    mgr = nec_stack.mgr
    mgr.client.create(make_request("r2", size=1, target_node="node0", model="mi355x"))
    assert mgr.wait_for(
        lambda: (req := mgr.client.try_get(ComposabilityRequest, "r2")) is not None
        and req.status.state == "Running",
        timeout=20,
    )
    mgr.client.delete(ComposabilityRequest, "r2")
    assert mgr.wait_for(
        lambda: mgr.client.try_get(ComposabilityRequest, "r2") is None, timeout=20
    )
