"""NEC CDIM and Sunfish backend protocol tests."""

import json

import httpx
import pytest

from cro_amd.api.v1alpha1.types import Node
from cro_amd.fabric.base import FabricError, WaitingDeviceAttaching, WaitingDeviceDetaching
from cro_amd.fabric.nec import NECClient, build_endpoint
from cro_amd.fabric.sunfish import SunfishClient
from tests.conftest import make_resource
from tests.fakes import FakeNECServer

NODE_ID = "nec-node-001"


@pytest.fixture(autouse=True)
def nec_env(monkeypatch):
    monkeypatch.setenv("NEC_PROVISIONAL_GPU_UUID", "GPU-aaaaaaaa-bbbb-cccc-dddd-eeeeeeeeeeee")


def nec_client(client, server):
    return NECClient(
        client,
        ip="10.0.0.1",
        layout_apply_port="8000",
        configuration_manager_port="8001",
        transport=server.transport(),
        poll_interval=0.01,
    )


def seed_nec_world(client, server):
    """One node with the eesv→eeio adapter chain and one free healthy GPU."""
    n = Node()
    n.metadata.name = "node0"
    n.status.provider_id = NODE_ID
    client.create(n)

    host = FakeNECServer.adapter(
        "host-adapter", "sourceFabricAdapter", "eesv",
        links=[{"type": "destinationFabricAdapter", "deviceID": "io-adapter"}],
    )
    io = FakeNECServer.adapter("io-adapter", "destinationFabricAdapter", "eeio")
    gpu = FakeNECServer.gpu("nec-gpu-1")
    server.resources = [host, io, gpu]
    server.nodes = [{"id": NODE_ID, "name": "node0", "resources": [host, io]}]
    return gpu


def test_build_endpoint():
    assert build_endpoint("10.0.0.1", "8000") == "http://10.0.0.1:8000/cdim/api/v1"
    with pytest.raises(ValueError):
        build_endpoint("", "8000")


def test_nec_attach_connect_flow(client):
    server = FakeNECServer()
    seed_nec_world(client, server)
    c = nec_client(client, server)
    did, cdi = c.add_resource(make_resource("gpu-1"))
    assert did == "GPU-aaaaaaaa-bbbb-cccc-dddd-eeeeeeeeeeee"  # provisional
    assert cdi == "nec-gpu-1"
    procedure = server.layout_calls[0]["procedures"][0]
    assert procedure["operation"] == "connect"
    assert procedure["sourceDeviceID"] == "io-adapter"
    assert procedure["destinationDeviceID"] == "nec-gpu-1"


def test_nec_attach_no_free_gpu(client):
    server = FakeNECServer()
    gpu = seed_nec_world(client, server)
    gpu["device"]["links"] = [{"type": "eeio", "deviceID": "io-adapter"}]  # taken
    c = nec_client(client, server)
    with pytest.raises(FabricError, match="no available GPU"):
        c.add_resource(make_resource("gpu-1"))


def test_nec_attach_unhealthy_gpu_skipped(client):
    server = FakeNECServer()
    gpu = seed_nec_world(client, server)
    gpu["device"]["status"]["health"] = "Critical"
    c = nec_client(client, server)
    with pytest.raises(FabricError, match="no available GPU"):
        c.add_resource(make_resource("gpu-1"))


def test_nec_attach_in_progress_then_completed(client):
    server = FakeNECServer()
    seed_nec_world(client, server)
    server.apply_status_script = ["IN_PROGRESS", "IN_PROGRESS", "COMPLETED"]
    c = nec_client(client, server)
    did, _ = c.add_resource(make_resource("gpu-1"))
    assert did.startswith("GPU-")


def test_nec_attach_conflict_maps_to_waiting(client):
    server = FakeNECServer()
    seed_nec_world(client, server)
    server.post_conflict = True
    c = nec_client(client, server)
    with pytest.raises(WaitingDeviceAttaching):
        c.add_resource(make_resource("gpu-1"))


def test_nec_attach_apply_failed(client):
    server = FakeNECServer()
    seed_nec_world(client, server)
    server.apply_status_script = ["FAILED"]
    c = nec_client(client, server)
    with pytest.raises(FabricError, match="layout-apply failed"):
        c.add_resource(make_resource("gpu-1"))


def test_nec_attach_timeout_raises_waiting(client):
    server = FakeNECServer()
    seed_nec_world(client, server)
    server.apply_status_script = ["IN_PROGRESS"] * 10
    c = nec_client(client, server)
    with pytest.raises(WaitingDeviceAttaching):
        c.add_resource(make_resource("gpu-1"))


def test_nec_detach_disconnect_flow(client):
    server = FakeNECServer()
    gpu = seed_nec_world(client, server)
    gpu["device"]["links"] = [
        {"type": "destinationFabricAdapter", "deviceID": "io-adapter"}
    ]
    c = nec_client(client, server)
    r = make_resource("gpu-1")
    r.status.cdi_device_id = "nec-gpu-1"
    c.remove_resource(r)
    procedure = server.layout_calls[0]["procedures"][0]
    assert procedure["operation"] == "disconnect"
    assert procedure["sourceDeviceID"] == "io-adapter"
    assert procedure["destinationDeviceID"] == "nec-gpu-1"


def test_nec_detach_already_detached(client):
    server = FakeNECServer()
    seed_nec_world(client, server)  # gpu has no destinationFabricAdapter link
    c = nec_client(client, server)
    r = make_resource("gpu-1")
    r.status.cdi_device_id = "nec-gpu-1"
    c.remove_resource(r)  # no-op, no layout call
    assert server.layout_calls == []


def test_nec_health(client):
    server = FakeNECServer()
    gpu = seed_nec_world(client, server)
    c = nec_client(client, server)
    r = make_resource("gpu-1")
    r.status.cdi_device_id = "nec-gpu-1"
    c.check_resource(r)
    gpu["device"]["status"]["state"] = "Disabled"
    with pytest.raises(FabricError, match="not healthy"):
        c.check_resource(r)


def test_nec_get_resources(client):
    server = FakeNECServer()
    seed_nec_world(client, server)
    gpu_on_node = FakeNECServer.gpu("nec-gpu-attached")
    server.nodes[0]["resources"].append(gpu_on_node)
    c = nec_client(client, server)
    infos = c.get_resources()
    assert len(infos) == 1
    assert infos[0].cdi_device_id == "nec-gpu-attached"
    assert infos[0].node_name == "node0"
    assert infos[0].device_id.startswith("GPU-")  # provisional uuid


def test_nec_missing_provisional_uuid(client, monkeypatch):
    monkeypatch.delenv("NEC_PROVISIONAL_GPU_UUID")
    server = FakeNECServer()
    seed_nec_world(client, server)
    c = nec_client(client, server)
    with pytest.raises(FabricError, match="NEC_PROVISIONAL_GPU_UUID"):
        c.add_resource(make_resource("gpu-1"))


# -- Sunfish ----------------------------------------------------------------


def test_sunfish_attach_detach_shape():
    calls = []

    def handler(request: httpx.Request) -> httpx.Response:
        calls.append((request.method, request.url.path, json.loads(request.content)))
        return httpx.Response(204)

    c = SunfishClient(endpoint="sunfish.example:5060", transport=httpx.MockTransport(handler))
    r = make_resource("gpu-1", model="AMD-Instinct-MI355X", target_node="node0")
    did, cdi = c.add_resource(r)
    assert (did, cdi) == ("", "")
    c.remove_resource(r)
    assert calls[0][0] == "PATCH"
    assert calls[0][1] == "/redfish/v1/Systems/System"
    assert calls[0][2]["Name"] == "node0"
    member = calls[0][2]["Processors"]["Members"][0]
    assert member["@Redfish.RequestCount"] == 1
    assert member["ProcessorType"] == "GPU"
    assert calls[1][2]["Processors"]["Members"][0]["@Redfish.RequestCount"] == 0


def test_sunfish_error_status():
    def handler(request):
        return httpx.Response(500)

    c = SunfishClient(endpoint="s.example", transport=httpx.MockTransport(handler))
    with pytest.raises(FabricError, match="500"):
        c.add_resource(make_resource("gpu-1", model="AMD-Instinct-MI355X"))


def test_sunfish_noop_surfaces():
    c = SunfishClient(endpoint="s.example")
    assert c.check_resource(make_resource("gpu-1")) is None
    assert c.get_resources() == []
