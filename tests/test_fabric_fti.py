"""FTI CM/FM backend protocol tests against the fake fabric server."""

import time

import pytest

from cro_amd.api.v1alpha1.types import Node
from cro_amd.fabric.base import FabricError, WaitingDeviceAttaching, WaitingDeviceDetaching
from cro_amd.fabric.fti.cm import FTICMClient
from cro_amd.fabric.fti.fm import FTIFMClient
from cro_amd.fabric.fti.machines import (
    BareMetalHost,
    Machine,
    MachineResolutionError,
    resolve_machine_id,
    resolve_machine_id_openshift,
)
from cro_amd.fabric.fti.token import CachedToken, TokenError
from tests.conftest import make_resource
from tests.fakes import FakeFTIServer, make_jwt

CREDS = lambda: {  # noqa: E731
    "username": "u", "password": "p", "client_id": "c",
    "client_secret": "s", "realm": "r",
}
MACHINE_UUID = "11111111-2222-3333-4444-555555555555"


def seed_chain(client, node="node0", machine_uuid=MACHINE_UUID):
    n = Node()
    n.metadata.name = node
    n.metadata.annotations["machine.openshift.io/machine"] = "openshift-machine-api/m0"
    client.create(n)
    m = Machine()
    m.metadata.name = "openshift-machine-api/m0"
    m.metadata.annotations["metal3.io/BareMetalHost"] = "openshift-machine-api/bmh0"
    client.create(m)
    b = BareMetalHost()
    b.metadata.name = "openshift-machine-api/bmh0"
    b.metadata.annotations["cluster-manager.cdi.io/machine"] = machine_uuid
    client.create(b)


def cm_client(client, server):
    return FTICMClient(
        client,
        endpoint="fabric.example",
        tenant_id="tenant-1",
        cluster_id="cluster-1",
        token=CachedToken("fabric.example", credentials=CREDS, transport=server.transport()),
        transport=server.transport(),
    )


def fm_client(client, server, cluster_id="cluster-1"):
    return FTIFMClient(
        client,
        endpoint="fabric.example",
        tenant_id="tenant-1",
        cluster_id=cluster_id,
        token=CachedToken("fabric.example", credentials=CREDS, transport=server.transport()),
        transport=server.transport(),
    )


# -- token cache ------------------------------------------------------------


def test_token_fetch_and_cache():
    server = FakeFTIServer()
    tok = CachedToken("fabric.example", credentials=CREDS, transport=server.transport())
    t1 = tok.get_token()
    t2 = tok.get_token()
    assert t1 == t2
    assert server.token_calls == 1


def test_token_refresh_when_near_expiry():
    server = FakeFTIServer()
    server.token_exp = time.time() + 10  # inside the 30 s leeway
    tok = CachedToken("fabric.example", credentials=CREDS, transport=server.transport())
    tok.get_token()
    tok.get_token()
    assert server.token_calls == 2  # never served from cache


def test_token_bad_credentials():
    server = FakeFTIServer()
    server.token_persona = "bad-creds"
    tok = CachedToken("fabric.example", credentials=CREDS, transport=server.transport())
    with pytest.raises(TokenError, match="401"):
        tok.get_token()


def test_token_non_json_body():
    server = FakeFTIServer()
    server.token_persona = "non-json"
    tok = CachedToken("fabric.example", credentials=CREDS, transport=server.transport())
    with pytest.raises(TokenError):
        tok.get_token()


def test_token_malformed_jwt():
    server = FakeFTIServer()
    server.token_persona = "malformed-jwt"
    tok = CachedToken("fabric.example", credentials=CREDS, transport=server.transport())
    with pytest.raises(TokenError, match="invalid access token"):
        tok.get_token()


# -- machine resolution ------------------------------------------------------


def test_openshift_chain_resolution(client):
    seed_chain(client)
    assert resolve_machine_id_openshift(client, "node0") == MACHINE_UUID


def test_chain_missing_annotation(client):
    n = Node()
    n.metadata.name = "bare"
    client.create(n)
    with pytest.raises(MachineResolutionError, match="machine.openshift.io/machine"):
        resolve_machine_id_openshift(client, "bare")


def test_rke2_provider_id(client):
    n = Node()
    n.metadata.name = "rke2-node"
    n.status.provider_id = f"fsas-cdi://{MACHINE_UUID}"
    client.create(n)
    assert resolve_machine_id(client, "rke2-node", "") == MACHINE_UUID
    with pytest.raises(MachineResolutionError):
        n2 = Node()
        n2.metadata.name = "bad"
        n2.status.provider_id = "aws://i-123"
        client.create(n2)
        resolve_machine_id(client, "bad", "")


# -- CM ----------------------------------------------------------------------


def test_cm_adopts_unused_complete_device(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.cm_machines[MACHINE_UUID] = server.cm_machine(
        devices=[server.cm_device("GPU-free")]
    )
    c = cm_client(client, server)
    did, cdi = c.add_resource(make_resource("gpu-1"))
    assert did == "GPU-free"
    assert cdi == "res-GPU-free"
    assert server.resize_calls == []  # no resize needed


def test_cm_resize_when_no_free_device(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.cm_machines[MACHINE_UUID] = server.cm_machine(devices=[], device_count=0)
    c = cm_client(client, server)
    with pytest.raises(WaitingDeviceAttaching):
        c.add_resource(make_resource("gpu-1"))
    machine_id, body = server.resize_calls[0]
    assert machine_id == MACHINE_UUID
    assert body == {"increase_resource_count": {"spec_uuid": "spec-1", "device_count": 1}}


def test_cm_add_failed_device_raises(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.cm_machines[MACHINE_UUID] = server.cm_machine(
        devices=[server.cm_device("GPU-bad", status="ADD_FAILED", reason="psu fault")]
    )
    c = cm_client(client, server)
    with pytest.raises(FabricError, match="psu fault"):
        c.add_resource(make_resource("gpu-1"))


def test_cm_skips_devices_owned_by_other_crs(client):
    server = FakeFTIServer()
    seed_chain(client)
    owner = make_resource("gpu-owner")
    client.create(owner)
    got = client.get(type(owner), "gpu-owner")
    got.status.device_id = "GPU-used"
    client.update_status(got)
    server.cm_machines[MACHINE_UUID] = server.cm_machine(
        devices=[server.cm_device("GPU-used"), server.cm_device("GPU-free2")]
    )
    c = cm_client(client, server)
    did, _ = c.add_resource(make_resource("gpu-1"))
    assert did == "GPU-free2"


def test_cm_remove_issues_scaledown(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.cm_machines[MACHINE_UUID] = server.cm_machine(
        devices=[server.cm_device("GPU-x")], device_count=1
    )
    c = cm_client(client, server)
    r = make_resource("gpu-1")
    r.status.device_id = "GPU-x"
    with pytest.raises(WaitingDeviceDetaching):
        c.remove_resource(r)
    _, body = server.resize_calls[0]
    assert body == {
        "remove_resources": {"spec_uuid": "spec-1", "device_count": 0, "devices": ["GPU-x"]}
    }


def test_cm_remove_unknown_device_is_noop(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.cm_machines[MACHINE_UUID] = server.cm_machine(devices=[])
    c = cm_client(client, server)
    r = make_resource("gpu-1")
    r.status.device_id = "GPU-ghost"
    c.remove_resource(r)  # no exception, no resize
    assert server.resize_calls == []


def test_cm_health_digits(client):
    server = FakeFTIServer()
    seed_chain(client)
    c = cm_client(client, server)
    r = make_resource("gpu-1")
    r.status.device_id = "GPU-x"
    for op_status, expectation in (
        ("0", None),
        ("020", None),  # only the first digit matters
        ("1", "Warning"),
        ("2", "Critical"),
        ("9", "unknown status"),
    ):
        server.cm_machines[MACHINE_UUID] = server.cm_machine(
            devices=[server.cm_device("GPU-x", op_status=op_status)]
        )
        if expectation is None:
            c.check_resource(r)
        else:
            with pytest.raises(FabricError, match=expectation):
                c.check_resource(r)


def test_cm_check_missing_device(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.cm_machines[MACHINE_UUID] = server.cm_machine(devices=[])
    c = cm_client(client, server)
    r = make_resource("gpu-1")
    r.status.device_id = "GPU-gone"
    with pytest.raises(FabricError, match="cannot be found"):
        c.check_resource(r)


def test_cm_get_resources(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.cm_machines[MACHINE_UUID] = server.cm_machine(
        devices=[server.cm_device("GPU-a"), server.cm_device("GPU-b")]
    )
    c = cm_client(client, server)
    infos = c.get_resources()
    assert {i.device_id for i in infos} == {"GPU-a", "GPU-b"}
    assert all(i.node_name == "node0" for i in infos)


# -- FM ----------------------------------------------------------------------


def test_fm_attach_synchronous_success(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.fm_scaleup_response = server.fm_machine(
        resources=[server.fm_resource("GPU-new")]
    )
    c = fm_client(client, server)
    did, cdi = c.add_resource(make_resource("gpu-1"))
    assert did == "GPU-new"
    assert cdi == "res-GPU-new"
    method, machine_id, body = server.fm_update_calls[0]
    assert (method, machine_id) == ("PATCH", MACHINE_UUID)
    spec = body["tenants"]["machines"][0]["resources"][0]["res_specs"][0]
    assert spec["res_type"] == "gpu" and spec["res_num"] == 1
    assert spec["res_spec"]["condition"][0] == {
        "column": "model", "operator": "eq", "value": "mi355x",
    }


def test_fm_attach_warning_accepted_critical_fails(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.fm_scaleup_response = server.fm_machine(
        resources=[server.fm_resource("GPU-warn", op_status="1")]
    )
    c = fm_client(client, server)
    did, _ = c.add_resource(make_resource("gpu-1"))
    assert did == "GPU-warn"
    server.fm_scaleup_response = server.fm_machine(
        resources=[server.fm_resource("GPU-crit", op_status="2")]
    )
    with pytest.raises(FabricError, match="Critical"):
        c.add_resource(make_resource("gpu-2"))


def test_fm_attach_http_error_surfaces_detail(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.fm_scaleup_status = 500
    c = fm_client(client, server)
    with pytest.raises(FabricError, match="scaleup failed"):
        c.add_resource(make_resource("gpu-1"))


def test_fm_detach_skips_when_gone(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.fm_machines[MACHINE_UUID] = server.fm_machine(resources=[])
    c = fm_client(client, server)
    r = make_resource("gpu-1")
    r.status.cdi_device_id = "res-GPU-x"
    c.remove_resource(r)
    assert server.fm_update_calls == []  # idempotent skip (fm/client.go:231-242)


def test_fm_detach_issues_delete(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.fm_machines[MACHINE_UUID] = server.fm_machine(
        resources=[server.fm_resource("GPU-x")]
    )
    c = fm_client(client, server)
    r = make_resource("gpu-1")
    r.status.device_id = "GPU-x"
    r.status.cdi_device_id = "res-GPU-x"
    c.remove_resource(r)
    method, machine_id, body = server.fm_update_calls[0]
    assert method == "DELETE"
    spec = body["tenants"]["machines"][0]["resources"][0]["res_specs"][0]
    assert spec == {"res_type": "gpu", "res_uuid": "res-GPU-x", "res_num": 1}


def test_fm_health_and_missing(client):
    server = FakeFTIServer()
    seed_chain(client)
    c = fm_client(client, server)
    r = make_resource("gpu-1")
    r.status.device_id = "GPU-x"
    server.fm_machines[MACHINE_UUID] = server.fm_machine(
        resources=[server.fm_resource("GPU-x", op_status="0")]
    )
    c.check_resource(r)
    server.fm_machines[MACHINE_UUID] = server.fm_machine(
        resources=[server.fm_resource("GPU-x", op_status="2")]
    )
    with pytest.raises(FabricError, match="Critical"):
        c.check_resource(r)
    server.fm_machines[MACHINE_UUID] = server.fm_machine(resources=[])
    with pytest.raises(FabricError, match="cannot be found"):
        c.check_resource(r)


def test_fm_get_resources_with_model(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.fm_machines[MACHINE_UUID] = server.fm_machine(
        resources=[server.fm_resource("GPU-a", model="mi355x")]
    )
    c = fm_client(client, server)
    infos = c.get_resources()
    assert len(infos) == 1
    assert infos[0].model == "mi355x"
    assert infos[0].device_id == "GPU-a"
    assert infos[0].cdi_device_id == "res-GPU-a"


def test_fm_rke2_machine_resolution(client):
    server = FakeFTIServer()
    n = Node()
    n.metadata.name = "rke2-node"
    n.status.provider_id = f"fsas-cdi://{MACHINE_UUID}"
    client.create(n)
    server.fm_scaleup_response = server.fm_machine(
        resources=[server.fm_resource("GPU-r")]
    )
    c = fm_client(client, server, cluster_id="")  # RKE2: no cluster id
    did, _ = c.add_resource(make_resource("gpu-1", target_node="rke2-node"))
    assert did == "GPU-r"
