"""Fabric error personas through the FULL manager (the reference's
scenario-encoded machine UUIDs, composableresource_controller_test.go:792-997):
every failure must surface in status and the system must keep converging."""

import httpx
import pytest

from cro_amd.api.v1alpha1.types import ComposabilityRequest, Node
from cro_amd.controllers import build_manager
from cro_amd.fabric.adapter import Adapter
from cro_amd.fabric.fti.fm import FTIFMClient
from cro_amd.fabric.fti.token import CachedToken
from cro_amd.nodeops.amdgpu import MockNodeOps
from tests.conftest import make_request
from tests.fakes import FakeFTIServer

MACHINE_UUID = "55555555-aaaa-bbbb-cccc-dddddddddddd"
CREDS = lambda: {"username": "u", "password": "p", "client_id": "c", "client_secret": "s", "realm": "r"}  # noqa: E731


def build_stack(server: FakeFTIServer):
    transport = httpx.MockTransport(server.handler)
    mgr = build_manager(Adapter("DRA", None), None)
    provider = FTIFMClient(
        mgr.client, endpoint="fabric.example", tenant_id="tenant-1", cluster_id="",
        token=CachedToken("fabric.example", credentials=CREDS, transport=transport),
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

    orig_add = provider.add_resource

    def add_resource(resource):
        did, cdi = orig_add(resource)
        ops.fabric_composed(resource.spec.target_node, did)
        return did, cdi

    provider.add_resource = add_resource
    return mgr, ops


def test_fm_scaleup_500_surfaces_and_recovers():
    server = FakeFTIServer()
    server.fm_machines[MACHINE_UUID] = server.fm_machine(resources=[])
    server.fm_scaleup_status = 500
    mgr, ops = build_stack(server)
    mgr.start()
    try:
        mgr.client.create(make_request("r1", size=1, target_node="node0"))
        # the failure reaches the request's per-device status map
        assert mgr.wait_for(
            lambda: any(
                "scaleup failed" in (v.error or "")
                for v in mgr.client.get(ComposabilityRequest, "r1").status.resources.values()
            ),
            timeout=10,
        )
        # fabric heals → request converges
        server.fm_scaleup_status = 200
        server.fm_scaleup_response = server.fm_machine(
            resources=[server.fm_resource("GPU-healed")]
        )
        assert mgr.wait_for(
            lambda: mgr.client.get(ComposabilityRequest, "r1").status.state == "Running",
            timeout=15,
        )
    finally:
        mgr.stop()


def test_fm_critical_attach_device_rejected_then_recovers():
    server = FakeFTIServer()
    server.fm_machines[MACHINE_UUID] = server.fm_machine(resources=[])
    server.fm_scaleup_response = server.fm_machine(
        resources=[server.fm_resource("GPU-crit", op_status="2")]
    )
    mgr, ops = build_stack(server)
    mgr.start()
    try:
        mgr.client.create(make_request("r1", size=1, target_node="node0"))
        assert mgr.wait_for(
            lambda: any(
                "Critical" in (v.error or "")
                for v in mgr.client.get(ComposabilityRequest, "r1").status.resources.values()
            ),
            timeout=10,
        )
        server.fm_scaleup_response = server.fm_machine(
            resources=[server.fm_resource("GPU-good")]
        )
        assert mgr.wait_for(
            lambda: mgr.client.get(ComposabilityRequest, "r1").status.state == "Running",
            timeout=15,
        )
    finally:
        mgr.stop()


def test_token_failure_blocks_then_recovers():
    server = FakeFTIServer()
    server.fm_machines[MACHINE_UUID] = server.fm_machine(resources=[])
    server.token_persona = "bad-creds"
    server.fm_scaleup_response = server.fm_machine(
        resources=[server.fm_resource("GPU-x")]
    )
    mgr, ops = build_stack(server)
    mgr.start()
    try:
        mgr.client.create(make_request("r1", size=1, target_node="node0"))
        assert mgr.wait_for(
            lambda: any(
                "401" in (v.error or "")
                for v in mgr.client.get(ComposabilityRequest, "r1").status.resources.values()
            ),
            timeout=10,
        )
        server.token_persona = "ok"
        assert mgr.wait_for(
            lambda: mgr.client.get(ComposabilityRequest, "r1").status.state == "Running",
            timeout=15,
        )
    finally:
        mgr.stop()


def test_online_health_critical_surfaces_without_detach():
    server = FakeFTIServer()
    server.fm_machines[MACHINE_UUID] = server.fm_machine(resources=[])
    server.fm_scaleup_response = server.fm_machine(
        resources=[server.fm_resource("GPU-x")]
    )
    mgr, ops = build_stack(server)
    mgr.resource_reconciler.config.online_health_period = 0.05
    mgr.start()
    try:
        mgr.client.create(make_request("r1", size=1, target_node="node0"))
        assert mgr.wait_for(
            lambda: mgr.client.get(ComposabilityRequest, "r1").status.state == "Running",
            timeout=15,
        )
        # flip the machine's view of the device to Critical
        server.fm_machines[MACHINE_UUID] = server.fm_machine(
            resources=[server.fm_resource("GPU-x", op_status="2")]
        )
        assert mgr.wait_for(
            lambda: any(
                "Critical" in (v.error or "")
                for v in mgr.client.get(ComposabilityRequest, "r1").status.resources.values()
            ),
            timeout=10,
        )
        # device stays Online (health errors surface, they do not detach —
        # composableresource_controller.go:317-331)
        req = mgr.client.get(ComposabilityRequest, "r1")
        assert all(v.state == "Online" for v in req.status.resources.values())
    finally:
        mgr.stop()
