"""Conformance: FM backend + DEVICE_PLUGIN mode — the reference suite's
second matrix block (composableresource_controller_test.go:6028ff): direct
visibility checks instead of ResourceSlices, whole-node load semantics, and
device-plugin/metrics daemonset rolling restarts on attach and detach."""

import json

import httpx
import pytest

from cro_amd.api.v1alpha1.types import (
    ComposabilityRequest,
    DaemonSet,
    Node,
)
from cro_amd.controllers import build_manager
from cro_amd.fabric.adapter import Adapter
from cro_amd.fabric.fti.fm import FTIFMClient
from cro_amd.fabric.fti.token import CachedToken
from cro_amd.nodeops.amdgpu import MockNodeOps
from cro_amd.nodeops.nodes import RESTARTED_AT_ANNOTATION
from tests.conftest import make_request
from tests.fakes import FakeFTIServer

MACHINE_UUID = "77777777-aaaa-bbbb-cccc-dddddddddddd"


@pytest.fixture
def dp_stack():
    server = FakeFTIServer()
    attached = []
    counter = {"n": 0}

    orig_handler = server.handler

    def handler(request):
        if "fabric_manager" in request.url.path and request.method == "PATCH":
            counter["n"] += 1
            serial = f"GPU-dp-{counter['n']}"
            attached.append(serial)
            server.fm_scaleup_response = server.fm_machine(
                resources=[server.fm_resource(serial)]
            )
            server.fm_machines[MACHINE_UUID] = server.fm_machine(
                resources=[server.fm_resource(s) for s in attached]
            )
        if "fabric_manager" in request.url.path and request.method == "DELETE":
            body = json.loads(request.content)
            res_uuid = body["tenants"]["machines"][0]["resources"][0]["res_specs"][0]["res_uuid"]
            serial = res_uuid.replace("res-", "")
            if serial in attached:
                attached.remove(serial)
            server.fm_machines[MACHINE_UUID] = server.fm_machine(
                resources=[server.fm_resource(s) for s in attached]
            )
        return orig_handler(request)

    transport = httpx.MockTransport(handler)
    creds = lambda: {"username": "u", "password": "p", "client_id": "c", "client_secret": "s", "realm": "r"}  # noqa: E731
    server.fm_machines[MACHINE_UUID] = server.fm_machine(resources=[])

    mgr = build_manager(Adapter("DEVICE_PLUGIN", None), None)
    provider = FTIFMClient(
        mgr.client, endpoint="fabric.example", tenant_id="tenant-1", cluster_id="",
        token=CachedToken("fabric.example", credentials=creds, transport=transport),
        transport=transport,
    )
    mgr.resource_reconciler.adapter = Adapter("DEVICE_PLUGIN", provider)
    ops = MockNodeOps(client=mgr.client)
    mgr.resource_reconciler.node_ops = ops

    node = Node()
    node.metadata.name = "node0"
    node.status.provider_id = f"fsas-cdi://{MACHINE_UUID}"
    mgr.client.create(node)
    ops.set_driver("node0", True)
    for name in ("amd-device-plugin", "amd-metrics-exporter"):
        ds = DaemonSet()
        ds.metadata.name = f"amd-gpu-operator/{name}"
        ds.status.desired_number_scheduled = 1
        ds.status.number_ready = 1
        ds.status.current_number_scheduled = 1
        mgr.client.create(ds)

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
    s.mgr, s.ops, s.attached = mgr, ops, attached
    yield s
    mgr.stop()


def test_fm_device_plugin_lifecycle_with_daemonset_restarts(dp_stack):
    mgr = dp_stack.mgr
    mgr.client.create(make_request("r1", size=1, target_node="node0"))
    assert mgr.wait_for(
        lambda: (req := mgr.client.try_get(ComposabilityRequest, "r1")) is not None
        and req.status.state == "Running",
        timeout=15,
    )
    # attach rolled the plugin daemonsets (restartedAt stamped)
    for name in ("amd-device-plugin", "amd-metrics-exporter"):
        ds = mgr.client.get(DaemonSet, f"amd-gpu-operator/{name}")
        assert RESTARTED_AT_ANNOTATION in ds.spec.template_annotations

    # whole-node load on an unrelated device blocks detach in DP mode
    dp_stack.ops.add_load("node0", "GPU-unrelated")
    mgr.client.delete(ComposabilityRequest, "r1")
    import time

    time.sleep(0.3)
    assert mgr.client.try_get(ComposabilityRequest, "r1") is not None  # blocked
    dp_stack.ops.clear_loads("node0")
    assert mgr.wait_for(
        lambda: mgr.client.try_get(ComposabilityRequest, "r1") is None, timeout=15
    )
    assert dp_stack.attached == []  # FM released the device
