"""The backend×mode×state×failure conformance matrix.

The reference enumerates per-state Describe blocks for the CM+DRA and
FM+DEVICE_PLUGIN matrices with injected status-update failures in every
state (composableresource_controller_test.go:1008, 6028, 9299;
suite_test.go:244-294).  This file closes the product systematically:

* Part A — full-manager lifecycle for EVERY {CM, FM, NEC} × {DRA,
  DEVICE_PLUGIN} combination (adds the previously missing
  CM+DEVICE_PLUGIN, FM+DRA, NEC+DEVICE_PLUGIN cells).  FM+DEVICE_PLUGIN
  uses the OpenShift metal3 chain (DEVICE_PLUGIN is forbidden on RKE2,
  composableresource_adapter.go:58-61 — enforced in new_adapter and
  tested in test_adapter.py), FM+DRA uses the RKE2 providerID path.
* Part B — hand-driven per-state write-failure injection: for each
  {mode} × {state} × {ApiError, Conflict}, the reconciler surfaces the
  failed write (workqueue retries with backoff) and converges on the
  retry.  The write seams are backend-independent (the same
  update/update_status calls run for every backend), so Part B uses the
  mock fabric; backend-side failure personas live in
  test_error_personas.py / test_fabric_edge_personas.py.
"""

import json

import httpx
import pytest

from cro_amd.api.v1alpha1.types import (
    ComposabilityRequest,
    ComposableResource,
    Node,
)
from cro_amd.controllers import build_manager
from cro_amd.controllers.composabilityrequest import ComposabilityRequestReconciler
from cro_amd.controllers.composableresource import (
    ComposableResourceReconciler,
    ReconcileConfig,
)
from cro_amd.fabric.adapter import Adapter
from cro_amd.fabric.mock import MockFabric
from cro_amd.nodeops.amdgpu import MockNodeOps
from cro_amd.runtime.client import Client
from cro_amd.runtime.errors import ApiError, ConflictError
from cro_amd.runtime.store import InMemoryStore
from tests.conftest import make_node, make_request
from tests.fakes import FakeFTIServer, FakeNECServer
from tests.test_fabric_fti import CREDS, MACHINE_UUID, seed_chain

MODES = ("DRA", "DEVICE_PLUGIN")


# -- Part A: backend × mode lifecycle ---------------------------------------


def _bridge(provider, ops):
    orig_add = provider.add_resource

    def add_resource(resource):
        did, cdi = orig_add(resource)
        ops.fabric_composed(resource.spec.target_node, did)
        return did, cdi

    provider.add_resource = add_resource


def build_cm_stack(mode):
    from cro_amd.fabric.fti.cm import FTICMClient
    from cro_amd.fabric.fti.token import CachedToken

    server = FakeFTIServer()
    state = {"devices": [], "counter": 0, "pending": 0}

    def refresh():
        server.cm_machines[MACHINE_UUID] = server.cm_machine(
            devices=[server.cm_device(d) for d in state["devices"]],
            device_count=len(state["devices"]),
        )

    refresh()
    orig_handler = server.handler

    def handler(request):
        if request.url.path.endswith("/actions/resize"):
            body = json.loads(request.content)
            if "increase_resource_count" in body:
                state["pending"] += 1
            else:
                for d in body["remove_resources"]["devices"]:
                    if d in state["devices"]:
                        state["devices"].remove(d)
                refresh()
            return httpx.Response(202, json={})
        if "cluster_manager" in request.url.path and state["pending"]:
            while state["pending"]:
                state["counter"] += 1
                state["devices"].append(f"GPU-cm-{state['counter']}")
                state["pending"] -= 1
            refresh()
        return orig_handler(request)

    transport = httpx.MockTransport(handler)
    mgr = build_manager(Adapter(mode, None), None)
    provider = FTICMClient(
        mgr.client, endpoint="fabric.example", tenant_id="tenant-1",
        cluster_id="cluster-1",
        token=CachedToken("fabric.example", credentials=CREDS, transport=transport),
        transport=transport,
    )
    mgr.resource_reconciler.adapter = Adapter(mode, provider)
    ops = MockNodeOps(client=mgr.client)
    mgr.resource_reconciler.node_ops = ops
    seed_chain(mgr.client)
    ops.set_driver("node0", True)
    _bridge(provider, ops)
    return mgr, state


def build_fm_stack(mode):
    from cro_amd.fabric.fti.fm import FTIFMClient
    from cro_amd.fabric.fti.token import CachedToken

    server = FakeFTIServer()
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
            server.fm_machines[MACHINE_UUID] = server.fm_machine(
                resources=[server.fm_resource(s) for s in attached]
            )
        if request.method == "DELETE" and "fabric_manager" in request.url.path:
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
    server.fm_machines[MACHINE_UUID] = server.fm_machine(resources=[])

    mgr = build_manager(Adapter(mode, None), None)
    # DEVICE_PLUGIN ⇒ OpenShift (metal3 chain); DRA ⇒ RKE2 providerID
    cluster_id = "cluster-1" if mode == "DEVICE_PLUGIN" else ""
    provider = FTIFMClient(
        mgr.client, endpoint="fabric.example", tenant_id="tenant-1",
        cluster_id=cluster_id,
        token=CachedToken("fabric.example", credentials=CREDS, transport=transport),
        transport=transport,
    )
    mgr.resource_reconciler.adapter = Adapter(mode, provider)
    ops = MockNodeOps(client=mgr.client)
    mgr.resource_reconciler.node_ops = ops
    if cluster_id:
        seed_chain(mgr.client)
    else:
        node = Node()
        node.metadata.name = "node0"
        node.status.provider_id = f"fsas-cdi://{MACHINE_UUID}"
        mgr.client.create(node)
    ops.set_driver("node0", True)
    _bridge(provider, ops)
    return mgr, attached


def build_nec_stack(mode, monkeypatch):
    from cro_amd.fabric.nec import NECClient

    PROVISIONAL = "GPU-aaaaaaaa-bbbb-cccc-dddd-eeeeeeeeeeee"
    monkeypatch.setenv("NEC_PROVISIONAL_GPU_UUID", PROVISIONAL)
    server = FakeNECServer()
    host = FakeNECServer.adapter(
        "host-adapter", "sourceFabricAdapter", "eesv",
        links=[{"type": "destinationFabricAdapter", "deviceID": "io-adapter"}],
    )
    io = FakeNECServer.adapter("io-adapter", "destinationFabricAdapter", "eeio")
    gpu = FakeNECServer.gpu("nec-gpu-1")
    server.resources = [host, io, gpu]
    server.nodes = [{"id": "nec-node-001", "name": "node0", "resources": [host, io]}]
    orig_handler = server.handler

    def handler(request):
        resp = orig_handler(request)
        if request.url.path.endswith("/layout-apply") and request.method == "POST" \
                and resp.status_code == 200:
            proc = json.loads(request.content)["procedures"][0]
            if proc["operation"] == "connect":
                gpu["device"]["links"] = [
                    {"type": "eeio", "deviceID": proc["sourceDeviceID"]},
                    {"type": "destinationFabricAdapter", "deviceID": proc["sourceDeviceID"]},
                ]
            else:
                gpu["device"]["links"] = []
        return resp

    transport = httpx.MockTransport(handler)
    mgr = build_manager(Adapter(mode, None), None)
    provider = NECClient(
        mgr.client, ip="10.0.0.1", layout_apply_port="8000",
        configuration_manager_port="8001", transport=transport, poll_interval=0.01,
    )
    mgr.resource_reconciler.adapter = Adapter(mode, provider)
    ops = MockNodeOps(client=mgr.client)
    mgr.resource_reconciler.node_ops = ops
    node = Node()
    node.metadata.name = "node0"
    node.status.provider_id = "nec-node-001"
    mgr.client.create(node)
    ops.set_driver("node0", True)
    _bridge(provider, ops)
    return mgr, None


@pytest.mark.parametrize("mode", MODES)
@pytest.mark.parametrize("backend", ["CM", "FM", "NEC"])
def test_lifecycle_matrix(backend, mode, monkeypatch):
    """Full attach→Running→delete through every backend×mode cell."""
    size = 1 if backend == "NEC" else 2  # the NEC fake owns one GPU
    if backend == "CM":
        mgr, _ = build_cm_stack(mode)
    elif backend == "FM":
        mgr, _ = build_fm_stack(mode)
    else:
        mgr, _ = build_nec_stack(mode, monkeypatch)
    mgr.start()
    try:
        mgr.client.create(make_request("r1", size=size, target_node="node0"))
        assert mgr.wait_for(
            lambda: (req := mgr.client.try_get(ComposabilityRequest, "r1")) is not None
            and req.status.state == "Running",
            timeout=30,
        ), (lambda r: f"state={r.status.state if r else 'gone'} err={r.status.error if r else ''}")(
            mgr.client.try_get(ComposabilityRequest, "r1"))
        req = mgr.client.get(ComposabilityRequest, "r1")
        assert len(req.status.resources) == size
        assert all(v.state == "Online" for v in req.status.resources.values())

        mgr.client.delete(ComposabilityRequest, "r1")
        assert mgr.wait_for(
            lambda: mgr.client.try_get(ComposabilityRequest, "r1") is None,
            timeout=30,
        )
        assert mgr.client.list(ComposableResource) == []
    finally:
        mgr.stop()


# -- Part B: mode × state × failure injection -------------------------------


class FailNClient(Client):
    """Injects an exception on a verb for the next N calls."""

    def __init__(self, store):
        super().__init__(store)
        self.inject = {}  # verb -> [exc_class, remaining]

    def _maybe(self, verb):
        entry = self.inject.get(verb)
        if entry and entry[1] > 0:
            entry[1] -= 1
            raise entry[0](f"injected {verb} failure")

    def update(self, obj):
        self._maybe("update")
        return super().update(obj)

    def update_status(self, obj):
        self._maybe("update_status")
        return super().update_status(obj)


def build_world(mode):
    client = FailNClient(InMemoryStore())
    fabric = MockFabric(models={"mi355x": 8})
    ops = MockNodeOps(client=client)
    _bridge(fabric, ops)

    class World:
        pass

    w = World()
    w.client = client
    w.resource_rec = ComposableResourceReconciler(
        client, Adapter(mode, fabric), ops, ReconcileConfig()
    )
    w.request_rec = ComposabilityRequestReconciler(client)
    make_node(client, "node0")
    ops.set_driver("node0", True)
    return w


def drive_child_to(w, child, state, n=30):
    for _ in range(n):
        cur = w.client.try_get(ComposableResource, child)
        if cur is not None and cur.status.state == state:
            return cur
        w.resource_rec.reconcile(child)
    cur = w.client.try_get(ComposableResource, child)
    raise AssertionError(
        f"never reached {state}: {cur.status.state if cur else 'gone'}")


STATES = ("", "Attaching", "Online", "Detaching")
FAILURES = (ApiError, ConflictError)


@pytest.mark.parametrize("failure", FAILURES, ids=lambda f: f.__name__)
@pytest.mark.parametrize("state", STATES, ids=lambda s: s or "None")
@pytest.mark.parametrize("mode", MODES)
def test_state_write_failure_converges(mode, state, failure):
    """A failed write at state S's write point surfaces (workqueue retries
    on raise — requeueOnErr parity, composableresource_controller.go:436)
    and the SAME key converges on the retry with no duplicate side
    effects."""
    w = build_world(mode)
    w.client.create(make_request("r1", size=1, target_node="node0"))
    for _ in range(5):
        w.request_rec.reconcile("r1")
    children = w.client.list(ComposableResource)
    assert len(children) == 1
    child = children[0].metadata.name

    if state == "Detaching":
        drive_child_to(w, child, "Online")
        w.client.delete(ComposableResource, child)  # deletionTimestamp set
        drive_child_to(w, child, "Detaching")
    elif state:
        drive_child_to(w, child, state)
    if state == "Online":
        # the Online write point is the deletion edge → Detaching
        w.client.delete(ComposableResource, child)

    w.client.inject["update_status"] = [failure, 1]
    raised = None
    for _ in range(5):  # first write in this state fails
        try:
            w.resource_rec.reconcile(child)
        except failure as exc:
            raised = exc
            break
    # the injected write MUST have been attempted and handled: an ApiError
    # propagates so the workqueue retries with backoff (requeueOnErr
    # parity); a Conflict may propagate OR be absorbed by a re-get+retry
    # inside the handler — both are correct, silent loss is not
    assert w.client.inject["update_status"][1] == 0
    if failure is ApiError:
        assert raised is not None

    # convergence: finish the lifecycle cleanly
    if state in ("Online", "Detaching"):
        for _ in range(30):
            if w.client.try_get(ComposableResource, child) is None:
                break
            w.resource_rec.reconcile(child)
        assert w.client.try_get(ComposableResource, child) is None
    else:
        drive_child_to(w, child, "Online")
        got = w.client.get(ComposableResource, child)
        assert got.status.device_id  # exactly one device attached
        # fabric must not have double-attached on the retry
        req_devices = [
            c.status.device_id for c in w.client.list(ComposableResource)
        ]
        assert len(req_devices) == len(set(req_devices)) == 1


@pytest.mark.parametrize("mode", MODES)
def test_none_state_finalizer_update_failure(mode):
    """The None state's FIRST write is the finalizer update (not status):
    inject there too (suite_test.go MockUpdate parity)."""
    w = build_world(mode)
    w.client.create(make_request("r1", size=1, target_node="node0"))
    for _ in range(5):
        w.request_rec.reconcile("r1")
    child = w.client.list(ComposableResource)[0].metadata.name

    w.client.inject["update"] = [ApiError, 1]
    with pytest.raises(ApiError):
        w.resource_rec.reconcile(child)
    drive_child_to(w, child, "Online")
    got = w.client.get(ComposableResource, child)
    assert len(got.metadata.finalizers) == 1  # added once, not twice


# -- Part C: request-controller state × failure matrix ----------------------

REQUEST_STATES = ("", "NodeAllocating", "Updating", "Running", "Cleaning")


def drive_request_to(w, name, state, n=40):
    for _ in range(n):
        cur = w.client.try_get(ComposabilityRequest, name)
        if cur is not None and cur.status.state == state:
            return cur
        w.request_rec.reconcile(name)
        # children must make progress for Updating→Running, and their
        # status must sync into the parent via the dual-kind path (the
        # hand-driven analog of the ComposableResource watch)
        for c in w.client.list(ComposableResource):
            w.resource_rec.reconcile(c.metadata.name)
            w.request_rec.reconcile(c.metadata.name)
    cur = w.client.try_get(ComposabilityRequest, name)
    raise AssertionError(
        f"never reached {state!r}: {cur.status.state if cur else 'gone'}")


@pytest.mark.parametrize("failure", FAILURES, ids=lambda f: f.__name__)
@pytest.mark.parametrize("state", REQUEST_STATES, ids=lambda s: s or "None")
def test_request_state_write_failure_converges(state, failure):
    """The fleet controller's per-state write points survive injected
    ApiError/Conflict and converge (composabilityrequest_controller
    requeueOnErr parity, :627-637)."""
    w = build_world("DRA")
    w.client.create(make_request("r1", size=2, target_node="node0"))

    if state == "Cleaning":
        # Cleaning is entered from Running on deletion
        drive_request_to(w, "r1", "Running")
        w.client.delete(ComposabilityRequest, "r1")
        for _ in range(10):
            cur = w.client.try_get(ComposabilityRequest, "r1")
            if cur is not None and cur.status.state == "Cleaning":
                break
            w.request_rec.reconcile("r1")
        assert w.client.get(ComposabilityRequest, "r1").status.state == "Cleaning"
    elif state:
        drive_request_to(w, "r1", state)
    if state == "Running":
        # Running's write point is the drift edge → NodeAllocating
        cur = w.client.get(ComposabilityRequest, "r1")
        cur.spec.resource.size = 1
        w.client.update(cur)
    if state == "Cleaning":
        pass  # next write: child deletion bookkeeping → Deleting edge
    elif state == "":
        pass  # first write: finalizer (update) then state (update_status)

    # inject once; keep driving the WHOLE machine (request + children +
    # dual-kind sync) until the poisoned write fires — some states' write
    # points only unlock after children progress (Updating→Running needs
    # Online children; Cleaning→Deleting needs children gone)
    w.client.inject["update_status"] = [failure, 1]
    raised = None
    for _ in range(20):
        if w.client.inject["update_status"][1] == 0:
            break
        try:
            w.request_rec.reconcile("r1")
            for c in w.client.list(ComposableResource):
                w.resource_rec.reconcile(c.metadata.name)
                if w.client.try_get(ComposableResource, c.metadata.name) is not None:
                    w.request_rec.reconcile(c.metadata.name)
        except failure as exc:
            raised = exc
    assert w.client.inject["update_status"][1] == 0, (
        "the injected write was never attempted in this state")
    if failure is ApiError:
        assert raised is not None

    # convergence to Running at the (possibly updated) size
    want = 1 if state == "Running" else 2
    if state == "Cleaning":  # noqa: SIM108
        # Cleaning only goes forward to deletion
        for _ in range(60):
            if w.client.try_get(ComposabilityRequest, "r1") is None:
                break
            w.request_rec.reconcile("r1")
            for c in w.client.list(ComposableResource):
                w.resource_rec.reconcile(c.metadata.name)
                if w.client.try_get(ComposableResource, c.metadata.name) is not None:
                    w.request_rec.reconcile(c.metadata.name)
        assert w.client.try_get(ComposabilityRequest, "r1") is None
        return
    # drive until Running AT the wanted size (a plain state check would
    # return early for the drift case, which stays "Running" until the
    # drift edge is processed)
    for _ in range(60):
        cur = w.client.try_get(ComposabilityRequest, "r1")
        if (cur is not None and cur.status.state == "Running"
                and len(cur.status.resources) == want
                and len(w.client.list(ComposableResource)) == want):
            break
        w.request_rec.reconcile("r1")
        for c in w.client.list(ComposableResource):
            w.resource_rec.reconcile(c.metadata.name)
            if w.client.try_get(ComposableResource, c.metadata.name) is not None:
                w.request_rec.reconcile(c.metadata.name)
    cur = w.client.get(ComposabilityRequest, "r1")
    assert cur.status.state == "Running"
    assert len(cur.status.resources) == want
    # no orphaned children beyond the wanted set
    assert len(w.client.list(ComposableResource)) == want


def test_request_cleaning_reached_and_survives_delete_failure():
    """Deletion path: Cleaning's child-delete call failing once must not
    wedge the teardown."""
    w = build_world("DRA")
    w.client.create(make_request("r1", size=2, target_node="node0"))
    drive_request_to(w, "r1", "Running")
    w.client.delete(ComposabilityRequest, "r1")

    real_delete = w.client.delete
    calls = {"n": 0}

    def flaky_delete(*a, **kw):
        calls["n"] += 1
        if calls["n"] == 1:
            raise ApiError("injected delete failure")
        return real_delete(*a, **kw)

    w.client.delete = flaky_delete
    with pytest.raises(ApiError):
        for _ in range(6):
            w.request_rec.reconcile("r1")
    for _ in range(60):
        if w.client.try_get(ComposabilityRequest, "r1") is None:
            break
        w.request_rec.reconcile("r1")
        for c in w.client.list(ComposableResource):
            w.resource_rec.reconcile(c.metadata.name)
            if w.client.try_get(ComposableResource, c.metadata.name) is not None:
                w.request_rec.reconcile(c.metadata.name)
    assert w.client.try_get(ComposabilityRequest, "r1") is None
    assert w.client.list(ComposableResource) == []
