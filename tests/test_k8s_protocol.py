"""kube-apiserver wire-protocol conformance.

The reference operates against a real kube-apiserver (envtest,
suite_test.go:318-409). No apiserver/etcd binaries exist in this offline
image, so exact *protocol* conformance is the testable contract instead:
this suite pins every wire behavior a k8s client library relies on —
list metadata, watch framing, resourceVersion semantics, 410 Gone,
bookmarks, the WatchList protocol, metav1.Status errors, and the real
resource.k8s.io / coordination.k8s.io groups — against the API server
(server/api.py) and the client (runtime/remote.py).  docs/K8S_COMPAT.md
is the human-readable matrix of what is and is not k8s-exact.
"""

import json

import pytest
from fastapi.testclient import TestClient

from cro_amd.api.v1alpha1.types import (
    ComposabilityRequest,
    Lease,
    Node,
    ResourceSlice,
)
from cro_amd.controllers import build_manager
from cro_amd.fabric.adapter import Adapter
from cro_amd.fabric.mock import MockFabric
from cro_amd.server.api import build_app
from tests.conftest import make_request

GROUP = "/apis/cro.hpsys.ibm.ie.com/v1alpha1"


@pytest.fixture
def stack():
    mgr = build_manager(Adapter("DRA", MockFabric()), None, enable_webhook=False)
    return TestClient(build_app(mgr.client)), mgr


# -- list semantics ----------------------------------------------------------


def test_list_carries_listmeta_resourceversion(stack):
    http, mgr = stack
    mgr.client.create(make_request("a", target_node="n0"))
    body = http.get(f"{GROUP}/composabilityrequests").json()
    assert body["kind"] == "ComposabilityRequestList"
    assert body["apiVersion"] == "cro.hpsys.ibm.ie.com/v1alpha1"
    rv = int(body["metadata"]["resourceVersion"])
    assert rv >= int(body["items"][0]["metadata"]["resourceVersion"])


def test_object_rv_and_list_rv_share_one_space(stack):
    """etcd-revision semantics: the list rv advances exactly with object
    writes, and a created object's rv slots into the same sequence."""
    http, mgr = stack
    rv0 = int(http.get(f"{GROUP}/nodes").json()["metadata"]["resourceVersion"])
    n = Node()
    n.metadata.name = "n1"
    created = mgr.client.create(n)
    rv1 = int(http.get(f"{GROUP}/nodes").json()["metadata"]["resourceVersion"])
    assert int(created.metadata.resourceVersion) == rv1
    assert rv1 > rv0


# -- watch framing -----------------------------------------------------------


def event_lines(resp_text):
    return [json.loads(l) for l in resp_text.splitlines() if l.strip()]


def test_watch_event_has_no_out_of_band_fields(stack):
    """WatchEvent = {"type", "object"} exactly; the resume token is the
    object's metadata.resourceVersion (no vendor framing field)."""
    http, mgr = stack
    listed = http.get(f"{GROUP}/composabilityrequests").json()
    rv = listed["metadata"]["resourceVersion"]
    mgr.client.create(make_request("w", target_node="n0"))
    with http.stream(
        "GET",
        f"{GROUP}/composabilityrequests",
        params={"watch": "true", "resourceVersion": rv, "timeoutSeconds": 2},
    ) as resp:
        for line in resp.iter_lines():
            if line.strip():
                ev = json.loads(line)
                break
    assert set(ev.keys()) == {"type", "object"}
    assert ev["type"] == "ADDED"
    assert int(ev["object"]["metadata"]["resourceVersion"]) > int(rv)


def test_watch_from_now_streams_no_replay(stack):
    """watch without resourceVersion starts at 'now' — informers do
    list-then-watch; the stream must NOT re-send existing objects."""
    http, mgr = stack
    mgr.client.create(make_request("pre", target_node="n0"))
    with http.stream(
        "GET",
        f"{GROUP}/composabilityrequests",
        params={"watch": "true", "timeoutSeconds": 1},
    ) as resp:
        got = [json.loads(l) for l in resp.iter_lines() if l.strip()]
    assert got == []


def test_watch_410_gone_contract(stack):
    """An aged-out resourceVersion gets ONE ERROR event whose object is a
    metav1.Status (code 410, reason Expired) and the stream ENDS."""
    http, mgr = stack
    for i in range(3):
        mgr.client.create(make_request(f"g{i}", model=f"m{i}", target_node="n0"))
    while mgr.store._event_log:
        mgr.store._event_log.popleft()
    with http.stream(
        "GET",
        f"{GROUP}/composabilityrequests",
        params={"watch": "true", "resourceVersion": "1", "timeoutSeconds": 5},
    ) as resp:
        got = [json.loads(l) for l in resp.iter_lines() if l.strip()]
    assert len(got) == 1
    st = got[0]
    assert st["type"] == "ERROR"
    assert st["object"]["kind"] == "Status"
    assert st["object"]["apiVersion"] == "v1"
    assert st["object"]["status"] == "Failure"
    assert st["object"]["code"] == 410
    assert st["object"]["reason"] == "Expired"
    assert "too old resource version" in st["object"]["message"]


def test_watch_bookmarks(stack):
    """allowWatchBookmarks → BOOKMARK events carry a fresh rv for idle
    clients (k8s bookmark contract: only kind/apiVersion/metadata.rv)."""
    http, mgr = stack
    mgr.client.create(make_request("b", target_node="n0"))
    with http.stream(
        "GET",
        f"{GROUP}/composabilityrequests",
        params={
            "watch": "true", "allowWatchBookmarks": "true",
            "timeoutSeconds": 2.5,
        },
    ) as resp:
        got = [json.loads(l) for l in resp.iter_lines() if l.strip()]
    bms = [g for g in got if g["type"] == "BOOKMARK"]
    assert bms, got
    bm = bms[0]["object"]
    assert bm["kind"] == "ComposabilityRequest"
    assert int(bm["metadata"]["resourceVersion"]) > 0


def test_watchlist_protocol(stack):
    """sendInitialEvents=true (k8s 1.27+): ADDED per existing object, then
    the k8s.io/initial-events-end BOOKMARK, then live events only."""
    http, mgr = stack
    mgr.client.create(make_request("x1", target_node="n0"))
    mgr.client.create(make_request("x2", model="m2", target_node="n0"))
    with http.stream(
        "GET",
        f"{GROUP}/composabilityrequests",
        params={
            "watch": "true", "sendInitialEvents": "true",
            "allowWatchBookmarks": "true", "timeoutSeconds": 1,
        },
    ) as resp:
        got = [json.loads(l) for l in resp.iter_lines() if l.strip()]
    assert [g["type"] for g in got[:3]] == ["ADDED", "ADDED", "BOOKMARK"]
    assert got[2]["object"]["metadata"]["annotations"] == {
        "k8s.io/initial-events-end": "true"
    }


# -- error bodies ------------------------------------------------------------


@pytest.mark.parametrize(
    "do,code,reason",
    [
        (lambda http: http.get(f"{GROUP}/composabilityrequests/nope"),
         404, "NotFound"),
        (lambda http: http.post(
            f"{GROUP}/nodes",
            json={"apiVersion": "v1", "kind": "Node", "metadata": {"name": "dup"}}),
         409, "AlreadyExists"),
        (lambda http: http.put(
            f"{GROUP}/nodes/dup",
            json={"apiVersion": "v1", "kind": "Node",
                  "metadata": {"name": "dup", "resourceVersion": "999999"}}),
         409, "Conflict"),
    ],
)
def test_errors_are_metav1_status(stack, do, code, reason):
    http, mgr = stack
    n = Node()
    n.metadata.name = "dup"
    mgr.client.create(n)
    resp = do(http)
    assert resp.status_code == code
    body = resp.json()
    assert body["kind"] == "Status"
    assert body["apiVersion"] == "v1"
    assert body["status"] == "Failure"
    assert body["reason"] == reason
    assert body["code"] == code


def test_delete_returns_success_status(stack):
    http, mgr = stack
    n = Node()
    n.metadata.name = "bye"
    mgr.client.create(n)
    resp = http.delete(f"{GROUP}/nodes/bye")
    assert resp.status_code == 200
    assert resp.json()["kind"] == "Status"
    assert resp.json()["status"] == "Success"


# -- real k8s API groups -----------------------------------------------------


def test_resource_k8s_io_group(stack):
    """ResourceSlices/DeviceTaintRules are served at their canonical
    resource.k8s.io/v1alpha3 paths (the group the reference writes,
    internal/utils/gpus.go:894-989)."""
    http, mgr = stack
    sl = ResourceSlice()
    sl.metadata.name = "node0-slice"
    sl.spec.node_name = "node0"
    mgr.client.create(sl)
    body = http.get("/apis/resource.k8s.io/v1alpha3/resourceslices").json()
    assert body["apiVersion"] == "resource.k8s.io/v1alpha3"
    assert [i["metadata"]["name"] for i in body["items"]] == ["node0-slice"]
    one = http.get(
        "/apis/resource.k8s.io/v1alpha3/resourceslices/node0-slice").json()
    assert one["apiVersion"] == "resource.k8s.io/v1alpha3"
    assert one["kind"] == "ResourceSlice"


def test_coordination_k8s_io_group(stack):
    """Leases live at coordination.k8s.io/v1 (leader election kind)."""
    http, mgr = stack
    lease = Lease()
    lease.metadata.name = "c5744f42.hpsys.ibm.ie.com"
    lease.spec.holderIdentity = "me"
    resp = http.post(
        "/apis/coordination.k8s.io/v1/leases",
        json=lease.model_dump(by_alias=True),
    )
    assert resp.status_code == 201
    got = http.get(
        "/apis/coordination.k8s.io/v1/leases/c5744f42.hpsys.ibm.ie.com").json()
    assert got["apiVersion"] == "coordination.k8s.io/v1"
    assert got["spec"]["holderIdentity"] == "me"


# -- client (informer) side --------------------------------------------------


def test_remote_client_list_then_watch(stack):
    """RemoteClient runs the informer protocol against the server over a
    real socket: LIST → synthetic ADDED replay → WATCH from
    ListMeta.resourceVersion → live events; Status errors round-trip to
    the exact error classes."""
    import socket
    import threading
    import time

    import httpx
    import uvicorn

    from cro_amd.runtime.errors import ConflictError
    from cro_amd.runtime.remote import RemoteClient

    _, mgr = stack
    mgr.client.create(make_request("rc1", target_node="n0"))

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    server = uvicorn.Server(uvicorn.Config(
        build_app(mgr.client), host="127.0.0.1", port=port, log_level="error"))
    t = threading.Thread(target=server.run, daemon=True)
    t.start()
    deadline = time.monotonic() + 15
    while time.monotonic() < deadline:
        try:
            if httpx.get(f"http://127.0.0.1:{port}/healthz", timeout=1).status_code == 200:
                break
        except Exception:
            time.sleep(0.05)
    else:
        pytest.fail("server did not come up")

    rc = RemoteClient(f"http://127.0.0.1:{port}")
    try:
        items = rc.list(ComposabilityRequest)
        assert [i.metadata.name for i in items] == ["rc1"]

        q = rc.watch(["ComposabilityRequest"])
        # informer replay: the existing object arrives as synthetic ADDED
        ev = q.get(timeout=10)
        assert ev.type == "ADDED" and ev.object.metadata.name == "rc1"
        # live event after the replay boundary
        mgr.client.create(make_request("rc2", model="m2", target_node="n0"))
        ev = q.get(timeout=10)
        assert ev.type == "ADDED" and ev.object.metadata.name == "rc2"

        # conflict round-trips as ConflictError off the Status body
        stale = items[0].model_copy(deep=True)
        stale.metadata.resourceVersion = "999999"
        with pytest.raises(ConflictError):
            rc.update(stale)
    finally:
        rc.close()
        server.should_exit = True
        t.join(timeout=5)


def test_field_selector_metadata_name(stack):
    """kubectl's fieldSelector=metadata.name=<x> form (the supported field
    for custom resources); unsupported fields get a 422 Status."""
    http, mgr = stack
    mgr.client.create(make_request("fs1", target_node="n0"))
    mgr.client.create(make_request("fs2", model="m2", target_node="n0"))
    body = http.get(
        f"{GROUP}/composabilityrequests",
        params={"fieldSelector": "metadata.name=fs2"},
    ).json()
    assert [i["metadata"]["name"] for i in body["items"]] == ["fs2"]
    resp = http.get(
        f"{GROUP}/composabilityrequests",
        params={"fieldSelector": "spec.nodeName=x"},
    )
    assert resp.status_code == 422
    assert resp.json()["kind"] == "Status"


def test_merge_patch_rfc7386(stack):
    """PATCH with application/merge-patch+json (kubectl patch default for
    CRs): recursive merge, null deletes, unknown content-type → 422,
    conflicts retried server-side; the status subresource patches only
    status."""
    http, mgr = stack
    mgr.client.create(make_request("p1", target_node="n0"))

    # spec field merge
    resp = http.patch(
        f"{GROUP}/composabilityrequests/p1",
        json={"spec": {"resource": {"size": 5}}},
        headers={"Content-Type": "application/merge-patch+json"},
    )
    assert resp.status_code == 200, resp.text
    body = resp.json()
    assert body["spec"]["resource"]["size"] == 5
    assert body["spec"]["resource"]["model"] == "mi355x"  # untouched

    # null deletes a key (resets to schema default on validate)
    resp = http.patch(
        f"{GROUP}/composabilityrequests/p1",
        json={"metadata": {"labels": None}},
        headers={"Content-Type": "application/merge-patch+json"},
    )
    assert resp.status_code == 200
    assert resp.json()["metadata"]["labels"] == {}

    # status subresource patch touches only status
    resp = http.patch(
        f"{GROUP}/composabilityrequests/p1/status",
        json={"status": {"state": "NodeAllocating"}},
        headers={"Content-Type": "application/merge-patch+json"},
    )
    assert resp.status_code == 200
    got = mgr.client.get(ComposabilityRequest, "p1")
    assert got.status.state == "NodeAllocating"
    assert got.spec.resource.size == 5  # spec untouched by status patch

    # wrong content type → 422 Status
    resp = http.patch(
        f"{GROUP}/composabilityrequests/p1",
        content=b"not json",
        headers={"Content-Type": "application/strategic-merge-patch+json"},
    )
    assert resp.status_code == 422
    assert resp.json()["kind"] == "Status"

    # schema still enforced through the patch path
    resp = http.patch(
        f"{GROUP}/composabilityrequests/p1",
        json={"spec": {"resource": {"type": "not-a-type"}}},
        headers={"Content-Type": "application/merge-patch+json"},
    )
    assert resp.status_code == 422
