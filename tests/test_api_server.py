"""REST API server tests: CRUD through apiserver-shaped routes, admission
enforcement, end-to-end operator drive over HTTP."""

import pytest
from fastapi.testclient import TestClient

from cro_amd.api.v1alpha1.types import ComposabilityRequest
from cro_amd.bench_harness import build_local_stack
from cro_amd.server.api import BASE, build_app
from tests.conftest import make_request

PLURAL = f"{BASE}/composabilityrequests"


@pytest.fixture
def api_stack():
    stack = build_local_stack(node_name="node0", use_gpu=False)
    stack.mgr.start()
    http = TestClient(build_app(stack.mgr.client))
    yield http, stack
    stack.mgr.stop()


def test_create_get_list_delete(api_stack):
    http, stack = api_stack
    body = make_request("r1", target_node="node0").model_dump(by_alias=True)
    resp = http.post(PLURAL, json=body)
    assert resp.status_code == 201, resp.text
    created = resp.json()
    assert created["metadata"]["uid"]

    assert http.get(f"{PLURAL}/r1").status_code == 200
    listed = http.get(PLURAL).json()
    assert len(listed["items"]) == 1
    assert listed["kind"] == "ComposabilityRequestList"

    # operator drives it to Running; status visible over the API
    assert stack.mgr.wait_for(
        lambda: http.get(f"{PLURAL}/r1").json()["status"]["state"] == "Running",
        timeout=10,
    )

    resp = http.delete(f"{PLURAL}/r1")
    assert resp.status_code == 200  # metav1.Status success body
    assert resp.json()["kind"] == "Status" and resp.json()["status"] == "Success"
    assert stack.mgr.wait_for(
        lambda: http.get(f"{PLURAL}/r1").status_code == 404, timeout=10
    )


def test_admission_enforced_over_http(api_stack):
    http, _ = api_stack
    bad = make_request("r1", policy="differentnode", target_node="node0")
    resp = http.post(PLURAL, json=bad.model_dump(by_alias=True))
    assert resp.status_code == 403
    assert resp.json()["kind"] == "Status"
    assert "TargetNode cannot be specified" in resp.json()["message"]


def test_schema_validation_over_http(api_stack):
    http, _ = api_stack
    bad = make_request("r1").model_dump(by_alias=True)
    bad["spec"]["resource"]["type"] = "tpu"
    resp = http.post(PLURAL, json=bad)
    assert resp.status_code == 422


def test_conflict_on_duplicate_create(api_stack):
    http, _ = api_stack
    body = make_request("r1", target_node="node0").model_dump(by_alias=True)
    assert http.post(PLURAL, json=body).status_code == 201
    assert http.post(PLURAL, json=body).status_code == 409


def test_update_spec_over_http(api_stack):
    http, stack = api_stack
    body = make_request("r1", target_node="node0").model_dump(by_alias=True)
    http.post(PLURAL, json=body)
    assert stack.mgr.wait_for(
        lambda: http.get(f"{PLURAL}/r1").json()["status"]["state"] == "Running",
        timeout=10,
    )
    current = http.get(f"{PLURAL}/r1").json()
    current["spec"]["resource"]["size"] = 2
    resp = http.put(f"{PLURAL}/r1", json=current)
    assert resp.status_code == 200
    assert stack.mgr.wait_for(
        lambda: len(http.get(f"{PLURAL}/r1").json()["status"]["resources"]) == 2
        and http.get(f"{PLURAL}/r1").json()["status"]["state"] == "Running",
        timeout=10,
    )


def test_label_selector_list(api_stack):
    http, stack = api_stack
    from cro_amd.api.v1alpha1.types import ComposableResource
    from tests.conftest import make_resource

    stack.mgr.client.create(make_resource("a", managed_by="rx"))
    stack.mgr.client.create(make_resource("b", managed_by="ry"))
    items = http.get(
        f"{BASE}/composableresources",
        params={"labelSelector": "app.kubernetes.io/managed-by=rx"},
    ).json()["items"]
    assert [i["metadata"]["name"] for i in items] == ["a"]
    # cleanup before manager teardown races with reconciles
    stack.mgr.client.delete(ComposableResource, "a")
    stack.mgr.client.delete(ComposableResource, "b")


def test_metrics_endpoint(api_stack):
    http, _ = api_stack
    resp = http.get("/metrics")
    assert resp.status_code == 200
    assert b"cro_reconcile_total" in resp.content


def test_metrics_token_auth(api_stack, monkeypatch):
    http, _ = api_stack
    monkeypatch.setenv("CRO_METRICS_TOKEN", "sekrit")
    assert http.get("/metrics").status_code == 401
    assert http.get("/metrics", headers={"Authorization": "Bearer wrong"}).status_code == 401
    ok = http.get("/metrics", headers={"Authorization": "Bearer sekrit"})
    assert ok.status_code == 200 and b"cro_reconcile_total" in ok.content


def test_api_token_enforced_on_apis_routes():
    """CRO_API_TOKEN (or token=) gates every /apis route with a k8s
    Status-shaped 401; /healthz+/readyz stay open for kubelet probes."""
    from fastapi.testclient import TestClient

    from cro_amd.controllers import build_manager
    from cro_amd.fabric.adapter import Adapter
    from cro_amd.fabric.mock import MockFabric
    from cro_amd.server.api import build_app

    mgr = build_manager(Adapter("DRA", MockFabric()), None, enable_webhook=False)
    app = build_app(mgr.client, token="sekrit")
    c = TestClient(app)

    r = c.get("/apis/cro.hpsys.ibm.ie.com/v1alpha1/nodes")
    assert r.status_code == 401
    assert r.json()["kind"] == "Status"
    assert r.json()["reason"] == "Unauthorized"
    r = c.post("/apis/cro.hpsys.ibm.ie.com/v1alpha1/nodes", json={
        "apiVersion": "v1", "kind": "Node", "metadata": {"name": "n"}})
    assert r.status_code == 401
    # health probes stay open
    assert c.get("/healthz").status_code == 200
    assert c.get("/readyz").status_code == 200
    # correct token passes
    ok = {"Authorization": "Bearer sekrit"}
    assert c.get("/apis/cro.hpsys.ibm.ie.com/v1alpha1/nodes", headers=ok).status_code == 200
    # wrong token still refused
    bad = {"Authorization": "Bearer wrong"}
    assert c.get("/apis/cro.hpsys.ibm.ie.com/v1alpha1/nodes", headers=bad).status_code == 401
