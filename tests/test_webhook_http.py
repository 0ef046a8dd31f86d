"""AdmissionReview HTTP endpoint tests (webhook_suite_test.go analog, with
httpx ASGITransport instead of a TLS envtest endpoint)."""

import httpx
import pytest

from cro_amd.api.v1alpha1.types import ComposabilityRequest
from cro_amd.webhook.server import WEBHOOK_PATH, build_app
from tests.conftest import make_request


@pytest.fixture
def webhook_client(client):
    from fastapi.testclient import TestClient

    app = build_app(lambda: client.list(ComposabilityRequest))
    return TestClient(app), client


def review(obj: dict, operation="CREATE", uid="uid-1") -> dict:
    return {
        "apiVersion": "admission.k8s.io/v1",
        "kind": "AdmissionReview",
        "request": {"uid": uid, "operation": operation, "object": obj},
    }


def test_health_endpoints(webhook_client):
    http, _ = webhook_client
    assert http.get("/healthz").json() == {"status": "ok"}
    assert http.get("/readyz").json() == {"status": "ok"}


def test_allowed_create(webhook_client):
    http, _ = webhook_client
    body = review(make_request("r1").model_dump(by_alias=True))
    resp = http.post(WEBHOOK_PATH, json=body).json()
    assert resp["response"]["allowed"] is True
    assert resp["response"]["uid"] == "uid-1"


def test_rejected_differentnode_with_target(webhook_client):
    http, _ = webhook_client
    bad = make_request("r1", policy="differentnode", target_node="node0")
    resp = http.post(WEBHOOK_PATH, json=review(bad.model_dump(by_alias=True))).json()
    assert resp["response"]["allowed"] is False
    assert "TargetNode cannot be specified" in resp["response"]["status"]["message"]
    assert resp["response"]["status"]["code"] == 403


def test_rejected_duplicate(webhook_client):
    http, store_client = webhook_client
    store_client.create(make_request("existing", target_node="node0"))
    dup = make_request("r2", target_node="node0")
    resp = http.post(WEBHOOK_PATH, json=review(dup.model_dump(by_alias=True))).json()
    assert resp["response"]["allowed"] is False
    assert "already exists" in resp["response"]["status"]["message"]


def test_update_operation_validated(webhook_client):
    http, store_client = webhook_client
    store_client.create(make_request("existing", target_node="node0"))
    dup = make_request("r2", target_node="node0")
    resp = http.post(
        WEBHOOK_PATH, json=review(dup.model_dump(by_alias=True), operation="UPDATE")
    ).json()
    assert resp["response"]["allowed"] is False


def test_delete_operation_always_allowed(webhook_client):
    http, store_client = webhook_client
    bad = make_request("r1", policy="differentnode", target_node="node0")
    resp = http.post(
        WEBHOOK_PATH, json=review(bad.model_dump(by_alias=True), operation="DELETE")
    ).json()
    assert resp["response"]["allowed"] is True


def test_malformed_object_rejected(webhook_client):
    http, _ = webhook_client
    resp = http.post(WEBHOOK_PATH, json=review({"spec": {"resource": {"type": 42}}})).json()
    assert resp["response"]["allowed"] is False
    assert "invalid ComposabilityRequest" in resp["response"]["status"]["message"]
