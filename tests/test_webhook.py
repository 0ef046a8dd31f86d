"""Admission validation: the three webhook rules
(composabilityrequest_webhook.go:84-131 parity), exercised through the store
admission chain the way the reference exercises the real TLS endpoint via
envtest WebhookInstallOptions."""

import pytest

from cro_amd.api.v1alpha1.types import ComposabilityRequest, ScalarResourceStatus
from cro_amd.runtime.errors import AdmissionDenied
from cro_amd.webhook.validator import admission_validator, validate_composability_request
from tests.conftest import make_request


@pytest.fixture
def guarded_client(client, store):
    store.register_admission("ComposabilityRequest", admission_validator(client))
    return client


def test_differentnode_with_target_node_rejected(guarded_client):
    req = make_request("r1", policy="differentnode", target_node="node0")
    with pytest.raises(AdmissionDenied, match="TargetNode cannot be specified"):
        guarded_client.create(req)


def test_duplicate_differentnode_type_model_rejected(guarded_client):
    guarded_client.create(make_request("r1", policy="differentnode"))
    with pytest.raises(AdmissionDenied, match="already exists"):
        guarded_client.create(make_request("r2", policy="differentnode"))


def test_differentnode_different_model_allowed(guarded_client):
    guarded_client.create(make_request("r1", policy="differentnode", model="mi355x"))
    guarded_client.create(make_request("r2", policy="differentnode", model="mi300x"))


def test_duplicate_samenode_target_type_model_rejected(guarded_client):
    guarded_client.create(make_request("r1", target_node="node0"))
    with pytest.raises(AdmissionDenied, match="already exists"):
        guarded_client.create(make_request("r2", target_node="node0"))


def test_samenode_different_nodes_allowed(guarded_client):
    guarded_client.create(make_request("r1", target_node="node0"))
    guarded_client.create(make_request("r2", target_node="node1"))


def test_samenode_implicit_target_resolved_from_status(guarded_client):
    # r1 has no explicit target but its status pins node0
    r1 = guarded_client.create(make_request("r1"))
    r1.status.resources = {"gpu-x": ScalarResourceStatus(node_name="node0")}
    guarded_client.update_status(r1)
    with pytest.raises(AdmissionDenied, match="already exists"):
        guarded_client.create(make_request("r2", target_node="node0"))


def test_update_validated_too(guarded_client):
    guarded_client.create(make_request("r1", target_node="node0"))
    r2 = guarded_client.create(make_request("r2", target_node="node1"))
    r2.spec.resource.target_node = "node0"
    with pytest.raises(AdmissionDenied, match="already exists"):
        guarded_client.update(r2)


def test_self_update_not_a_conflict(guarded_client):
    created = guarded_client.create(make_request("r1", target_node="node0"))
    created.spec.resource.size = 4
    guarded_client.update(created)  # same name — excluded from conflict scan


def test_two_unpinned_samenode_requests_conflict(guarded_client):
    """Two samenode requests with no target node and the same (type, model)
    collide on the empty implicit target — rejected, matching the
    reference's comparison semantics (webhook :107-128: "" == "")."""
    guarded_client.create(make_request("r1"))
    with pytest.raises(AdmissionDenied, match="already exists"):
        guarded_client.create(make_request("r2"))
    # a different model is fine
    guarded_client.create(make_request("r3", model="mi300x"))


def test_samenode_update_resolves_incoming_implicit_target(client):
    """An UPDATE of an allocated no-target samenode request collides with
    an explicit request for the node it actually occupies (both sides of
    rule 3 resolve implicit targets, webhook :107-128)."""
    mine = make_request("mine", size=1)  # no target_node
    mine.status.resources["gpu-x"] = ScalarResourceStatus(node_name="node7")
    explicit = make_request("explicit", size=1, target_node="node7")

    msg = validate_composability_request(mine, [explicit])
    assert msg is not None and "explicit" in msg

    # different node → no collision
    elsewhere = make_request("elsewhere", size=1, target_node="node8")
    assert validate_composability_request(mine, [elsewhere]) is None


def test_terminating_requests_still_conflict(guarded_client):
    """Reference rule scope: the webhook lists ALL requests — a request
    mid-deletion (deletionTimestamp set, finalizer pending) still counts
    for duplicate detection (composabilityrequest_webhook.go:85-89 lists
    without filtering)."""
    from cro_amd.api.v1alpha1.types import ComposabilityRequest

    guarded_client.create(make_request("dying", target_node="nodeA"))
    got = guarded_client.get(ComposabilityRequest, "dying")
    got.metadata.finalizers.append("com.ie.ibm.hpsys/finalizer")
    guarded_client.update(got)
    guarded_client.delete(ComposabilityRequest, "dying")  # deletionTimestamp set
    assert guarded_client.get(
        ComposabilityRequest, "dying").metadata.deletionTimestamp is not None

    with pytest.raises(AdmissionDenied, match="already exists"):
        guarded_client.create(make_request("newer", target_node="nodeA"))


def test_differentnode_duplicate_across_terminating(guarded_client):
    from cro_amd.api.v1alpha1.types import ComposabilityRequest

    guarded_client.create(make_request("d1", policy="differentnode"))
    got = guarded_client.get(ComposabilityRequest, "d1")
    got.metadata.finalizers.append("com.ie.ibm.hpsys/finalizer")
    guarded_client.update(got)
    guarded_client.delete(ComposabilityRequest, "d1")
    with pytest.raises(AdmissionDenied, match="already exists"):
        guarded_client.create(make_request("d2", policy="differentnode"))
