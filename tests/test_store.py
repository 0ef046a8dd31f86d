"""Store semantics: the apiserver behaviors the reconcilers rely on."""

import queue

import pytest

from cro_amd.api.v1alpha1.types import ComposabilityRequest, ComposableResource
from cro_amd.runtime.errors import (
    AdmissionDenied,
    AlreadyExistsError,
    ConflictError,
    NotFoundError,
)
from tests.conftest import make_request, make_resource


def test_create_assigns_metadata(client):
    req = client.create(make_request("r1"))
    assert req.metadata.uid
    assert req.metadata.resourceVersion
    assert req.metadata.creationTimestamp
    assert req.metadata.generation == 1


def test_create_duplicate_rejected(client):
    client.create(make_request("r1"))
    with pytest.raises(AlreadyExistsError):
        client.create(make_request("r1"))


def test_generate_name(client):
    r = make_resource("")
    r.metadata.name = ""
    r.metadata.generateName = "gpu-"
    created = client.create(r)
    assert created.metadata.name.startswith("gpu-")
    assert len(created.metadata.name) > len("gpu-")


def test_get_returns_copy(client):
    client.create(make_request("r1"))
    a = client.get(ComposabilityRequest, "r1")
    a.spec.resource.size = 99
    b = client.get(ComposabilityRequest, "r1")
    assert b.spec.resource.size == 1


def test_update_conflict_on_stale_rv(client):
    created = client.create(make_request("r1"))
    stale = created.model_copy(deep=True)
    created.spec.resource.size = 2
    client.update(created)
    stale.spec.resource.size = 3
    with pytest.raises(ConflictError):
        client.update(stale)


def test_update_does_not_touch_status(client):
    created = client.create(make_request("r1"))
    created.status.state = "Running"
    client.update_status(created)
    fresh = client.get(ComposabilityRequest, "r1")
    fresh.spec.resource.size = 4
    fresh.status.state = "Bogus"  # must be ignored by spec update
    client.update(fresh)
    assert client.get(ComposabilityRequest, "r1").status.state == "Running"


def test_status_update_does_not_touch_spec(client):
    created = client.create(make_request("r1", size=1))
    created.spec.resource.size = 7  # must be ignored by status update
    created.status.state = "NodeAllocating"
    client.update_status(created)
    got = client.get(ComposabilityRequest, "r1")
    assert got.spec.resource.size == 1
    assert got.status.state == "NodeAllocating"


def test_generation_bumps_only_on_spec_change(client):
    created = client.create(make_request("r1"))
    created.metadata.labels["x"] = "y"
    updated = client.update(created)
    assert updated.metadata.generation == 1
    updated.spec.resource.size = 5
    updated2 = client.update(updated)
    assert updated2.metadata.generation == 2


def test_finalizer_delete_flow(client):
    r = make_resource("gpu-1")
    r.metadata.finalizers = ["cro.amd.com/finalizer"]
    created = client.create(r)
    client.delete(ComposableResource, "gpu-1")
    # still present, with deletionTimestamp
    got = client.get(ComposableResource, "gpu-1")
    assert got.metadata.deletionTimestamp is not None
    # clearing finalizers removes the object
    got.metadata.finalizers = []
    client.update(got)
    with pytest.raises(NotFoundError):
        client.get(ComposableResource, "gpu-1")


def test_delete_without_finalizers_is_immediate(client):
    client.create(make_resource("gpu-1"))
    client.delete(ComposableResource, "gpu-1")
    with pytest.raises(NotFoundError):
        client.get(ComposableResource, "gpu-1")


def test_watch_event_sequence(client, store):
    q = store.watch(["ComposableResource"])
    client.create(make_resource("gpu-1"))
    created = client.get(ComposableResource, "gpu-1")
    created.status.state = "Attaching"
    client.update_status(created)
    client.delete(ComposableResource, "gpu-1")
    types = [q.get(timeout=1).type for _ in range(3)]
    assert types == ["ADDED", "MODIFIED", "DELETED"]
    with pytest.raises(queue.Empty):
        q.get(timeout=0.05)


def test_watch_filters_kinds(client, store):
    q = store.watch(["ComposabilityRequest"])
    client.create(make_resource("gpu-1"))
    client.create(make_request("r1"))
    ev = q.get(timeout=1)
    assert ev.object.kind == "ComposabilityRequest"


def test_watch_update_carries_old_object(client, store):
    client.create(make_resource("gpu-1"))
    q = store.watch(["ComposableResource"])
    got = client.get(ComposableResource, "gpu-1")
    got.status.state = "Attaching"
    client.update_status(got)
    ev = q.get(timeout=1)
    assert ev.type == "MODIFIED"
    assert ev.old_object.status.state == ""
    assert ev.object.status.state == "Attaching"


def test_label_selector_list(client):
    client.create(make_resource("a", managed_by="r1"))
    client.create(make_resource("b", managed_by="r2"))
    client.create(make_resource("c", managed_by="r1"))
    names = {r.metadata.name for r in client.list(ComposableResource, {"app.kubernetes.io/managed-by": "r1"})}
    assert names == {"a", "c"}


def test_schema_validation_on_create(client):
    bad = make_request("r1")
    bad.spec.resource.type = "tpu"
    with pytest.raises(ValueError):
        client.create(bad)
    bad2 = make_request("r2", size=1)
    bad2.spec.resource.size = -1
    with pytest.raises(ValueError):
        client.create(bad2)
    bad3 = make_request("r3", policy="samenode")
    bad3.spec.resource.allocation_policy = "anynode"
    with pytest.raises(ValueError):
        client.create(bad3)


def test_admission_hook_rejects(client, store):
    def deny(op, old, new):
        raise AdmissionDenied("nope")

    store.register_admission("ComposabilityRequest", deny)
    with pytest.raises(AdmissionDenied):
        client.create(make_request("r1"))
    # other kinds unaffected
    client.create(make_resource("gpu-1"))


def test_admission_not_called_for_status_update(client, store):
    calls = []

    def track(op, old, new):
        calls.append(op)

    created = client.create(make_request("r1"))
    store.register_admission("ComposabilityRequest", track)
    created.status.state = "Running"
    client.update_status(created)
    assert calls == []  # status subresource bypasses admission
    created2 = client.get(ComposabilityRequest, "r1")
    created2.spec.resource.size = 2
    client.update(created2)
    assert calls == ["UPDATE"]
