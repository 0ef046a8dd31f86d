import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X (run via gpurun)")


@pytest.fixture
def store():
    from cro_amd.runtime.store import InMemoryStore

    return InMemoryStore()


@pytest.fixture
def client(store):
    from cro_amd.runtime.client import Client

    return Client(store)


def make_node(client, name: str, milli_cpu=64000, memory=1 << 40, pods=128):
    from cro_amd.api.v1alpha1.types import Node

    n = Node()
    n.metadata.name = name
    n.status.capacity.milli_cpu = milli_cpu
    n.status.capacity.memory = memory
    n.status.capacity.ephemeral_storage = 1 << 40
    n.status.capacity.allowed_pod_number = pods
    return client.create(n)


def make_request(name, *, rtype="gpu", model="mi355x", size=1, target_node="",
                 policy="samenode", force_detach=False, other_spec=None):
    from cro_amd.api.v1alpha1.types import (
        ComposabilityRequest,
        ComposabilityRequestSpec,
        ScalarResourceDetails,
    )

    req = ComposabilityRequest(
        spec=ComposabilityRequestSpec(
            resource=ScalarResourceDetails(
                type=rtype, model=model, size=size, target_node=target_node,
                allocation_policy=policy, force_detach=force_detach,
                other_spec=other_spec,
            )
        )
    )
    req.metadata.name = name
    return req


def make_resource(name, *, rtype="gpu", model="mi355x", target_node="node0",
                  force_detach=False, managed_by=None, labels=None):
    from cro_amd.api.v1alpha1.types import ComposableResource, ComposableResourceSpec

    r = ComposableResource(
        spec=ComposableResourceSpec(
            type=rtype, model=model, target_node=target_node, force_detach=force_detach
        )
    )
    r.metadata.name = name
    if managed_by:
        r.metadata.labels["app.kubernetes.io/managed-by"] = managed_by
    for k, v in (labels or {}).items():
        r.metadata.labels[k] = v
    return r


@pytest.fixture
def mock_world(client):
    """Hand-driven reconcile world: client + mock fabric + mock node ops +
    both reconcilers (no manager threads — tests call reconcile directly,
    the reference's drive-by-hand pattern)."""
    from cro_amd.controllers.composabilityrequest import ComposabilityRequestReconciler
    from cro_amd.controllers.composableresource import (
        ComposableResourceReconciler,
        ReconcileConfig,
    )
    from cro_amd.fabric.adapter import Adapter
    from cro_amd.fabric.mock import MockFabric
    from cro_amd.nodeops.amdgpu import MockNodeOps

    fabric = MockFabric(models={"mi355x": 8, "mi300x": 2})
    adapter = Adapter("DRA", fabric)
    ops = MockNodeOps(client=client)

    orig_add = fabric.add_resource

    def add(resource):
        did, cdi = orig_add(resource)
        ops.fabric_composed(resource.spec.target_node, did)
        return did, cdi

    fabric.add_resource = add

    class World:
        pass

    from cro_amd.runtime.events import EventRecorder

    w = World()
    w.client = client
    w.fabric = fabric
    w.adapter = adapter
    w.ops = ops
    w.recorder = EventRecorder(client)
    w.resource_rec = ComposableResourceReconciler(
        client, adapter, ops, ReconcileConfig(), recorder=w.recorder
    )
    w.request_rec = ComposabilityRequestReconciler(client, recorder=w.recorder)
    return w


def drive(reconciler, name, n=20):
    """Reconcile a key repeatedly until quiescent (no requeue) or n times."""
    last = None
    for _ in range(n):
        last = reconciler.reconcile(name)
    return last
