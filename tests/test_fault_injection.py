"""Fault injection through a flaky client wrapper — the analog of the
reference's MyClient per-verb overrides (suite_test.go:244-294): API write
failures at every seam must surface into status and retry to success."""

import threading

import pytest

from cro_amd.api.v1alpha1.types import ComposabilityRequest, ComposableResource
from cro_amd.runtime.client import Client
from cro_amd.runtime.errors import ApiError
from tests.conftest import drive, make_node, make_request


class FlakyClient(Client):
    """Fails selected verbs N times before letting them through."""

    def __init__(self, store):
        super().__init__(store)
        self.fail_counts = {}  # verb -> remaining failures
        self._lock = threading.Lock()

    def _maybe_fail(self, verb):
        with self._lock:
            n = self.fail_counts.get(verb, 0)
            if n > 0:
                self.fail_counts[verb] = n - 1
                raise ApiError(f"injected {verb} failure")

    def create(self, obj):
        self._maybe_fail("create")
        return super().create(obj)

    def update(self, obj):
        self._maybe_fail("update")
        return super().update(obj)

    def update_status(self, obj):
        self._maybe_fail("update_status")
        return super().update_status(obj)

    def delete(self, *a, **kw):
        self._maybe_fail("delete")
        return super().delete(*a, **kw)


@pytest.fixture
def flaky_world(mock_world):
    flaky = FlakyClient(mock_world.client.store)
    mock_world.resource_rec.client = flaky
    mock_world.request_rec.client = flaky
    mock_world.flaky = flaky
    return mock_world


def test_status_update_failure_retries(flaky_world):
    make_node(flaky_world.client, "node0")
    flaky_world.client.create(make_request("r1", size=1, target_node="node0"))
    flaky_world.flaky.fail_counts["update_status"] = 1
    with pytest.raises(ApiError):
        flaky_world.request_rec.reconcile("r1")  # None state write fails
    # retry succeeds and the machine continues normally
    flaky_world.request_rec.reconcile("r1")
    got = flaky_world.client.get(ComposabilityRequest, "r1")
    assert got.status.state == "NodeAllocating"


def test_child_create_failure_recovers(flaky_world):
    make_node(flaky_world.client, "node0")
    flaky_world.client.create(make_request("r1", size=2, target_node="node0"))
    flaky_world.request_rec.reconcile("r1")
    flaky_world.request_rec.reconcile("r1")  # → Updating, names chosen
    flaky_world.flaky.fail_counts["create"] = 1
    with pytest.raises(ApiError):
        flaky_world.request_rec.reconcile("r1")  # first child create fails
    got = flaky_world.client.get(ComposabilityRequest, "r1")
    assert "injected create failure" in got.status.error
    flaky_world.request_rec.reconcile("r1")  # creates BOTH (idempotent names)
    children = flaky_world.client.list(
        ComposableResource, {"app.kubernetes.io/managed-by": "r1"}
    )
    assert len(children) == 2


def test_fabric_flap_then_recovery_full_manager():
    """A fabric that fails the first two attaches still converges under the
    running manager (backoff + retry)."""
    from cro_amd.bench_harness import attach_detach_cycle, build_local_stack
    from cro_amd.fabric.mock import MockFabricConfig

    stack = build_local_stack(
        node_name="node0", use_gpu=False,
        fabric_config=MockFabricConfig(fail_attach=2),
    )
    stack.mgr.start()
    try:
        timing = attach_detach_cycle(stack, "flap", size=1, timeout=20)
        assert timing["attach_ms"] < 20000
    finally:
        stack.mgr.stop()


def test_detach_fabric_failure_blocks_then_recovers():
    from cro_amd.api.v1alpha1.types import ComposabilityRequest as CR
    from cro_amd.bench_harness import build_local_stack
    from tests.conftest import make_request as mk

    stack = build_local_stack(node_name="node0", use_gpu=False)
    stack.mgr.start()
    try:
        stack.mgr.client.create(mk("r1", size=1, target_node="node0"))
        assert stack.mgr.wait_for(
            lambda: (r := stack.mgr.client.try_get(CR, "r1")) is not None
            and r.status.state == "Running",
            timeout=10,
        )
        stack.fabric.config.fail_detach = 2
        stack.mgr.client.delete(CR, "r1")
        assert stack.mgr.wait_for(
            lambda: stack.mgr.client.try_get(CR, "r1") is None, timeout=20
        )
        assert stack.fabric.attached_to("node0") == []
    finally:
        stack.mgr.stop()


def test_concurrent_requests_stress():
    """BASELINE config #5 writ large: 6 requests × 1 device churn against an
    8-device pool with concurrent reconcile workers; the pool must end
    empty with every request gone."""
    from cro_amd.api.v1alpha1.types import ComposabilityRequest as CR, Node
    from cro_amd.bench_harness import build_local_stack
    from tests.conftest import make_request as mk

    stack = build_local_stack(node_name="node0", use_gpu=False)
    stack.mgr.start()
    try:
        for i in range(1, 6):
            n = Node()
            n.metadata.name = f"node{i}"
            stack.mgr.client.create(n)
            stack.ops.set_driver(f"node{i}", True)
        names = []
        for i in range(6):
            name = f"stress-{i}"
            stack.mgr.client.create(mk(name, size=1, target_node=f"node{i % 6}"))
            names.append(name)
        assert stack.mgr.wait_for(
            lambda: all(
                (r := stack.mgr.client.try_get(CR, n)) is not None
                and r.status.state == "Running"
                for n in names
            ),
            timeout=30,
        )
        for n in names:
            stack.mgr.client.delete(CR, n)
        assert stack.mgr.wait_for(
            lambda: all(stack.mgr.client.try_get(CR, n) is None for n in names),
            timeout=30,
        )
        for i in range(6):
            assert stack.fabric.attached_to(f"node{i}") == []
    finally:
        stack.mgr.stop()


def test_fabric_double_allocation_refused(mock_world):
    """A fabric that hands out a device another CR already claims must be
    refused (dual-mapping one GPU into two workloads), not silently
    accepted — a guard the reference lacks."""
    from tests.conftest import drive, make_node, make_request
    from cro_amd.api.v1alpha1.types import ComposabilityRequest, ComposableResource

    w = mock_world
    make_node(w.client, "node0")
    w.ops.set_driver("node0", True)
    w.client.create(make_request("r1", size=1, target_node="node0"))
    drive(w.request_rec, "r1")
    req = w.client.get(ComposabilityRequest, "r1")
    child1 = next(iter(req.status.resources))
    drive(w.resource_rec, child1)
    first = w.client.get(ComposableResource, child1)
    assert first.status.state == "Online"

    # poison the fabric: always return the already-claimed device
    dup_id = first.status.device_id

    def poisoned(resource):
        return dup_id, f"amd.com/gpu={dup_id}"

    w.fabric.add_resource = poisoned
    w.client.create(make_request("r2", model="mi300x", target_node="node0"))
    drive(w.request_rec, "r2")
    req2 = w.client.get(ComposabilityRequest, "r2")
    child2 = next(iter(req2.status.resources))
    for _ in range(3):
        try:
            w.resource_rec.reconcile(child2)
        except Exception:
            pass
    second = w.client.get(ComposableResource, child2)
    assert second.status.state != "Online"
    assert "already claimed" in second.status.error
    assert second.status.device_id == ""  # identity never persisted


def test_flaky_node_exec_lifecycle_converges():
    """Intermittent node-agent/exec failures (every 3rd call raises) must
    only slow the lifecycle down, never wedge or corrupt it — the
    ExecError → status.error → backoff-requeue path, end to end."""
    from cro_amd.api.v1alpha1.types import ComposabilityRequest, ComposableResource
    from cro_amd.controllers.composabilityrequest import ComposabilityRequestReconciler
    from cro_amd.controllers.composableresource import (
        ComposableResourceReconciler,
        ReconcileConfig,
    )
    from cro_amd.fabric.adapter import Adapter
    from cro_amd.fabric.mock import MockFabric
    from cro_amd.nodeops.amdgpu import AmdNodeOps
    from cro_amd.nodeops.execs import ExecError, MockNodeExec
    from cro_amd.runtime.client import Client
    from cro_amd.runtime.store import InMemoryStore
    from tests.conftest import make_node, make_request
    from tests.test_nodeops import kfd_fixture

    class FlakyExec:
        """Seeded 10% failure per call: multi-syscall operations (a KFD
        enumeration is ~10 reads) still succeed sometimes, so retries can
        make progress — a deterministic every-Nth injector would starve
        them forever."""

        def __init__(self, inner, rate=0.10, seed=7):
            import random

            self.inner = inner
            self.rng = random.Random(seed)
            self.rate = rate
            self.calls = 0
            self.failures = 0

        def _maybe_fail(self):
            self.calls += 1
            if self.rng.random() < self.rate:
                self.failures += 1
                raise ExecError("injected agent failure")

        def run(self, *a, **k):
            self._maybe_fail()
            return self.inner.run(*a, **k)

        def read_file(self, *a, **k):
            self._maybe_fail()
            return self.inner.read_file(*a, **k)

        def write_file(self, *a, **k):
            self._maybe_fail()
            return self.inner.write_file(*a, **k)

        def list_dir(self, *a, **k):
            self._maybe_fail()
            return self.inner.list_dir(*a, **k)

        def path_exists(self, *a, **k):
            self._maybe_fail()
            return self.inner.path_exists(*a, **k)

    client = Client(InMemoryStore())
    inner = MockNodeExec()
    ids = kfd_fixture(inner, 1, node="node0")
    flaky = FlakyExec(inner)
    ops = AmdNodeOps(
        flaky, client=client, cdi_dir="/etc/cdi",
        destructive=False, initially_detached=ids,
    )
    ops.enum_cache_ttl = 0.0
    fabric = MockFabric(bind_inventory=[
        {"device_id": ids[0], "cdi_device_id": f"amd.com/gpu={ids[0]}", "model": "mi355x"}
    ])
    orig_add = fabric.add_resource

    def add(resource):
        did, cdi = orig_add(resource)
        ops.simulate_compose(resource.spec.target_node, did)
        return did, cdi

    fabric.add_resource = add
    adapter = Adapter("DRA", fabric)
    res_rec = ComposableResourceReconciler(client, adapter, ops, ReconcileConfig())
    req_rec = ComposabilityRequestReconciler(client)
    make_node(client, "node0")

    client.create(make_request("r1", size=1, target_node="node0"))

    def crank(rounds=60):
        for _ in range(rounds):
            for req in client.list(ComposabilityRequest):
                try:
                    req_rec.reconcile(req.metadata.name)
                except Exception:
                    pass
            for res in client.list(ComposableResource):
                for rec in (res_rec, req_rec):
                    try:
                        rec.reconcile(res.metadata.name)
                    except Exception:
                        pass
            r = client.try_get(ComposabilityRequest, "r1")
            if r is not None and r.status.state == "Running":
                return True
        return False

    assert crank(), client.list(ComposableResource)[0].status
    assert flaky.failures > 5  # failures actually fired along the way

    client.delete(ComposabilityRequest, "r1")
    for _ in range(120):
        for name in [r.metadata.name for r in client.list(ComposabilityRequest)] + [
            r.metadata.name for r in client.list(ComposableResource)
        ]:
            for rec in (req_rec, res_rec):
                try:
                    rec.reconcile(name)
                except Exception:
                    pass
        if not client.list(ComposabilityRequest) and not client.list(ComposableResource):
            break
    assert client.list(ComposableResource) == []
    assert fabric.attached_to("node0") == []
