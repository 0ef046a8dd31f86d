"""Property-based state-machine semantics: random interleavings of
reconciles, deletions, fabric failures and load flips must keep the
system's invariants:

* a CR in Online always carries a device_id;
* a CR never terminates (disappears) while the fabric still holds its
  device;
* quiescing (reconciling everything repeatedly with a healthy fabric and
  no loads) always converges to terminal states with no leaked fabric
  attachments or taint rules.
"""

import hypothesis.strategies as st
from hypothesis import settings
from hypothesis.stateful import RuleBasedStateMachine, invariant, rule

from cro_amd.api.v1alpha1.types import (
    ComposabilityRequest,
    ComposableResource,
    DeviceTaintRule,
    Node,
)
from cro_amd.controllers.composabilityrequest import ComposabilityRequestReconciler
from cro_amd.controllers.upstreamsyncer import UpstreamSyncer
from cro_amd.controllers.composableresource import (
    ComposableResourceReconciler,
    ReconcileConfig,
)
from cro_amd.fabric.adapter import Adapter
from cro_amd.fabric.mock import MockFabric
from cro_amd.nodeops.amdgpu import MockNodeOps
from cro_amd.runtime.client import Client
from cro_amd.runtime.store import InMemoryStore
from tests.conftest import make_request

REQS = ["p0", "p1"]


class ControllerMachine(RuleBasedStateMachine):
    def __init__(self):
        super().__init__()
        self.client = Client(InMemoryStore())
        self.fabric = MockFabric(models={"mi355x": 6})
        self.ops = MockNodeOps(client=self.client)
        adapter = Adapter("DRA", self.fabric)
        self.res_rec = ComposableResourceReconciler(
            self.client, adapter, self.ops, ReconcileConfig()
        )
        self.req_rec = ComposabilityRequestReconciler(self.client)
        self.syncer = UpstreamSyncer(self.client, adapter, self.ops, grace_period=0.0)

        orig_add = self.fabric.add_resource

        def add(resource):
            did, cdi = orig_add(resource)
            self.ops.fabric_composed(resource.spec.target_node, did)
            return did, cdi

        self.fabric.add_resource = add
        for i in range(2):
            n = Node()
            n.metadata.name = f"node{i}"
            self.client.create(n)
            self.ops.set_driver(f"node{i}", True)

    # -- operations --------------------------------------------------------

    def _reconcile_all_once(self):
        for req in self.client.list(ComposabilityRequest):
            try:
                self.req_rec.reconcile(req.metadata.name)
            except Exception:
                pass
        for res in self.client.list(ComposableResource):
            try:
                self.res_rec.reconcile(res.metadata.name)
            except Exception:
                pass
            try:
                self.req_rec.reconcile(res.metadata.name)  # dual-kind sync
            except Exception:
                pass

    @rule(name=st.sampled_from(REQS), node=st.sampled_from(["node0", "node1"]),
          size=st.integers(0, 2))
    def create_request(self, name, node, size):
        try:
            self.client.create(make_request(name, size=size, target_node=node))
        except Exception:
            pass  # exists / admission conflict — fine

    @rule(name=st.sampled_from(REQS))
    def delete_request(self, name):
        try:
            self.client.delete(ComposabilityRequest, name)
        except Exception:
            pass

    @rule(name=st.sampled_from(REQS), size=st.integers(0, 2))
    def scale_request(self, name, size):
        req = self.client.try_get(ComposabilityRequest, name)
        if req is None or req.metadata.deletionTimestamp:
            return
        req.spec.resource.size = size
        try:
            self.client.update(req)
        except Exception:
            pass

    @rule()
    def reconcile_round(self):
        self._reconcile_all_once()

    @rule(n=st.integers(1, 2))
    def fabric_flaps(self, n):
        self.fabric.config.fail_attach = n

    @rule(node=st.sampled_from(["node0", "node1"]))
    def fabric_drift(self, node):
        # a device composes behind the operator's back (out-of-band)
        free = [d for d in self.fabric._pool.values() if not d.attached_node]
        if free:
            self.fabric.force_attach(free[0].device_id, node)
            self.ops.visible.setdefault(node, set()).add(free[0].device_id)

    @rule()
    def syncer_round(self):
        try:
            self.syncer.sync()
        except Exception:
            pass

    @rule(node=st.sampled_from(["node0", "node1"]))
    def load_flips(self, node):
        if self.ops.loads.get(node):
            self.ops.clear_loads(node)
        else:
            self.ops.add_load(node, "*")

    # -- invariants --------------------------------------------------------

    @invariant()
    def online_implies_device(self):
        for res in self.client.list(ComposableResource):
            if res.status.state == "Online":
                assert res.status.device_id, res

    @invariant()
    def no_phantom_claims(self):
        """Every device an ONLINE CR claims is genuinely attached on the
        fabric (the converse — unclaimed fabric attachments — is legal
        drift until the syncer repairs it, which teardown proves)."""
        attached = set(self.fabric.attached_to("node0")) | set(
            self.fabric.attached_to("node1")
        )
        for r in self.client.list(ComposableResource):
            if r.status.state == "Online" and r.status.device_id:
                assert r.status.device_id in attached, (
                    r.metadata.name,
                    r.status.device_id,
                )

    def teardown(self):
        # quiesce: heal everything, delete everything, reconcile to drain
        self.fabric.config.fail_attach = 0
        self.fabric.config.fail_detach = 0
        for node in ("node0", "node1"):
            self.ops.clear_loads(node)
        for req in self.client.list(ComposabilityRequest):
            try:
                self.client.delete(ComposabilityRequest, req.metadata.name)
            except Exception:
                pass
        import time as _time

        for _ in range(60):
            try:
                self.syncer.sync()  # grace 0: drift → detach CRs
            except Exception:
                pass
            _time.sleep(0.001)
            self._reconcile_all_once()
            if (
                not self.client.list(ComposabilityRequest)
                and not self.client.list(ComposableResource)
                and self.fabric.attached_to("node0") == []
                and self.fabric.attached_to("node1") == []
            ):
                break
        assert self.client.list(ComposabilityRequest) == []
        assert self.client.list(ComposableResource) == []
        assert self.fabric.attached_to("node0") == []
        assert self.fabric.attached_to("node1") == []
        assert self.client.list(DeviceTaintRule) == []


ControllerMachine.TestCase.settings = settings(
    max_examples=25, stateful_step_count=25, deadline=None
)
TestControllerProperties = ControllerMachine.TestCase
