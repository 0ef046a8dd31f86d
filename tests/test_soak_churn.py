"""Bounded churn soak: random create/scale/delete against the running
manager for a few seconds, then convergence invariants.  Shakes out races
between the fan-out workers, the dual-kind watch and the store's optimistic
concurrency that targeted tests won't."""

import random

from cro_amd.api.v1alpha1.types import (
    ComposabilityRequest,
    ComposableResource,
    DeviceTaintRule,
    Node,
)
from cro_amd.bench_harness import build_local_stack
from cro_amd.runtime.errors import AdmissionDenied, ConflictError, NotFoundError
from tests.conftest import make_request


import pytest


@pytest.mark.parametrize("seed", [1234, 99, 31337])
def test_churn_soak_converges(seed):
    rng = random.Random(seed)
    stack = build_local_stack(node_name="node0", use_gpu=False)
    stack.mgr.start()
    try:
        for i in range(1, 4):
            n = Node()
            n.metadata.name = f"node{i}"
            stack.mgr.client.create(n)
            stack.ops.set_driver(f"node{i}", True)

        live = set()
        for step in range(120):
            op = rng.random()
            try:
                if op < 0.4 and len(live) < 4:
                    name = f"soak-{step}"
                    node = f"node{rng.randrange(4)}"
                    stack.mgr.client.create(
                        make_request(name, size=rng.randrange(1, 3), target_node=node)
                    )
                    live.add(name)
                elif op < 0.7 and live:
                    name = rng.choice(sorted(live))
                    req = stack.mgr.client.try_get(ComposabilityRequest, name)
                    if req is not None and req.metadata.deletionTimestamp is None:
                        req.spec.resource.size = rng.randrange(0, 3)
                        stack.mgr.client.update(req)
                elif live:
                    name = rng.choice(sorted(live))
                    stack.mgr.client.delete(ComposabilityRequest, name)
                    live.discard(name)
            except (AdmissionDenied, ConflictError, NotFoundError):
                pass  # races with reconciles are expected; soak continues
            if rng.random() < 0.2:
                import time

                time.sleep(0.02)

        # drain everything and assert convergence
        for name in sorted(live):
            try:
                stack.mgr.client.delete(ComposabilityRequest, name)
            except NotFoundError:
                pass
        assert stack.mgr.wait_for(
            lambda: stack.mgr.client.list(ComposabilityRequest) == [], timeout=30
        ), [r.metadata.name for r in stack.mgr.client.list(ComposabilityRequest)]
        assert stack.mgr.wait_for(
            lambda: stack.mgr.client.list(ComposableResource) == [], timeout=30
        ), [
            (r.metadata.name, r.status.state, r.status.error)
            for r in stack.mgr.client.list(ComposableResource)
        ]
        # every fabric device released, no taints left behind
        assert stack.mgr.wait_for(
            lambda: all(
                stack.fabric.attached_to(f"node{i}") == [] for i in range(4)
            ),
            timeout=30,
        )
        assert stack.mgr.client.list(DeviceTaintRule) == []
    finally:
        stack.mgr.stop()
