"""Single-node embedded operator stack for bench.py, smoke() and GPU tests.

Builds the full operator (manager + both controllers + admission + mock
fabric) against either:

* the REAL amdgpu node path — KFD enumeration, CDI spec writes, gfx950
  health probe — in non-destructive lifecycle mode (no physical CXL fabric
  exists on a bench box, so PCI hot-remove is simulated while every read
  path and the GPU compute probe are real); this is BASELINE.json config #2
  measured end to end; or
* MockNodeOps for CPU-only runs (BASELINE.json config #1: "envtest
  apiserver + mock fabric backend, CPU-only").

The mock fabric is bound to the real enumerated inventory so the IDs the
fabric "composes" are devices that actually exist on the node.
"""

from __future__ import annotations

import os
import time
from dataclasses import dataclass, field
from typing import List, Optional

from .api.v1alpha1.types import (
    ComposabilityRequest,
    ComposabilityRequestSpec,
    Node,
    ScalarResourceDetails,
)
from .controllers import build_manager
from .controllers.composableresource import ReconcileConfig
from .controllers.composabilityrequest import RequestReconcileConfig
from .fabric.adapter import Adapter
from .fabric.mock import MockFabric, MockFabricConfig
from .nodeops.amdgpu import AmdNodeOps, MockNodeOps
from .nodeops.execs import LocalNodeExec


@dataclass
class LocalStack:
    mgr: object
    fabric: MockFabric
    ops: object
    node_name: str
    device_ids: List[str] = field(default_factory=list)
    gpu: bool = False


def gpu_available() -> bool:
    return os.path.exists("/dev/kfd")


def build_local_stack(
    node_name: str = "bench-node",
    cdi_dir: Optional[str] = None,
    mode: str = "DRA",
    use_gpu: Optional[bool] = None,
    gpu_index: Optional[int] = None,
    gpu_count: Optional[int] = None,
    mock_devices: int = 8,
    fabric_config: Optional[MockFabricConfig] = None,
    enable_probe: bool = True,
    max_concurrent_reconciles: int = 8,
    syncer_period: Optional[float] = 1.0,
    syncer_grace: float = 10.0,
    record_events: bool = True,
) -> LocalStack:
    if use_gpu is None:
        use_gpu = gpu_available()
    if cdi_dir is None:
        cdi_dir = os.path.join(os.environ.get("TMPDIR", "/tmp"), f"cro-cdi-{os.getpid()}")

    if use_gpu:
        execer = LocalNodeExec()
        from .nodeops.kfd import enumerate_gpus

        gpus = enumerate_gpus(execer, node_name)
        if not gpus:
            raise RuntimeError("no GPUs in KFD topology")
        if gpu_index is not None:
            gpus = [gpus[gpu_index % len(gpus)]]
        elif gpu_count is not None:
            gpus = gpus[:gpu_count]
        device_ids = [g.device_id for g in gpus]
        probe_fn = None
        if enable_probe:
            from .nodeops.probe import probe_fn_for_nodeops

            probe_fn = probe_fn_for_nodeops
        fabric = MockFabric(
            config=fabric_config,
            bind_inventory=[
                {"device_id": g.device_id, "cdi_device_id": f"amd.com/gpu={g.device_id}", "model": "mi355x"}
                for g in gpus
            ],
        )
        adapter = Adapter(mode, fabric)
        mgr = build_manager(
            adapter,
            None,  # node_ops installed below (needs mgr.client)
            resource_config=ReconcileConfig(),
            request_config=RequestReconcileConfig(),
            max_concurrent_reconciles=max_concurrent_reconciles,
            syncer_period=syncer_period,
            syncer_grace=syncer_grace,
            record_events=record_events,
        )
        ops = AmdNodeOps(
            execer,
            client=mgr.client,
            cdi_dir=cdi_dir,
            destructive=False,
            initially_detached=device_ids,  # "in the fabric pool, not composed"
            probe_fn=probe_fn,
        )
        mgr.resource_reconciler.node_ops = ops
        if hasattr(mgr, "syncer"):
            mgr.syncer.node_ops = ops

        # bridge fabric composition → simulated hot-add (a real fabric would
        # make the device appear on the PCIe bus; here the silicon is already
        # present, so composition clears the simulated-detached mark)
        orig_add = fabric.add_resource

        def add_resource(resource):
            device_id, cdi_id = orig_add(resource)
            ops.simulate_compose(resource.spec.target_node, device_id)
            return device_id, cdi_id

        fabric.add_resource = add_resource
    else:
        fabric = MockFabric(config=fabric_config, models={"mi355x": mock_devices})
        adapter = Adapter(mode, fabric)
        mgr = build_manager(
            adapter,
            None,
            resource_config=ReconcileConfig(),
            request_config=RequestReconcileConfig(),
            max_concurrent_reconciles=max_concurrent_reconciles,
            syncer_period=syncer_period,
            syncer_grace=syncer_grace,
            record_events=record_events,
        )
        ops = MockNodeOps(client=mgr.client)
        mgr.resource_reconciler.node_ops = ops
        if hasattr(mgr, "syncer"):
            mgr.syncer.node_ops = ops
        ops.set_driver(node_name, True)
        orig_add = fabric.add_resource

        def add_resource(resource):
            device_id, cdi_id = orig_add(resource)
            ops.fabric_composed(resource.spec.target_node, device_id)
            return device_id, cdi_id

        fabric.add_resource = add_resource
        device_ids = []

    node = Node()
    node.metadata.name = node_name
    node.status.capacity.milli_cpu = 128000
    node.status.capacity.memory = 2 << 40
    node.status.capacity.allowed_pod_number = 256
    mgr.client.create(node)

    stack = LocalStack(
        mgr=mgr, fabric=fabric, ops=ops, node_name=node_name,
        device_ids=device_ids, gpu=use_gpu,
    )
    # one long-lived request watch for event-driven waits in
    # attach_detach_cycle (polling would add its granularity to the
    # measured latency)
    stack.request_events = mgr.store.watch(["ComposabilityRequest"])
    return stack


def attach_detach_cycle(
    stack: LocalStack, name: str, size: int = 1, timeout: float = 60.0,
    force_detach: bool = False, events=None,
) -> dict:
    """One full ComposabilityRequest lifecycle; returns timing samples.

    attach_ms = create → Running (every device Online with CDI written);
    detach_ms = delete → object gone (device drained + fabric detach done).

    ``events``: a dedicated ComposabilityRequest watch queue for this
    caller.  Concurrent driver threads MUST each pass their own
    (``stack.mgr.store.watch([...])``) — the shared ``stack.request_events``
    fallback is single-consumer and threads would steal each other's
    events from it.
    """
    import queue as _queue

    mgr = stack.mgr
    req = ComposabilityRequest(
        spec=ComposabilityRequestSpec(
            resource=ScalarResourceDetails(
                type="gpu", model="mi355x", size=size, target_node=stack.node_name,
                force_detach=force_detach,
            )
        )
    )
    req.metadata.name = name

    if events is None:
        events = getattr(stack, "request_events", None)

    def wait_event(pred) -> bool:
        """Event-driven wait on the stack's request watch (no poll
        granularity in the measured latency); falls back to polling when
        no watch is attached."""
        deadline = time.monotonic() + timeout
        while True:
            remaining = deadline - time.monotonic()
            if remaining <= 0:
                return False
            try:
                ev = events.get(timeout=remaining)
            except _queue.Empty:
                return False
            if ev.object.metadata.name == name and pred(ev):
                return True

    t0 = time.monotonic()
    mgr.client.create(req)
    if events is not None:
        ok = wait_event(
            lambda ev: ev.type != "DELETED" and ev.object.status.state == "Running"
        )
    else:
        ok = mgr.wait_for(
            lambda: mgr.client.try_get(ComposabilityRequest, name) is not None
            and mgr.client.get(ComposabilityRequest, name).status.state == "Running",
            timeout=timeout,
        )
    t1 = time.monotonic()
    if not ok:
        cur = mgr.client.try_get(ComposabilityRequest, name)
        raise RuntimeError(
            f"request {name} did not reach Running in {timeout}s "
            f"(state={cur.status.state if cur else 'gone'}, "
            f"err={cur.status.error if cur else ''})"
        )

    mgr.client.delete(ComposabilityRequest, name)
    if events is not None:
        ok = wait_event(lambda ev: ev.type == "DELETED")
    else:
        ok = mgr.wait_for(
            lambda: mgr.client.try_get(ComposabilityRequest, name) is None,
            timeout=timeout,
        )
    t2 = time.monotonic()
    if not ok:
        from .api.v1alpha1.types import ComposableResource

        children = [
            (c.metadata.name, c.status.state, c.status.error)
            for c in mgr.client.list(
                ComposableResource, {"app.kubernetes.io/managed-by": name}
            )
        ]
        cur = mgr.client.try_get(ComposabilityRequest, name)
        raise RuntimeError(
            f"request {name} did not tear down in {timeout}s "
            f"(state={cur.status.state if cur else 'gone'}, children={children})"
        )
    return {"attach_ms": (t1 - t0) * 1e3, "detach_ms": (t2 - t1) * 1e3}


def churn_cycle(
    stack: LocalStack, name: str, sizes=(1, 4, 8, 0), timeout: float = 120.0,
    force_detach: bool = False,
) -> dict:
    """BASELINE config #4: scale a single CR through spec updates under the
    validating webhook (every update passes the admission chain), then
    delete.  Returns per-transition seconds keyed ``to_<size>`` plus
    ``delete_s`` and ``total_s``.

    Spec updates ride optimistic concurrency: get-latest → mutate → update,
    retrying Conflict (the controller updates status concurrently).
    """
    import time as _time

    from .runtime.errors import ConflictError

    mgr = stack.mgr
    c = mgr.client
    out = {}
    t_start = _time.monotonic()

    first, rest = sizes[0], sizes[1:]
    req = ComposabilityRequest(
        spec=ComposabilityRequestSpec(
            resource=ScalarResourceDetails(
                type="gpu", model="mi355x", size=first,
                target_node=stack.node_name, force_detach=force_detach,
            )
        )
    )
    req.metadata.name = name

    def settled(sz):
        cur = c.try_get(ComposabilityRequest, name)
        return (
            cur is not None
            and cur.status.state == "Running"
            and len(cur.status.resources) == sz
        )

    t0 = _time.monotonic()
    c.create(req)
    if not mgr.wait_for(lambda: settled(first), timeout=timeout):
        raise RuntimeError(f"churn {name}: size={first} never settled")
    out[f"to_{first}_s"] = _time.monotonic() - t0

    for sz in rest:
        t0 = _time.monotonic()
        for _ in range(50):
            cur = c.get(ComposabilityRequest, name)
            cur.spec.resource.size = sz
            try:
                c.update(cur)
                break
            except ConflictError:
                _time.sleep(0.01)
        else:
            raise RuntimeError(f"churn {name}: update to size={sz} kept conflicting")
        if not mgr.wait_for(lambda: settled(sz), timeout=timeout):
            cur = c.try_get(ComposabilityRequest, name)
            raise RuntimeError(
                f"churn {name}: size={sz} never settled "
                f"(state={cur.status.state if cur else 'gone'})"
            )
        out[f"to_{sz}_s"] = _time.monotonic() - t0

    t0 = _time.monotonic()
    c.delete(ComposabilityRequest, name)
    if not mgr.wait_for(
        lambda: c.try_get(ComposabilityRequest, name) is None, timeout=timeout
    ):
        raise RuntimeError(f"churn {name}: delete never completed")
    out["delete_s"] = _time.monotonic() - t0
    out["total_s"] = _time.monotonic() - t_start
    return out


def reconcile_count(stack: LocalStack) -> float:
    """Total reconcile invocations across both controllers (Prometheus)."""
    total = 0.0
    for metric in stack.mgr.metrics.reconcile_total.collect():
        for sample in metric.samples:
            if sample.name.endswith("_total"):
                total += sample.value
    return total
