"""Operator wiring: the cmd/main.go:137-201 analog.

``build_manager`` assembles a fully wired Manager: both reconcilers with
their watch sources and predicates (dual-kind watch with the status-change-
only predicate, composabilityrequest_controller.go:658-690), the in-process
admission validator (webhook analog), and optionally the upstream syncer
runnable.  Callers inject the fabric provider and NodeOps (mock or real) —
the same wiring serves tests (envtest analog), the bench, and production.
"""

from __future__ import annotations

from typing import Optional

from ..fabric import Adapter
from ..nodeops.amdgpu import NodeOps
from ..runtime.controller import Controller, Source
from ..runtime.manager import Manager
from ..runtime.store import InMemoryStore, WatchEvent
from ..webhook.validator import admission_validator
from .composabilityrequest import ComposabilityRequestReconciler, RequestReconcileConfig
from .composableresource import ComposableResourceReconciler, ReconcileConfig
from .upstreamsyncer import UpstreamSyncer


def _status_changed(ev: WatchEvent) -> bool:
    """UPDATE events pass only when .status changed (resourceStatusUpdate-
    Predicate parity, :658-678); DELETED events of managed children also
    pass — a deliberate improvement over the reference, whose Cleaning and
    Updating states poll at 30 s because child deletions never wake the
    parent (:612)."""
    if ev.type == "DELETED":
        return bool(ev.object.metadata.labels.get("app.kubernetes.io/managed-by"))
    if ev.type != "MODIFIED":
        return False
    if ev.old_object is None:
        return True
    return getattr(ev.object, "status", None) != getattr(ev.old_object, "status", None)


def _child_event_mapper(ev: WatchEvent):
    """Child status changes reconcile under the child's name (the dual-kind
    sync path); child deletions wake the owning request directly."""
    if ev.type == "DELETED":
        parent = ev.object.metadata.labels.get("app.kubernetes.io/managed-by", "")
        return [parent] if parent else []
    return [ev.object.metadata.name]


def build_manager(
    adapter: Adapter,
    node_ops: NodeOps,
    store: Optional[InMemoryStore] = None,
    resource_config: Optional[ReconcileConfig] = None,
    request_config: Optional[RequestReconcileConfig] = None,
    max_concurrent_reconciles: int = 8,
    enable_webhook: bool = True,
    syncer_period: Optional[float] = None,
    syncer_grace: float = 600.0,
    metrics_port: Optional[int] = None,
    client=None,
    record_events: bool = True,
) -> Manager:
    from ..runtime.events import EventRecorder, NullRecorder

    mgr = Manager(store=store, metrics_port=metrics_port, client=client)
    # buffered (k8s EventBroadcaster shape): reconciles only enqueue;
    # a daemon thread does the store writes off the hot path
    recorder = (
        EventRecorder(mgr.client, asynchronous=True) if record_events else NullRecorder()
    )
    mgr.recorder = recorder

    resource_reconciler = ComposableResourceReconciler(
        mgr.client, adapter, node_ops, resource_config, recorder=recorder
    )
    request_reconciler = ComposabilityRequestReconciler(
        mgr.client, request_config, recorder=recorder
    )

    mgr.add_controller(
        Controller(
            "composable_resource",
            resource_reconciler,
            sources=[Source(kind="ComposableResource")],
            max_concurrent_reconciles=max_concurrent_reconciles,
        )
    )
    mgr.add_controller(
        Controller(
            "composability_request",
            request_reconciler,
            sources=[
                Source(kind="ComposabilityRequest"),
                Source(
                    kind="ComposableResource",
                    predicate=_status_changed,
                    mapper=_child_event_mapper,
                ),
            ],
            max_concurrent_reconciles=max_concurrent_reconciles,
        )
    )

    if enable_webhook:  # ENABLE_WEBHOOKS!=false parity (cmd/main.go:196-201)
        mgr.register_admission("ComposabilityRequest", admission_validator(mgr.client))

    if syncer_period is not None:
        syncer = UpstreamSyncer(
            mgr.client, adapter, node_ops,
            grace_period=syncer_grace, recorder=recorder,
        )
        mgr.add_runnable(syncer_period, syncer.sync)
        mgr.syncer = syncer  # exposed for tests

    mgr.resource_reconciler = resource_reconciler
    mgr.request_reconciler = request_reconciler
    return mgr
