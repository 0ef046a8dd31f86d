"""ComposabilityRequest controller: the fleet-level 6-state machine.

Parity with composabilityrequest_controller.go:72-690:

    "" → NodeAllocating → Updating → Running   (spec drift → NodeAllocating)
                                   ↘ Cleaning → Deleting (on deletion)

Dual-kind reconcile: the same queue also receives ComposableResource
status-change events (watch predicate) and syncs child status into the
parent's ``status.resources`` map (:169-195).  Node allocation honors the
samenode/differentnode policies and ``other_spec`` capacity admission;
surplus devices are deleted through the 5-level LRU priority buckets keyed
by the ``cro.amd.com/last-used-time`` annotation (:309-359).

Performance departure: the Updating wait for children to come Online re-queues
at ``updating_wait`` (default 100 ms) instead of 30 s (:558) — with the
watch predicate the transition is usually event-driven anyway.
"""

from __future__ import annotations

import logging
from dataclasses import dataclass
from typing import Dict, List, Optional

from ..api.v1alpha1.types import (
    ComposabilityRequest,
    ComposableResource,
    ComposableResourceSpec,
    ScalarResourceStatus,
)
from ..nodeops.nodes import (
    check_node_capacity_sufficient,
    get_all_nodes,
    node_exists,
)
from ..runtime.client import Client
from ..runtime.controller import Reconciler, Result
from ..runtime.errors import ConflictError
from ..utils import generate_composable_resource_name
from .composableresource import FINALIZER, MANAGED_BY_LABEL, READY_TO_DETACH_LABEL

log = logging.getLogger(__name__)

LAST_USED_TIME_ANNOTATION = "cro.amd.com/last-used-time"
DELETE_DEVICE_ANNOTATION = "cro.amd.com/delete-device"


@dataclass
class RequestReconcileConfig:
    updating_wait: float = 0.1  # reference: 30 s (:558)
    running_heartbeat: float = 30.0  # reference: 30 s (:585)
    cleaning_wait: float = 0.1  # reference: 30 s (:612)


class ComposabilityRequestReconciler(Reconciler):
    def __init__(
        self,
        client: Client,
        config: Optional[RequestReconcileConfig] = None,
        recorder=None,
    ):
        from ..runtime.events import NullRecorder

        self.client = client
        self.config = config or RequestReconcileConfig()
        self.recorder = recorder or NullRecorder()

    # -- plumbing ----------------------------------------------------------

    def reconcile(self, name: str) -> Result:
        request = self.client.try_get(ComposabilityRequest, name)
        if request is not None:
            return self._handle_request(request)

        resource = self.client.try_get(ComposableResource, name)
        if resource is not None:
            return self._handle_resource_change(resource)
        return Result()

    def _handle_request(self, request: ComposabilityRequest) -> Result:
        if self._garbage_collect(request):
            return Result()
        handler = {
            "": self._handle_none,
            "NodeAllocating": self._handle_node_allocating,
            "Updating": self._handle_updating,
            "Running": self._handle_running,
            "Cleaning": self._handle_cleaning,
            "Deleting": self._handle_deleting,
        }.get(request.status.state)
        if handler is None:
            self._set_error(request, f"the composabilityRequest state '{request.status.state}' is invalid")
            raise ValueError(f"invalid state {request.status.state!r}")
        try:
            return handler(request)
        except ConflictError:
            raise
        except Exception as exc:
            self._set_error(request, str(exc))
            raise

    def _set_error(self, request: ComposabilityRequest, msg: str) -> None:
        try:
            fresh = self.client.get(ComposabilityRequest, request.metadata.name)
            fresh.status.error = msg
            self.client.update_status(fresh)
        except Exception:
            pass
        self.recorder.warning(request, "ReconcileError", msg)

    def _garbage_collect(self, request: ComposabilityRequest) -> bool:
        """Target node deleted → delete the request (:147-167)."""
        if not request.spec or not request.spec.resource.target_node:
            return False
        if node_exists(self.client, request.spec.resource.target_node):
            return False
        if request.metadata.deletionTimestamp is None:
            self.client.delete(request)
            return True
        return False

    def _handle_resource_change(self, resource: ComposableResource) -> Result:
        """Sync child ComposableResource status into the parent request
        (:169-195)."""
        if resource.metadata.labels.get(READY_TO_DETACH_LABEL, ""):
            return Result()  # syncer-created CR, unmanaged (:171-176)
        parent_name = resource.metadata.labels.get(MANAGED_BY_LABEL, "")
        if not parent_name:
            return Result()
        request = self.client.try_get(ComposabilityRequest, parent_name)
        if request is None:
            return Result()
        entry = request.status.resources.get(resource.metadata.name)
        if entry is not None:
            entry.state = resource.status.state
            entry.error = resource.status.error
            entry.device_id = resource.status.device_id
            entry.cdi_device_id = resource.status.cdi_device_id
            request.status.resources[resource.metadata.name] = entry
            self.client.update_status(request)
        return Result()

    # -- states ------------------------------------------------------------

    def _handle_none(self, request: ComposabilityRequest) -> Result:
        if FINALIZER not in request.metadata.finalizers:
            request.metadata.finalizers.append(FINALIZER)
            request = self.client.update(request)
        request.status.state = "NodeAllocating"
        request.status.error = ""
        request.status.scalarResource = request.spec.resource.model_copy(deep=True)
        self.client.update_status(request)
        return Result()

    def _handle_node_allocating(self, request: ComposabilityRequest) -> Result:
        if request.metadata.deletionTimestamp is not None:
            request.status.state = "Cleaning"
            self.client.update_status(request)
            return Result()

        managed = self.client.list(ComposableResource, {MANAGED_BY_LABEL: request.metadata.name})
        # exclude children already on their way out (:228-235)
        managed = [r for r in managed if r.status.state not in ("Detaching", "Deleting")]
        # read-only snapshots (never mutated below) — the deep-copy-free
        # form keeps NodeAllocating O(children) instead of O(fleet)
        all_requests = self.client.list(ComposabilityRequest, copy=False)
        nodes = get_all_nodes(self.client)

        spec = request.spec.resource
        to_allocate = spec.size
        to_delete = 0
        nodes_used_differentnode: Dict[str, bool] = {}
        samenode_target = ""

        # keep/evict pass over existing children (:253-305)
        for resource in managed:
            if to_allocate > 0:
                if (
                    resource.spec.type != spec.type
                    or resource.spec.model != spec.model
                    or resource.spec.force_detach != spec.force_detach
                ):
                    request.status.resources.pop(resource.metadata.name, None)
                    continue
                if spec.target_node and resource.spec.target_node != spec.target_node:
                    request.status.resources.pop(resource.metadata.name, None)
                    continue
                if spec.other_spec is not None:
                    if not check_node_capacity_sufficient(
                        self.client, resource.spec.target_node, spec.other_spec
                    ):
                        request.status.resources.pop(resource.metadata.name, None)
                        continue
                if spec.allocation_policy == "differentnode":
                    if nodes_used_differentnode.get(resource.spec.target_node):
                        request.status.resources.pop(resource.metadata.name, None)
                        continue
                    nodes_used_differentnode[resource.spec.target_node] = True
                elif spec.allocation_policy == "samenode":
                    if samenode_target == "":
                        samenode_target = resource.spec.target_node
                    elif samenode_target != resource.spec.target_node:
                        request.status.resources.pop(resource.metadata.name, None)
                        continue
                to_allocate -= 1
            else:
                to_delete += 1

        # surplus eviction through the 5-level priority buckets (:309-359)
        if to_delete > 0:
            buckets: List[List[tuple]] = [[] for _ in range(5)]
            for resource in managed:
                sort_key = resource.metadata.annotations.get(LAST_USED_TIME_ANNOTATION, "")
                if not sort_key:
                    sort_key = resource.metadata.creationTimestamp or ""
                st = resource.status.state
                if st in ("", "None") or (st == "Attaching" and resource.status.device_id == ""):
                    level = 0
                elif st == "Online" and resource.metadata.annotations.get(DELETE_DEVICE_ANNOTATION) == "true":
                    level = 1
                elif st == "Attaching":
                    level = 2
                elif st == "Online":
                    level = 3
                else:
                    level = 4
                buckets[level].append((sort_key, resource.metadata.name))
            for level in buckets:
                level.sort()
            done = False
            for level in buckets:
                for _, rname in level:
                    if to_delete == 0:
                        done = True
                        break
                    request.status.resources.pop(rname, None)
                    to_delete -= 1
                if done:
                    break

        # node allocation by policy (:361-467)
        allocating: List[str] = []
        if spec.allocation_policy == "samenode" and spec.target_node:
            if not node_exists(self.client, spec.target_node):
                raise ValueError("the target node does not existed")
            if spec.other_spec is not None and not check_node_capacity_sufficient(
                self.client, spec.target_node, spec.other_spec
            ):
                raise ValueError("TargetNode does not meet spec's requirements")
            allocating = [spec.target_node] * to_allocate
        elif spec.allocation_policy == "samenode":
            if samenode_target == "" and request.status.resources:
                # no surviving child carried the node (children deleted,
                # stale status entries remain): the implicit target lives
                # in status (reference resolves it from status.resources)
                for entry in request.status.resources.values():
                    if entry.node_name:
                        samenode_target = entry.node_name
                        break
            if request.status.resources and samenode_target:
                allocating = [samenode_target] * to_allocate
            else:
                chosen = None
                for node in nodes:
                    if spec.other_spec is not None and not check_node_capacity_sufficient(
                        self.client, node.metadata.name, spec.other_spec
                    ):
                        continue
                    if self._node_occupied(node.metadata.name, request, all_requests):
                        continue
                    chosen = node.metadata.name
                    break
                if chosen is not None:
                    allocating = [chosen] * to_allocate
                if len(allocating) != to_allocate:
                    raise ValueError("insufficient number of available nodes")
        elif spec.allocation_policy == "differentnode":
            for node in nodes:
                if spec.other_spec is not None and not check_node_capacity_sufficient(
                    self.client, node.metadata.name, spec.other_spec
                ):
                    continue
                if node.metadata.name in allocating or nodes_used_differentnode.get(node.metadata.name):
                    continue
                allocating.append(node.metadata.name)
                if len(allocating) == to_allocate:
                    break
            if len(allocating) != to_allocate:
                raise ValueError("insufficient number of available nodes")

        for node_name in allocating:
            rname = generate_composable_resource_name(spec.type)
            request.status.resources[rname] = ScalarResourceStatus(node_name=node_name)

        request.status.state = "Updating"
        request.status.error = ""
        request.status.scalarResource = spec.model_copy(deep=True)
        self.client.update_status(request)
        nodes_used = sorted({e.node_name for e in request.status.resources.values()})
        self.recorder.normal(
            request,
            "NodesAllocated",
            f"{len(request.status.resources)} device(s) on {', '.join(nodes_used)}",
        )
        return Result()

    def _node_occupied(self, node_name: str, request, all_requests) -> bool:
        """Is the node claimed by another samenode request (:397-426)?"""
        for req in all_requests:
            if req.metadata.name == request.metadata.name or req.spec is None:
                continue
            target = ""
            if req.spec.resource.allocation_policy == "samenode":
                if req.spec.resource.target_node == "":
                    for v in req.status.resources.values():
                        target = v.node_name
                        break
                else:
                    target = req.spec.resource.target_node
            if target == node_name:
                return True
        return False

    def _handle_updating(self, request: ComposabilityRequest) -> Result:
        if request.metadata.deletionTimestamp is not None:
            request.status.state = "Cleaning"
            self.client.update_status(request)
            return Result()

        if request.status.scalarResource != request.spec.resource:
            request.status.state = "NodeAllocating"
            request.status.scalarResource = request.spec.resource.model_copy(deep=True)
            self.client.update_status(request)
            return Result()

        managed = self.client.list(ComposableResource, {MANAGED_BY_LABEL: request.metadata.name})
        existing = set()
        for resource in managed:
            if resource.metadata.name not in request.status.resources:
                self.client.delete(resource)  # surplus child (:509-518)
            else:
                existing.add(resource.metadata.name)

        for rname, entry in request.status.resources.items():  # fan-out (:521-542)
            if rname not in existing:
                child = ComposableResource(
                    spec=ComposableResourceSpec(
                        type=request.spec.resource.type,
                        model=request.spec.resource.model,
                        target_node=entry.node_name,
                        force_detach=request.spec.resource.force_detach,
                    )
                )
                child.metadata.name = rname
                child.metadata.labels[MANAGED_BY_LABEL] = request.metadata.name
                self.client.create(child)

        if all(r.state == "Online" for r in request.status.resources.values()):
            request.status.state = "Running"
            request.status.error = ""
            request.status.scalarResource = request.spec.resource.model_copy(deep=True)
            self.client.update_status(request)
            self.recorder.normal(
                request,
                "Running",
                f"all {len(request.status.resources)} device(s) online",
            )
            return Result()
        return Result(requeue_after=self.config.updating_wait)

    def _handle_running(self, request: ComposabilityRequest) -> Result:
        if request.metadata.deletionTimestamp is not None:
            request.status.state = "Cleaning"
            self.client.update_status(request)
            return Result()

        if request.status.scalarResource != request.spec.resource:
            log.info("spec drift on %s, redoing NodeAllocating", request.metadata.name)
            request.status.state = "NodeAllocating"
            request.status.scalarResource = request.spec.resource.model_copy(deep=True)
            self.client.update_status(request)
            self.recorder.normal(
                request, "SpecChanged", "spec drift detected; re-allocating"
            )
            return Result()

        if request.status.error:
            request.status.error = ""
            self.client.update_status(request)
        return Result(requeue_after=self.config.running_heartbeat)

    def _handle_cleaning(self, request: ComposabilityRequest) -> Result:
        managed = self.client.list(ComposableResource, {MANAGED_BY_LABEL: request.metadata.name})
        if not managed:
            request.status.state = "Deleting"
            self.client.update_status(request)
            return Result()
        for resource in managed:
            self.client.delete(resource)
        return Result(requeue_after=self.config.cleaning_wait)

    def _handle_deleting(self, request: ComposabilityRequest) -> Result:
        if FINALIZER in request.metadata.finalizers:
            request.metadata.finalizers.remove(FINALIZER)
        self.client.update(request)
        return Result()
