"""Validating admission for ComposabilityRequest.

The three rules of the reference webhook
(composabilityrequest_webhook.go:84-131), failurePolicy=fail:

1. ``differentnode`` + explicit ``target_node`` is contradictory;
2. among differentnode requests, (type, model) must be unique cluster-wide;
3. among samenode requests, (resolved target node, type, model) must be
   unique — the implicit target node of a pending request is read from its
   first allocated resource status.

Served two ways with one rule function:
* in-process: registered on the store's admission chain (the envtest-with-
  webhook analog, used by tests/bench and the embedded runtime);
* over HTTP: an AdmissionReview endpoint (cro_amd/webhook/server.py) for
  real-cluster deployments.
"""

from __future__ import annotations

from typing import Optional

from ..api.v1alpha1.types import ComposabilityRequest
from ..runtime.errors import AdmissionDenied


def validate_composability_request(
    request: ComposabilityRequest, existing: list
) -> Optional[str]:
    """Returns a rejection message or None.  ``existing`` = all current
    ComposabilityRequests (the incoming one excluded by name)."""
    spec = request.spec.resource

    if spec.allocation_policy == "differentnode" and spec.target_node:
        return "TargetNode cannot be specified when AllocationPolicy is set to 'differentnode'"

    if spec.allocation_policy == "differentnode":
        for other in existing:
            if other.metadata.name == request.metadata.name or other.spec is None:
                continue
            o = other.spec.resource
            if (
                o.allocation_policy == "differentnode"
                and o.type == spec.type
                and o.model == spec.model
            ):
                return (
                    f"composabilityRequest resource {other.metadata.name} with type "
                    f"{spec.type} and model {spec.model} already exists"
                )
    elif spec.allocation_policy == "samenode":
        # resolve the INCOMING request's implicit target from its status
        # too (webhook :107-128 resolves both sides): an UPDATE of an
        # already-allocated no-target request must collide with an explicit
        # request on the node it actually occupies
        my_target = spec.target_node
        if not my_target:
            for v in request.status.resources.values():
                my_target = v.node_name
                break
        for other in existing:
            if other.metadata.name == request.metadata.name or other.spec is None:
                continue
            o = other.spec.resource
            target = o.target_node
            if not target:
                for v in other.status.resources.values():
                    target = v.node_name
                    break
            if target == my_target and o.type == spec.type and o.model == spec.model:
                return (
                    f"composabilityRequest resource {other.metadata.name} with type "
                    f"{spec.type} and model {spec.model} already exists"
                )
    return None


def admission_validator(client):
    """Store admission hook bound to a runtime client (CREATE/UPDATE only,
    status subresource excluded — webhook marker parity)."""

    def _validate(op: str, old, new) -> None:
        if new.spec is None:
            return
        # read-only snapshot: admission runs on every CREATE/UPDATE and a
        # deep copy of the whole fleet per write is the O(n^2) fleet term
        existing = client.list(ComposabilityRequest, copy=False)
        msg = validate_composability_request(new, existing)
        if msg:
            raise AdmissionDenied(msg)

    return _validate
