"""OpenAPI-level field validation matching the reference CRD schemas.

The reference encodes these as kubebuilder markers compiled into the CRD
OpenAPI schema (composabilityrequest_types.go:40-64,
composableresource_types.go:27-33); the apiserver enforces them on every
create/update.  Our in-process store applies :func:`validate_spec` at the same
point (cro_amd/runtime/store.py) so invalid specs are rejected before any
reconciler sees them.
"""

from __future__ import annotations

RESOURCE_TYPES = ("gpu", "cxlmemory")
ALLOCATION_POLICIES = ("samenode", "differentnode")


class SchemaValidationError(ValueError):
    pass


def validate_scalar_resource_details(d) -> None:
    if d.type not in RESOURCE_TYPES:
        raise SchemaValidationError(
            f"spec.resource.type must be one of {RESOURCE_TYPES}, got {d.type!r}"
        )
    if len(d.model) < 1:
        raise SchemaValidationError("spec.resource.model must be non-empty")
    if d.size < 0:
        raise SchemaValidationError("spec.resource.size must be >= 0")
    if d.allocation_policy not in ALLOCATION_POLICIES:
        raise SchemaValidationError(
            "spec.resource.allocation_policy must be one of "
            f"{ALLOCATION_POLICIES}, got {d.allocation_policy!r}"
        )
    if d.other_spec is not None:
        for field in ("milli_cpu", "memory", "ephemeral_storage", "allowed_pod_number"):
            if getattr(d.other_spec, field) < 0:
                raise SchemaValidationError(f"spec.resource.other_spec.{field} must be >= 0")


def validate_composable_resource_spec(s) -> None:
    if s.type not in RESOURCE_TYPES:
        raise SchemaValidationError(
            f"spec.type must be one of {RESOURCE_TYPES}, got {s.type!r}"
        )


def validate_spec(obj) -> None:
    """Dispatch schema validation by kind; no-op for kinds without rules."""
    kind = getattr(obj, "kind", "")
    spec = getattr(obj, "spec", None)
    if spec is None:
        return
    if kind == "ComposabilityRequest":
        validate_scalar_resource_details(spec.resource)
    elif kind == "ComposableResource":
        validate_composable_resource_spec(spec)
