"""API error taxonomy mirroring the k8s apimachinery errors the reference
branches on (k8serrors.IsNotFound / IsConflict / IsAlreadyExists)."""

from __future__ import annotations


class ApiError(Exception):
    pass


class NotFoundError(ApiError):
    pass


class ConflictError(ApiError):
    """Optimistic-concurrency failure: stale resourceVersion."""


class AlreadyExistsError(ApiError):
    pass


class AdmissionDenied(ApiError):
    """Raised by a registered admission validator; maps to a webhook reject
    with failurePolicy=fail (composabilityrequest_webhook.go:49)."""
