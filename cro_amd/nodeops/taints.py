"""DRA DeviceTaintRule helpers (gpus.go:894-989 parity).

During detach in DRA mode a NoSchedule taint rule keyed by the device uuid
blocks the scheduler from placing new claims on the device being pulled;
the rule is removed once the device is invisible.
"""

from __future__ import annotations

from ..api.v1alpha1.types import DeviceTaintRule, DeviceTaintRuleSpec
from ..runtime.client import Client
from ..runtime.errors import AlreadyExistsError, NotFoundError


def _rule_name(resource) -> str:
    return f"cro-detach-{resource.metadata.name}"


def create_device_taint(client: Client, resource) -> None:
    rule = DeviceTaintRule()
    rule.metadata.name = _rule_name(resource)
    rule.metadata.labels["app.kubernetes.io/managed-by"] = "cro-amd"
    rule.spec = DeviceTaintRuleSpec(
        device_uuid=resource.status.device_id,
        reason=f"composable device {resource.status.device_id} detaching",
    )
    try:
        client.create(rule)
    except AlreadyExistsError:
        pass  # idempotent across requeues


def has_device_taint(client: Client, resource) -> bool:
    try:
        client.get(DeviceTaintRule, _rule_name(resource))
        return True
    except NotFoundError:
        return False


def delete_device_taint(client: Client, resource) -> None:
    try:
        client.delete(DeviceTaintRule, _rule_name(resource))
    except NotFoundError:
        pass
