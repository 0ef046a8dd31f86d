"""Composable-fabric provider contract.

Parity with the reference's ``CdiProvider`` interface and sentinel errors
(internal/cdi/client.go:34-44): four methods plus the async "still in
progress" signals the resource controller converts into short requeues
(composableresource_controller.go:233-243,367-376).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List


@dataclass
class DeviceInfo:
    """Parity: cdi.DeviceInfo (internal/cdi/client.go:24-32)."""

    node_name: str = ""
    machine_uuid: str = ""
    device_type: str = ""
    model: str = ""
    device_id: str = ""
    cdi_device_id: str = ""


class FabricError(Exception):
    pass


class WaitingDeviceAttaching(FabricError):
    """Attach accepted but still in progress (ErrWaitingDeviceAttaching)."""


class WaitingDeviceDetaching(FabricError):
    """Detach accepted but still in progress (ErrWaitingDeviceDetaching)."""


class FabricProvider:
    """Interface every fabric backend implements.

    ``add_resource`` returns (device_id, cdi_device_id) once the device is
    composed; raises WaitingDeviceAttaching while in flight.
    """

    name = "abstract"

    def add_resource(self, resource) -> "tuple[str, str]":  # pragma: no cover
        raise NotImplementedError

    def remove_resource(self, resource) -> None:  # pragma: no cover
        raise NotImplementedError

    def check_resource(self, resource) -> None:
        """Raise FabricError when the fabric reports the device unhealthy."""
        raise NotImplementedError  # pragma: no cover

    def get_resources(self) -> List[DeviceInfo]:  # pragma: no cover
        raise NotImplementedError
