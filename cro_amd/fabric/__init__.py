from .base import (  # noqa: F401
    DeviceInfo,
    FabricError,
    FabricProvider,
    WaitingDeviceAttaching,
    WaitingDeviceDetaching,
)
from .adapter import new_adapter, Adapter  # noqa: F401
