"""In-process mock fabric manager.

The reference tests emulate the fabric with an httptest TLS server
(composableresource_controller_test.go:737-997); this mock is the in-process
equivalent and also BASELINE.json config #1's backend ("envtest apiserver +
mock fabric backend").  It models a pool of composable MI355X devices with:

* configurable attach/detach latency, optionally asynchronous (first call
  raises WaitingDeviceAttaching until the latency elapses — FTI CM semantics,
  fti/cm/client.go:114-187) or synchronous (FTI FM semantics);
* failure injection per operation (attach/detach/health);
* an inventory the upstream syncer can diff against
  (GetResources parity, fti/fm/client.go:361-414).

When ``bind_inventory`` is given real device IDs (from KFD enumeration on a
GPU node), attach hands out those IDs so the node path operates on devices
that actually exist — the bench's single-node configuration.
"""

from __future__ import annotations

import threading
import time
import uuid
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .base import (
    DeviceInfo,
    FabricError,
    FabricProvider,
    WaitingDeviceAttaching,
    WaitingDeviceDetaching,
)


@dataclass
class _PoolDevice:
    device_id: str
    cdi_device_id: str
    model: str
    attached_node: str = ""  # "" = free in the pool
    healthy: bool = True
    attach_deadline: float = 0.0  # monotonic time when an async attach lands
    detach_deadline: float = 0.0


@dataclass
class MockFabricConfig:
    attach_latency: float = 0.0  # seconds the fabric takes to compose
    detach_latency: float = 0.0
    asynchronous: bool = False  # True → CM-style resize+poll; False → FM-style
    fail_attach: int = 0  # fail the next N attach calls
    fail_detach: int = 0
    unhealthy_devices: set = field(default_factory=set)


class MockFabric(FabricProvider):
    name = "mock"

    def __init__(
        self,
        models: Optional[Dict[str, int]] = None,
        config: Optional[MockFabricConfig] = None,
        bind_inventory: Optional[List[Dict[str, str]]] = None,
    ):
        """models: model name -> pool size (ignored when bind_inventory given).

        bind_inventory: [{"device_id":..., "cdi_device_id":..., "model":...}]
        — real enumerated devices to hand out (single-node GPU bench).
        """
        self.config = config or MockFabricConfig()
        self._lock = threading.Lock()
        self._pool: Dict[str, _PoolDevice] = {}
        if bind_inventory:
            for d in bind_inventory:
                dev = _PoolDevice(
                    device_id=d["device_id"],
                    cdi_device_id=d.get("cdi_device_id", d["device_id"]),
                    model=d.get("model", "mi355x"),
                )
                self._pool[dev.device_id] = dev
        else:
            for model, count in (models or {"mi355x": 8}).items():
                for _ in range(count):
                    did = f"GPU-{uuid.uuid4()}"
                    self._pool[did] = _PoolDevice(
                        device_id=did, cdi_device_id=f"amd.com/gpu={did}", model=model
                    )

    # -- FabricProvider ----------------------------------------------------

    def add_resource(self, resource):
        with self._lock:
            if self.config.fail_attach > 0:
                self.config.fail_attach -= 1
                raise FabricError("mock fabric: injected attach failure")
            now = time.monotonic()
            # an in-flight async attach for this node+model?
            for dev in self._pool.values():
                if dev.attached_node == resource.spec.target_node and dev.attach_deadline:
                    if now >= dev.attach_deadline:
                        dev.attach_deadline = 0.0
                        return dev.device_id, dev.cdi_device_id
                    raise WaitingDeviceAttaching(
                        f"device {dev.device_id} still composing"
                    )
            free = [
                d
                for d in self._pool.values()
                if not d.attached_node and d.healthy and d.model == resource.spec.model
            ]
            if not free:
                raise FabricError(
                    f"mock fabric: no free {resource.spec.model} device in pool"
                )
            dev = free[0]
            dev.attached_node = resource.spec.target_node
            if self.config.asynchronous and self.config.attach_latency > 0:
                dev.attach_deadline = now + self.config.attach_latency
                raise WaitingDeviceAttaching(f"device {dev.device_id} composing")
            if self.config.attach_latency > 0:
                time.sleep(self.config.attach_latency)  # synchronous FM-style RTT
            return dev.device_id, dev.cdi_device_id

    def remove_resource(self, resource) -> None:
        with self._lock:
            if self.config.fail_detach > 0:
                self.config.fail_detach -= 1
                raise FabricError("mock fabric: injected detach failure")
            dev = self._pool.get(resource.status.device_id)
            if dev is None or not dev.attached_node:
                return  # idempotent: already gone (fti/fm/client.go:231-242)
            now = time.monotonic()
            if self.config.asynchronous and self.config.detach_latency > 0:
                if not dev.detach_deadline:
                    dev.detach_deadline = now + self.config.detach_latency
                    raise WaitingDeviceDetaching(f"device {dev.device_id} detaching")
                if now < dev.detach_deadline:
                    raise WaitingDeviceDetaching(f"device {dev.device_id} detaching")
                dev.detach_deadline = 0.0
            elif self.config.detach_latency > 0:
                time.sleep(self.config.detach_latency)
            dev.attached_node = ""

    def check_resource(self, resource) -> None:
        with self._lock:
            did = resource.status.device_id
            if did in self.config.unhealthy_devices:
                raise FabricError(f"mock fabric: device {did} status Critical")
            dev = self._pool.get(did)
            if dev is not None and not dev.healthy:
                raise FabricError(f"mock fabric: device {did} unhealthy")

    def get_resources(self) -> List[DeviceInfo]:
        with self._lock:
            now = time.monotonic()
            for d in self._pool.values():
                # an async compose completes fabric-side whether or not the
                # requester is still around (CM resize semantics): once the
                # deadline passes the device is attached ground truth — the
                # syncer must see it or an abandoned mid-compose attach
                # (CR deleted while WaitingDeviceAttaching) leaks forever
                if d.attached_node and d.attach_deadline and now >= d.attach_deadline:
                    d.attach_deadline = 0.0
            return [
                DeviceInfo(
                    node_name=d.attached_node,
                    device_type="gpu",
                    model=d.model,
                    device_id=d.device_id,
                    cdi_device_id=d.cdi_device_id,
                )
                for d in self._pool.values()
                if d.attached_node and not d.attach_deadline
            ]

    # -- test helpers ------------------------------------------------------

    def attached_to(self, node: str) -> List[str]:
        with self._lock:
            return [d.device_id for d in self._pool.values() if d.attached_node == node]

    def force_attach(self, device_id: str, node: str) -> None:
        """Simulate out-of-band fabric drift (device composed behind our back)."""
        with self._lock:
            self._pool[device_id].attached_node = node
