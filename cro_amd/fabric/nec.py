"""NEC CDIM backend.

Protocol parity with internal/cdi/nec/client.go:

* endpoints built from ``NEC_CDIM_IP`` + ``LAYOUT_APPLY_PORT`` /
  ``CONFIGURATION_MANAGER_PORT`` → ``http://ip:port/cdim/api/v1`` (:648-659);
* attach (:131-195): pick a free healthy GPU from ``/resources?detail=true``
  (detected, type gpu, not linked to an ``eeio`` adapter, enabled/ok,
  model match), resolve the node's FabricIODevice by walking
  sourceFabricAdapter(eesv) → destinationFabricAdapter(eeio) links
  (:483-557), POST a ``connect`` procedure to ``/layout-apply`` and poll
  ``/layout-apply/{id}`` up to 6×10 s for COMPLETED/FAILED (:352-377);
* 409 + ``E40010`` ("already running") maps to the waiting sentinels (:379-387);
* NEC CDIM does not expose GPU UUIDs — a provisional UUID from
  ``NEC_PROVISIONAL_GPU_UUID`` stands in (:186-194,712-723);
* node identity: kubernetes ``spec.providerID`` == CDIM node id (:437-463).
"""

from __future__ import annotations

import logging
import os
import time
from typing import List, Optional, Tuple

import httpx

from ..api.v1alpha1.types import ComposableResource, Node
from .base import (
    DeviceInfo,
    FabricError,
    FabricProvider,
    WaitingDeviceAttaching,
    WaitingDeviceDetaching,
)

log = logging.getLogger(__name__)

REQUEST_TIMEOUT = 30.0
DEFAULT_POLL_INTERVAL = 10.0
DEFAULT_POLL_ATTEMPTS = 6


def build_endpoint(ip: str, port: str) -> str:
    if not ip or not port:
        raise ValueError(f"env vars are required: NEC_CDIM_IP='{ip}', port='{port}'")
    return f"http://{ip}:{port}/cdim/api/v1"


def _provisional_gpu_uuid() -> str:
    value = os.environ.get("NEC_PROVISIONAL_GPU_UUID", "")
    if not value:
        raise FabricError(
            "NEC_PROVISIONAL_GPU_UUID is required for NEC prototype mode "
            "(example: GPU-xxxxxxxx-xxxx-xxxx-xxxx-xxxxxxxxxxxx)"
        )
    if not value.upper().startswith("GPU-"):
        value = "GPU-" + value
    return value


def _healthy(state: str, health: str) -> bool:
    return state.lower() == "enabled" and health.lower() == "ok"


def _is_gpu(device_type: str, requested: str) -> bool:
    if requested and requested.lower() != "gpu":
        return False
    return device_type.lower() == "gpu"


def _connected_to_eeio(links: list) -> bool:
    return any(link.get("type", "").lower() == "eeio" for link in links)


def _device_specific_status(device: dict) -> str:
    info = device.get("attribute", {}).get("deviceSpecificInformation", {})
    if not isinstance(info, dict):
        return ""
    return str(info.get("status", ""))


class NECClient(FabricProvider):
    name = "nec"

    def __init__(
        self,
        client,
        ip: Optional[str] = None,
        layout_apply_port: Optional[str] = None,
        configuration_manager_port: Optional[str] = None,
        transport: Optional[httpx.BaseTransport] = None,
        poll_interval: Optional[float] = None,
        poll_attempts: int = DEFAULT_POLL_ATTEMPTS,
    ):
        self.client = client
        ip = ip if ip is not None else os.environ.get("NEC_CDIM_IP", "")
        lport = layout_apply_port if layout_apply_port is not None else os.environ.get("LAYOUT_APPLY_PORT", "")
        cport = (
            configuration_manager_port
            if configuration_manager_port is not None
            else os.environ.get("CONFIGURATION_MANAGER_PORT", "")
        )
        self.layout_apply_endpoint = build_endpoint(ip, lport)
        self.configuration_manager_endpoint = build_endpoint(ip, cport)
        self.transport = transport
        self.poll_interval = (
            poll_interval
            if poll_interval is not None
            else float(os.environ.get("CRO_NEC_POLL_INTERVAL", DEFAULT_POLL_INTERVAL))
        )
        self.poll_attempts = poll_attempts
        self._http = httpx.Client(transport=transport, timeout=REQUEST_TIMEOUT)

    # -- HTTP --------------------------------------------------------------

    def _do(self, endpoint: str, method: str, path: str, payload=None) -> dict:
        resp = self._http.request(method, endpoint + path, json=payload)
        if not 200 <= resp.status_code < 300:
            raise FabricError(
                f"request failed: method={method} path={path} "
                f"status={resp.status_code} body={resp.text}"
            )
        if not resp.content:
            return {}
        try:
            return resp.json()
        except ValueError as exc:
            raise FabricError(f"failed to unmarshal {path} response: {exc}")

    # -- CDIM queries ------------------------------------------------------

    def _all_resources(self) -> list:
        return self._do(
            self.configuration_manager_endpoint, "GET", "/resources?detail=true"
        ).get("resources", [])

    def _all_nodes(self) -> list:
        return self._do(
            self.configuration_manager_endpoint, "GET", "/nodes?detail=true"
        ).get("nodes", [])

    def _resource_by_id(self, rid: str) -> dict:
        body = self._do(self.configuration_manager_endpoint, "GET", f"/resources/{rid}")
        return body.get("resource", body)

    def _node_id_for(self, node_name: str) -> str:
        node = self.client.get(Node, node_name)
        provider_id = node.status.provider_id
        if not provider_id:
            raise FabricError(f"node {node_name} has no providerID")
        for entry in self._all_nodes():
            if entry.get("id", "").lower() == provider_id.lower():
                return entry["id"]
        raise FabricError(f"node id not found: {provider_id}")

    def _k8s_node_for(self, nec_node_id: str) -> str:
        for node in self.client.list(Node):
            if node.status.provider_id.lower() == nec_node_id.lower():
                return node.metadata.name
        raise FabricError(f"kubernetes node not found for NEC node ID: {nec_node_id}")

    # -- fabric link walk (:483-557) ---------------------------------------

    def _resolve_attach_fabric_io_device(self, node_id: str) -> str:
        target = None
        for node in self._all_nodes():
            if node.get("id", "").lower() == node_id.lower():
                target = node
                break
        if target is None:
            raise FabricError(f"node not found while resolving attach destination: {node_id}")

        host_device_id = ""
        for res in target.get("resources", []):
            if not res.get("detected"):
                continue
            device = res.get("device", {})
            if (
                device.get("type", "").lower() == "sourcefabricadapter"
                and _device_specific_status(device).lower() == "eesv"
            ):
                host_device_id = device.get("deviceID", "")
                if host_device_id:
                    break
        if not host_device_id:
            raise FabricError(
                f"failed to resolve FabricHostDevice id from node resources: node={node_id}"
            )

        host = self._resource_by_id(host_device_id)
        io_device_id = ""
        for link in host.get("device", {}).get("links", []):
            if link.get("type", "").lower() == "destinationfabricadapter" and link.get("deviceID"):
                io_device_id = link["deviceID"]
                break
        if not io_device_id:
            raise FabricError(
                "failed to resolve FabricIODevice id from FabricHostDevice resource "
                f"links: resourceID={host_device_id}"
            )

        io_res = self._resource_by_id(io_device_id)
        io_dev = io_res.get("device", {})
        if not (
            io_dev.get("type", "").lower() == "destinationfabricadapter"
            and _device_specific_status(io_dev).lower() == "eeio"
        ):
            raise FabricError(
                f"linked resource is not a FabricIODevice: resourceID={io_dev.get('deviceID')} "
                f"type={io_dev.get('type')}"
            )
        return io_device_id

    # -- layout apply ------------------------------------------------------

    def _post_layout_apply(self, operation: str, source: str, destination: str) -> str:
        payload = {
            "procedures": [
                {
                    "operationID": 1,
                    "operation": operation,
                    "sourceDeviceID": source,
                    "destinationDeviceID": destination,
                    "dependencies": [],
                }
            ]
        }
        body = self._do(self.layout_apply_endpoint, "POST", "/layout-apply", payload)
        apply_id = body.get("applyID", "")
        if not apply_id:
            raise FabricError("/layout-apply response does not contain applyID")
        return apply_id

    def _wait_layout_apply(self, apply_id: str, waiting_exc) -> None:
        for attempt in range(self.poll_attempts):
            body = self._do(self.layout_apply_endpoint, "GET", f"/layout-apply/{apply_id}")
            status = str(body.get("status", "")).upper()
            if status == "COMPLETED":
                return
            if status in ("IN_PROGRESS", "CANCELING", ""):
                if attempt < self.poll_attempts - 1:
                    time.sleep(self.poll_interval)
                    continue
                raise waiting_exc(f"layout-apply {apply_id} still {status or 'pending'}")
            if status in ("FAILED", "SUSPENDED", "CANCELED"):
                raise FabricError(
                    f"layout-apply failed: applyID={apply_id} status={body.get('status')} "
                    f"rollbackStatus={body.get('rollbackStatus', '')}"
                )
            raise FabricError(
                f"layout-apply returned unknown status: applyID={apply_id} "
                f"status={body.get('status')}"
            )
        raise waiting_exc(f"layout-apply {apply_id} still pending")

    @staticmethod
    def _already_running(exc: Exception) -> bool:
        msg = str(exc)
        return "status=409" in msg and "E40010" in msg  # E40010: already running

    # -- FabricProvider ----------------------------------------------------

    def add_resource(self, resource: ComposableResource) -> Tuple[str, str]:
        node_name = resource.spec.target_node
        if not node_name:
            raise FabricError("spec.target_node (kubernetes node name) is required")

        resources = self._all_resources()
        node_id = self._node_id_for(node_name)
        io_device_id = self._resolve_attach_fabric_io_device(node_id)

        target = None
        for res in resources:
            device = res.get("device", {})
            if not res.get("detected"):
                continue
            if not _is_gpu(device.get("type", ""), resource.spec.type):
                continue
            if _connected_to_eeio(device.get("links", [])):
                continue
            if not _healthy(
                device.get("status", {}).get("state", ""),
                device.get("status", {}).get("health", ""),
            ):
                continue
            if resource.spec.model and device.get("model", "").lower() != resource.spec.model.lower():
                continue
            target = device
            break
        if target is None:
            raise FabricError(
                f"no available GPU found for node={node_id} model={resource.spec.model} "
                f"type={resource.spec.type}"
            )
        gpu_device_id = target.get("deviceID", "")
        if not gpu_device_id:
            raise FabricError("gpu deviceID is empty for selected resource")

        try:
            apply_id = self._post_layout_apply("connect", io_device_id, gpu_device_id)
        except FabricError as exc:
            if self._already_running(exc):
                raise WaitingDeviceAttaching(str(exc))
            raise
        self._wait_layout_apply(apply_id, WaitingDeviceAttaching)
        return _provisional_gpu_uuid(), gpu_device_id

    def remove_resource(self, resource: ComposableResource) -> None:
        rid = resource.status.cdi_device_id
        if not rid:
            raise FabricError("status.cdi_device_id is required")
        res = self._resource_by_id(rid)
        io_device_id = ""
        for link in res.get("device", {}).get("links", []):
            if link.get("type", "").lower() == "destinationfabricadapter":
                io_device_id = link.get("deviceID", "")
                break
        if not io_device_id:
            log.info("GPU already detached; destinationFabricAdapter link not found")
            return
        try:
            apply_id = self._post_layout_apply("disconnect", io_device_id, rid)
        except FabricError as exc:
            if self._already_running(exc):
                raise WaitingDeviceDetaching(str(exc))
            raise
        self._wait_layout_apply(apply_id, WaitingDeviceDetaching)

    def check_resource(self, resource: ComposableResource) -> None:
        rid = resource.status.cdi_device_id
        if not rid:
            raise FabricError("status.cdi_device_id is required")
        res = self._resource_by_id(rid)
        status = res.get("device", {}).get("status", {})
        if _healthy(status.get("state", ""), status.get("health", "")):
            return
        raise FabricError(
            f"resource is not healthy: id={rid} status={status.get('state')} "
            f"health={status.get('health')}"
        )

    def get_resources(self) -> List[DeviceInfo]:
        provisional = _provisional_gpu_uuid()
        infos: List[DeviceInfo] = []
        for node in self._all_nodes():
            node_id = node.get("id", "")
            if not node_id:
                continue
            try:
                k8s_name = self._k8s_node_for(node_id)
            except FabricError as exc:
                log.error("%s", exc)
                continue
            for res in node.get("resources", []):
                if not res.get("detected"):
                    continue
                device = res.get("device", {})
                if not _is_gpu(device.get("type", ""), "gpu"):
                    continue
                infos.append(
                    DeviceInfo(
                        node_name=k8s_name,
                        machine_uuid=node_id,
                        device_type=device.get("type", "").lower(),
                        model=device.get("model", ""),
                        device_id=provisional,
                        cdi_device_id=device.get("deviceID", ""),
                    )
                )
        return infos
