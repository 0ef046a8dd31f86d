"""FTI Cluster Manager (CM) backend — the asynchronous OpenShift path.

Protocol parity with internal/cdi/fti/cm/client.go:

* attach (``AddResource``, :114-187): GET machine info; if an unused
  ``ADD_COMPLETE`` device already exists on a matching resource spec, adopt
  it; else POST ``.../actions/resize`` with ``device_count+1`` and raise
  :class:`WaitingDeviceAttaching` (the controller re-polls);
* detach (:189-264): resize with the explicit device list and
  ``device_count-1``; raises :class:`WaitingDeviceDetaching` after accept;
* health (:266-317): the device's ``res_op_status`` first digit — 0 OK,
  1 Warning, 2 Critical;
* inventory (:319-361): walk every node's machine.

REST base: ``cluster_manager/cluster_autoscaler/v3/tenants/{t}/clusters/
{c}/machines/{m}``; bearer token from the shared cache; 60 s timeout.
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional, Tuple

import httpx

from ...api.v1alpha1.types import ComposableResource, Node
from ..base import (
    DeviceInfo,
    FabricError,
    FabricProvider,
    WaitingDeviceAttaching,
    WaitingDeviceDetaching,
)
from .machines import resolve_machine_id_openshift
from .token import CachedToken

CM_REQUEST_TIMEOUT = 60.0
ADD_COMPLETE = "ADD_COMPLETE"
ADD_FAILED = "ADD_FAILED"
REMOVE_FAILED = "REMOVE_FAILED"

STATUS_OK = "0"
STATUS_WARNING = "1"
STATUS_CRITICAL = "2"


class FTICMClient(FabricProvider):
    name = "fti-cm"

    def __init__(
        self,
        client,
        endpoint: Optional[str] = None,
        tenant_id: Optional[str] = None,
        cluster_id: Optional[str] = None,
        token: Optional[CachedToken] = None,
        transport: Optional[httpx.BaseTransport] = None,
        verify: bool = True,
    ):
        self.client = client
        endpoint = endpoint if endpoint is not None else os.environ.get("FTI_CDI_ENDPOINT", "")
        if not endpoint.endswith("/"):
            endpoint += "/"
        self.endpoint = endpoint
        self.tenant_id = tenant_id if tenant_id is not None else os.environ.get("FTI_CDI_TENANT_ID", "")
        self.cluster_id = cluster_id if cluster_id is not None else os.environ.get("FTI_CDI_CLUSTER_ID", "")
        self.transport = transport
        self.verify = verify
        self.token = token or CachedToken(endpoint, transport=transport, verify=verify)
        # persistent connection pool: health checks run every 30 s per
        # device — a TLS handshake per call would dominate fabric RTT
        self._http = httpx.Client(
            transport=transport, verify=verify, timeout=CM_REQUEST_TIMEOUT
        )

    # -- HTTP plumbing -----------------------------------------------------

    def _machine_path(self, machine_id: str, action: str = "") -> str:
        base = (
            f"https://{self.endpoint}cluster_manager/cluster_autoscaler/v3/"
            f"tenants/{self.tenant_id}/clusters/{self.cluster_id}/machines/{machine_id}"
        )
        return base + action

    def _request(self, method: str, url: str, json_body=None) -> httpx.Response:
        headers = {
            "Authorization": f"Bearer {self.token.get_token()}",
            "Content-Type": "application/json",
        }
        return self._http.request(method, url, json=json_body, headers=headers)

    def _get_machine_info(self, machine_id: str) -> dict:
        resp = self._request("GET", self._machine_path(machine_id))
        if not 200 <= resp.status_code < 300:
            raise FabricError(
                f"failed to process CM get request. http returned status: {resp.status_code}"
            )
        try:
            return resp.json()["data"]
        except (ValueError, KeyError) as exc:
            raise FabricError(
                f"failed to unmarshal CM get machine response body into machineData: {exc}"
            )

    def _machine_id(self, node_name: str) -> str:
        return resolve_machine_id_openshift(self.client, node_name)

    # -- FabricProvider ----------------------------------------------------

    def add_resource(self, resource: ComposableResource) -> Tuple[str, str]:
        machine_id = self._machine_id(resource.spec.target_node)
        data = self._get_machine_info(machine_id)

        existing = {
            r.status.device_id: True
            for r in self.client.list(ComposableResource)
            if r.status.device_id
        }

        spec_uuid, device_count = "", 0
        for spec in data["cluster"]["machine"].get("resspecs", []):
            if not _spec_matches(spec, resource):
                continue
            unused = _find_available_device(spec, existing)
            if unused is not None:
                if unused.get("status") == ADD_COMPLETE:
                    return unused["device_id"], unused.get("detail", {}).get("res_uuid", "")
                if unused.get("status") == ADD_FAILED:
                    raise FabricError(
                        "an error occurred with the resource in CM: "
                        f"'{unused.get('status_reason', '')}'"
                    )
            spec_uuid = spec.get("spec_uuid", "")
            device_count = spec.get("device_count", 0)
            break

        if not spec_uuid:
            raise FabricError(
                f"no resource spec in CM matches type={resource.spec.type} "
                f"model={resource.spec.model} on machine {machine_id}"
            )

        body = {
            "increase_resource_count": {
                "spec_uuid": spec_uuid,
                "device_count": device_count + 1,
            }
        }
        resp = self._request("POST", self._machine_path(machine_id, "/actions/resize"), body)
        if not 200 <= resp.status_code < 300:
            raise FabricError(
                f"failed to process CM scaleup request. http returned status: {resp.status_code}"
            )
        raise WaitingDeviceAttaching("CM resize accepted; device composing")

    def remove_resource(self, resource: ComposableResource) -> None:
        machine_id = self._machine_id(resource.spec.target_node)
        data = self._get_machine_info(machine_id)

        spec_uuid, device_count, failure = "", 0, None
        for spec in data["cluster"]["machine"].get("resspecs", []):
            if spec.get("type") != resource.spec.type:
                continue
            for device in spec.get("devices", []):
                if device.get("device_id") == resource.status.device_id:
                    if device.get("status") == REMOVE_FAILED:
                        failure = device.get("status_reason", "remove failed")
                    spec_uuid = spec.get("spec_uuid", "")
                    device_count = spec.get("device_count", 0)
                    break
            if spec_uuid:
                break
        if failure:
            # persist the upstream failure reason and STILL issue the resize
            # (cm/client.go:206-215: status update, then continue — the
            # caller's Waiting requeue would otherwise drop the message)
            resource.status.error = failure
            if self.client is not None:
                try:
                    fresh = self.client.get(ComposableResource, resource.metadata.name)
                    fresh.status.error = failure
                    self.client.update_status(fresh)
                except Exception:  # best-effort, like the reference's log-and-go
                    pass
        if not spec_uuid:
            return  # device already gone upstream — idempotent

        body = {
            "remove_resources": {
                "spec_uuid": spec_uuid,
                "device_count": device_count - 1,
                "devices": [resource.status.device_id],
            }
        }
        resp = self._request("POST", self._machine_path(machine_id, "/actions/resize"), body)
        if not 200 <= resp.status_code < 300:
            raise FabricError(
                f"failed to process CM scaledown request. http returned status: {resp.status_code}"
            )
        raise WaitingDeviceDetaching("CM resize accepted; device detaching")

    def check_resource(self, resource: ComposableResource) -> None:
        machine_id = self._machine_id(resource.spec.target_node)
        data = self._get_machine_info(machine_id)
        for spec in data["cluster"]["machine"].get("resspecs", []):
            if spec.get("type") != resource.spec.type:
                continue
            if not _spec_matches(spec, resource):
                continue
            for device in spec.get("devices", []):
                if device.get("device_id") != resource.status.device_id:
                    continue
                op_status = str(device.get("detail", {}).get("res_op_status", ""))
                if not op_status:
                    raise FabricError(
                        f"the target gpu '{resource.status.device_id}' on machine "
                        f"'{machine_id}' has empty status in CM"
                    )
                digit = op_status[:1]
                if digit == STATUS_OK:
                    return
                if digit == STATUS_WARNING:
                    raise FabricError(
                        f"the target gpu '{resource.status.device_id}' is showing a Warning status in CM"
                    )
                if digit == STATUS_CRITICAL:
                    raise FabricError(
                        f"the target gpu '{resource.status.device_id}' is showing a Critical status in CM"
                    )
                raise FabricError(
                    f"the target gpu '{resource.status.device_id}' has unknown status "
                    f"'{op_status}' in CM"
                )
        raise FabricError(
            f"the target device '{resource.status.device_id}' cannot be found in CDI system"
        )

    def get_resources(self) -> List[DeviceInfo]:
        infos: List[DeviceInfo] = []
        for node in self.client.list(Node):
            machine_id = self._machine_id(node.metadata.name)
            data = self._get_machine_info(machine_id)
            for spec in data["cluster"]["machine"].get("resspecs", []):
                if spec.get("type") != "gpu":
                    continue
                for device in spec.get("devices", []):
                    infos.append(
                        DeviceInfo(
                            node_name=node.metadata.name,
                            machine_uuid=machine_id,
                            device_type=spec.get("type", ""),
                            device_id=device.get("device_id", ""),
                            cdi_device_id=device.get("detail", {}).get("res_uuid", ""),
                        )
                    )
        return infos


def _spec_matches(spec: dict, resource: ComposableResource) -> bool:
    if spec.get("type") != resource.spec.type:
        return False
    conditions = spec.get("selector", {}).get("expression", {}).get("conditions", [])
    return any(
        c.get("column") == "model"
        and c.get("operator") == "eq"
        and c.get("value") == resource.spec.model
        for c in conditions
    )


def _find_available_device(spec: dict, existing: Dict[str, bool]) -> Optional[dict]:
    for device in spec.get("devices", []):
        if not existing.get(device.get("device_id", "")):
            return device
    return None
