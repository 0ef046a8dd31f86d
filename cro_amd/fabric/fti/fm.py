"""FTI Fabric Manager (FM) backend — the synchronous direct path.

Protocol parity with internal/cdi/fti/fm/client.go:

* attach (:100-214): PATCH ``fabric_manager/api/v1/machines/{m}/update``
  with the tenants/machines/res_specs body; the response carries the
  attached device's ``res_serial_num`` (GPU UUID) and ``res_uuid``
  immediately; ``res_op_status`` digit 0/1 accepted (1 logs Warning),
  2 fails;
* detach (:216-312): DELETE with the ``res_uuid``; skipped when the device
  is already gone upstream (idempotent, :231-242);
* health (:314-359): same status-digit scheme against GET machine info;
* node→machine: the OpenShift chain when ``FTI_CDI_CLUSTER_ID`` is set,
  else providerID ``fsas-cdi://`` (RKE2) — :416-464; 180 s timeout.
"""

from __future__ import annotations

import logging
import os
from typing import List, Optional, Tuple

import httpx

from ...api.v1alpha1.types import ComposableResource, Node
from ..base import DeviceInfo, FabricError, FabricProvider
from .machines import resolve_machine_id
from .token import CachedToken

FM_REQUEST_TIMEOUT = 180.0
log = logging.getLogger(__name__)


def _format_fm_error(detail: dict) -> str:
    return (
        f"code: {detail.get('code', '')}, message: {detail.get('message', '')}, "
        f"data: {detail.get('data', '')}"
    )


class FTIFMClient(FabricProvider):
    name = "fti-fm"

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
        # persistent connection pool (see cm.py)
        self._http = httpx.Client(
            transport=transport, verify=verify, timeout=FM_REQUEST_TIMEOUT
        )

    # -- HTTP plumbing -----------------------------------------------------

    def _request(self, method: str, machine_id: str, suffix: str = "", json_body=None) -> httpx.Response:
        url = f"https://{self.endpoint}fabric_manager/api/v1/machines/{machine_id}{suffix}"
        headers = {
            "Authorization": f"Bearer {self.token.get_token()}",
            "Content-Type": "application/json",
        }
        return self._http.request(
            method, url, params={"tenant_uuid": self.tenant_id},
            json=json_body, headers=headers,
        )

    def _machine_id(self, node_name: str) -> str:
        return resolve_machine_id(self.client, node_name, self.cluster_id)

    def _get_machine_info(self, machine_id: str) -> dict:
        resp = self._request("GET", machine_id)
        if resp.status_code != 200:
            detail = _error_detail(resp)
            raise FabricError(f"failed to process FM get request. FM returned {detail}")
        try:
            return resp.json()["data"]
        except (ValueError, KeyError) as exc:
            raise FabricError(
                f"failed to unmarshal FM get machine response body into machineData: {exc}"
            )

    # -- FabricProvider ----------------------------------------------------

    def add_resource(self, resource: ComposableResource) -> Tuple[str, str]:
        machine_id = self._machine_id(resource.spec.target_node)
        body = {
            "tenants": {
                "tenant_uuid": self.tenant_id,
                "machines": [
                    {
                        "mach_uuid": machine_id,
                        "resources": [
                            {
                                "res_specs": [
                                    {
                                        "res_type": resource.spec.type,
                                        "res_spec": {
                                            "condition": [
                                                {
                                                    "column": "model",
                                                    "operator": "eq",
                                                    "value": resource.spec.model,
                                                }
                                            ]
                                        },
                                        "res_num": 1,
                                    }
                                ]
                            }
                        ],
                    }
                ],
            }
        }
        resp = self._request("PATCH", machine_id, "/update", body)
        if resp.status_code != 200:
            raise FabricError(
                f"failed to process FM scaleup request. FM returned {_error_detail(resp)}"
            )
        try:
            machines = resp.json()["data"]["machines"]
        except (ValueError, KeyError) as exc:
            raise FabricError(
                f"failed to unmarshal FM scaleup response body into scaleUpResponse: {exc}"
            )
        if machines and machines[0].get("resources"):
            res = machines[0]["resources"][0]
            if res.get("res_type") == resource.spec.type:
                for cond in res.get("res_spec", {}).get("condition", []):
                    if (
                        cond.get("column") == "model"
                        and cond.get("operator") == "eq"
                        and cond.get("value") == resource.spec.model
                    ):
                        digit = str(res.get("res_op_status", ""))[:1]
                        if digit == "0":
                            return res["res_serial_num"], res["res_uuid"]
                        if digit == "1":
                            log.info("FM attached device is in Warning state")
                            return res["res_serial_num"], res["res_uuid"]
                        if digit == "2":
                            raise FabricError(
                                f"the FM attached device called by {resource.metadata.name} "
                                "is in Critical state in FM"
                            )
                        raise FabricError(
                            f"the FM attached device called by {resource.metadata.name} is in "
                            f"unknown state '{res.get('res_op_status', '')}' in FM"
                        )
        raise FabricError("can not find the added gpu when using FM to add gpu")

    def remove_resource(self, resource: ComposableResource) -> None:
        machine_id = self._machine_id(resource.spec.target_node)
        data = self._get_machine_info(machine_id)
        machines = data.get("machines", [])
        exists = any(
            r.get("res_type") == resource.spec.type
            and r.get("res_uuid") == resource.status.cdi_device_id
            for r in (machines[0].get("resources", []) if machines else [])
        )
        if not exists:
            log.info("resource does not exist in FM, skipping removal")
            return

        body = {
            "tenants": {
                "tenant_uuid": self.tenant_id,
                "machines": [
                    {
                        "mach_uuid": machine_id,
                        "resources": [
                            {
                                "res_specs": [
                                    {
                                        "res_type": resource.spec.type,
                                        "res_uuid": resource.status.cdi_device_id,
                                        "res_num": 1,
                                    }
                                ]
                            }
                        ],
                    }
                ],
            }
        }
        resp = self._request("DELETE", machine_id, "/update", body)
        if resp.status_code not in (200, 204):
            raise FabricError(
                f"failed to process FM scaledown request. FM returned {_error_detail(resp)}"
            )

    def check_resource(self, resource: ComposableResource) -> None:
        machine_id = self._machine_id(resource.spec.target_node)
        data = self._get_machine_info(machine_id)
        machines = data.get("machines", [])
        for res in (machines[0].get("resources", []) if machines else []):
            if res.get("res_type") != resource.spec.type:
                continue
            for cond in res.get("res_spec", {}).get("condition", []):
                if (
                    cond.get("column") != "model"
                    or cond.get("operator") != "eq"
                    or cond.get("value") != resource.spec.model
                ):
                    continue
                if res.get("res_serial_num") == resource.status.device_id:
                    digit = str(res.get("res_op_status", ""))[:1]
                    if digit == "0":
                        return
                    if digit == "1":
                        raise FabricError(
                            f"the target gpu '{resource.status.device_id}' is showing a Warning status in FM"
                        )
                    if digit == "2":
                        raise FabricError(
                            f"the target gpu '{resource.status.device_id}' is showing a Critical status in FM"
                        )
                    raise FabricError(
                        f"the target gpu '{resource.status.device_id}' has unknown status "
                        f"'{res.get('res_op_status', '')}' in FM"
                    )
        raise FabricError(
            f"the target device '{resource.status.device_id}' cannot be found in CDI system"
        )

    def get_resources(self) -> List[DeviceInfo]:
        infos: List[DeviceInfo] = []
        for node in self.client.list(Node):
            try:
                machine_id = self._machine_id(node.metadata.name)
                data = self._get_machine_info(machine_id)
            except Exception as exc:  # per-node failures skip, not abort
                log.error("failed to get machineInfo for node %s: %s", node.metadata.name, exc)
                continue
            machines = data.get("machines", [])
            if not machines:
                continue
            for res in machines[0].get("resources", []):
                if res.get("res_type") != "gpu":
                    continue
                model = ""
                for cond in res.get("res_spec", {}).get("condition", []):
                    if cond.get("column") == "model" and cond.get("operator") == "eq":
                        model = cond.get("value", "")
                        break
                infos.append(
                    DeviceInfo(
                        node_name=node.metadata.name,
                        machine_uuid=machine_id,
                        device_type=res.get("res_type", ""),
                        model=model,
                        device_id=res.get("res_serial_num", ""),
                        cdi_device_id=res.get("res_uuid", ""),
                    )
                )
        return infos


def _error_detail(resp: httpx.Response) -> str:
    try:
        detail = resp.json().get("detail", {})
    except ValueError:
        raise FabricError(
            "failed to unmarshal FM error response body into errBody. "
            f"Original body: {resp.text[:200]}"
        )
    return _format_fm_error(detail)
