"""Sunfish / Redfish backend — a minimal composition-service prototype.

Parity with internal/cdi/sunfish/client.go:30-146: PATCH
``http://{endpoint}/redfish/v1/Systems/System`` with
``{Name: targetNode, Processors: {Members: [{@Redfish.RequestCount,
ProcessorType: GPU, Model}]}}``; count 1 attaches, 0 detaches;
``check_resource``/``get_resources`` are no-ops.

The supported-model list is the MI355X build's own (the reference
hard-codes V100/A100 — AMD models belong here).
"""

from __future__ import annotations

import logging
import os
from typing import List, Optional, Tuple

import httpx

from ..api.v1alpha1.types import ComposableResource
from .base import DeviceInfo, FabricError, FabricProvider

log = logging.getLogger(__name__)

PROCESSOR_TYPE_GPU = "GPU"
SUPPORTED_MODELS = (
    "AMD-Instinct-MI355X",
    "AMD-Instinct-MI350X",
    "mi355x",
    "mi350x",
)
DEFAULT_ENDPOINT = "composition-service.cro-system.svc.cluster.local:5060"


class SunfishClient(FabricProvider):
    name = "sunfish"

    def __init__(
        self,
        endpoint: Optional[str] = None,
        transport: Optional[httpx.BaseTransport] = None,
    ):
        self.endpoint = endpoint or os.environ.get("SUNFISH_ENDPOINT", DEFAULT_ENDPOINT)
        self.transport = transport
        self._http = httpx.Client(transport=transport, timeout=30.0)

    def _patch(self, body: dict) -> None:
        url = f"http://{self.endpoint}/redfish/v1/Systems/System"
        resp = self._http.patch(url, json=body)
        if resp.status_code not in (200, 204):
            raise FabricError(f"http returned code {resp.status_code}")

    def _composition_request(self, resource: ComposableResource, count: int) -> dict:
        member = {}
        if resource.spec.model in SUPPORTED_MODELS:
            member = {
                "@Redfish.RequestCount": count,
                "ProcessorType": PROCESSOR_TYPE_GPU,
                "Model": resource.spec.model,
            }
        return {
            "Name": resource.spec.target_node,
            "Processors": {"Members": [member]},
        }

    def add_resource(self, resource: ComposableResource) -> Tuple[str, str]:
        self._patch(self._composition_request(resource, 1))
        # the Redfish prototype returns no device identity (sunfish/client.go:77-103)
        return "", ""

    def remove_resource(self, resource: ComposableResource) -> None:
        self._patch(self._composition_request(resource, 0))

    def check_resource(self, resource: ComposableResource) -> None:
        return None

    def get_resources(self) -> List[DeviceInfo]:
        return []
