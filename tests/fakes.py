"""Fake fabric-manager HTTP servers for backend tests.

The httpx.MockTransport analog of the reference's httptest TLS server
(composableresource_controller_test.go:737-997): one handler emulates the
complete id_manager + CM/FM REST surface, with scenario switches for the
failure personas (bad credentials, non-JSON bodies, attach/detach failures,
warning/critical health).
"""

from __future__ import annotations

import base64
import json
import time
from typing import Dict, Optional

import httpx


def make_jwt(exp: Optional[float] = None) -> str:
    """Structurally valid JWT with an exp claim (fti/token.go:158-172 seam)."""
    if exp is None:
        exp = time.time() + 3600
    header = base64.urlsafe_b64encode(b'{"alg":"none"}').rstrip(b"=").decode()
    payload = (
        base64.urlsafe_b64encode(json.dumps({"exp": int(exp)}).encode())
        .rstrip(b"=")
        .decode()
    )
    return f"{header}.{payload}.sig"


class FakeFTIServer:
    """CM + FM + id_manager in one transport handler.

    Scenario knobs mirror the reference fake's machine-UUID-encoded
    behaviors; here they are attributes.
    """

    def __init__(self):
        self.token_calls = 0
        self.token_persona = "ok"  # ok|bad-creds|non-json|malformed-jwt
        self.token_exp: Optional[float] = None
        # machine_id -> CM machine dict (the "data" payload)
        self.cm_machines: Dict[str, dict] = {}
        # machine_id -> FM machine dict (the "data" payload)
        self.fm_machines: Dict[str, dict] = {}
        self.resize_calls = []
        self.fm_update_calls = []
        self.fail_resize = False
        self.fm_scaleup_response: Optional[dict] = None
        self.fm_scaleup_status = 200
        self.fm_scaledown_status = 200

    # -- CM payload builders (machine.go schema) ---------------------------

    @staticmethod
    def cm_machine(spec_uuid="spec-1", rtype="gpu", model="mi355x", devices=None, device_count=None):
        devices = devices or []
        return {
            "tenant_uuid": "tenant-1",
            "cluster": {
                "cluster_uuid": "cluster-1",
                "machine": {
                    "uuid": "",
                    "name": "m",
                    "status": "",
                    "status_reason": "",
                    "resspecs": [
                        {
                            "spec_uuid": spec_uuid,
                            "type": rtype,
                            "selector": {
                                "version": "v1",
                                "expression": {
                                    "conditions": [
                                        {"column": "model", "operator": "eq", "value": model}
                                    ]
                                },
                            },
                            "min_resspec_count": 0,
                            "max_resspec_count": 8,
                            "device_count": device_count if device_count is not None else len(devices),
                            "devices": devices,
                        }
                    ],
                },
            },
        }

    @staticmethod
    def cm_device(device_id, status="ADD_COMPLETE", res_uuid=None, op_status="0", reason=""):
        return {
            "device_id": device_id,
            "status": status,
            "status_reason": reason,
            "detail": {
                "fabric_uuid": "f",
                "fabric_id": 1,
                "res_uuid": res_uuid or f"res-{device_id}",
                "res_op_status": op_status,
            },
        }

    # -- FM payload builders (fm/api schema) -------------------------------

    @staticmethod
    def fm_machine(resources=None):
        return {
            "machines": [
                {
                    "fabric_uuid": "f",
                    "fabric_id": 1,
                    "mach_uuid": "m",
                    "mach_id": 1,
                    "mach_name": "m",
                    "tenant_uuid": "tenant-1",
                    "mach_status": 0,
                    "mach_status_detail": "",
                    "resources": resources or [],
                }
            ]
        }

    @staticmethod
    def fm_resource(serial, res_uuid=None, model="mi355x", op_status="0", rtype="gpu"):
        return {
            "res_uuid": res_uuid or f"res-{serial}",
            "res_name": serial,
            "res_type": rtype,
            "res_status": 0,
            "res_op_status": op_status,
            "res_serial_num": serial,
            "res_spec": {
                "condition": [{"column": "model", "operator": "eq", "value": model}]
            },
        }

    # -- transport handler -------------------------------------------------

    def handler(self, request: httpx.Request) -> httpx.Response:
        path = request.url.path

        if "id_manager" in path and path.endswith("/token"):
            self.token_calls += 1
            from urllib.parse import parse_qsl

            self.last_token_request = dict(
                parse_qsl(request.content.decode(errors="replace"))
            )
            if self.token_persona == "bad-creds":
                return httpx.Response(401, json={"error": "invalid_grant"})
            if self.token_persona == "non-json":
                return httpx.Response(200, text="not json at all")
            if self.token_persona == "malformed-jwt":
                return httpx.Response(200, json={"access_token": "garbage", "token_type": "Bearer"})
            return httpx.Response(
                200,
                json={
                    "access_token": make_jwt(self.token_exp),
                    "expires_in": 3600,
                    "token_type": "Bearer",
                },
            )

        if "cluster_manager" in path:
            machine_id = path.split("/machines/")[1].split("/")[0]
            if path.endswith("/actions/resize"):
                self.resize_calls.append((machine_id, json.loads(request.content)))
                if self.fail_resize:
                    return httpx.Response(500, json={"status": 500, "detail": {"code": "E500", "message": "boom"}})
                return httpx.Response(202, json={})
            machine = self.cm_machines.get(machine_id)
            if machine is None:
                return httpx.Response(404, json={"status": 404, "detail": {"code": "E404", "message": "no machine"}})
            return httpx.Response(200, json={"data": machine})

        if "fabric_manager" in path:
            machine_id = path.split("/machines/")[1].split("/")[0]
            if path.endswith("/update"):
                self.fm_update_calls.append(
                    (request.method, machine_id, json.loads(request.content))
                )
                if request.method == "PATCH":
                    if self.fm_scaleup_status != 200:
                        return httpx.Response(
                            self.fm_scaleup_status,
                            json={"status": self.fm_scaleup_status, "detail": {"code": "E1", "message": "scaleup failed", "data": {}}},
                        )
                    return httpx.Response(200, json={"data": self.fm_scaleup_response or {"machines": []}})
                if request.method == "DELETE":
                    if self.fm_scaledown_status not in (200, 204):
                        return httpx.Response(
                            self.fm_scaledown_status,
                            json={"status": self.fm_scaledown_status, "detail": {"code": "E2", "message": "scaledown failed", "data": {}}},
                        )
                    return httpx.Response(self.fm_scaledown_status, json={"data": {}})
            machine = self.fm_machines.get(machine_id)
            if machine is None:
                return httpx.Response(404, json={"status": 404, "detail": {"code": "E404", "message": "no machine", "data": {}}})
            return httpx.Response(200, json={"data": machine})

        return httpx.Response(404, text=f"unhandled path {path}")

    def transport(self) -> httpx.MockTransport:
        return httpx.MockTransport(self.handler)


class FakeNECServer:
    """Configuration-manager + layout-apply surface (nec/client.go seams)."""

    def __init__(self):
        self.nodes = []  # /nodes?detail=true payload entries
        self.resources = []  # /resources?detail=true payload entries
        self.applies: Dict[str, dict] = {}
        self.apply_counter = 0
        # sequence of statuses each new apply walks through on successive polls
        self.apply_status_script = ["COMPLETED"]
        self.post_conflict = False  # 409 E40010 persona
        self.layout_calls = []

    @staticmethod
    def gpu(device_id, model="mi355x", links=None, state="Enabled", health="OK"):
        return {
            "device": {
                "deviceID": device_id,
                "type": "GPU",
                "model": model,
                "attribute": {},
                "status": {"state": state, "health": health},
                "powerState": "On",
                "links": links or [],
            },
            "detected": True,
            "nodeIDs": [],
        }

    @staticmethod
    def adapter(device_id, adapter_type, status, links=None):
        return {
            "device": {
                "deviceID": device_id,
                "type": adapter_type,
                "model": "",
                "attribute": {"deviceSpecificInformation": {"status": status}},
                "status": {"state": "Enabled", "health": "OK"},
                "powerState": "On",
                "links": links or [],
            },
            "detected": True,
            "nodeIDs": [],
        }

    def handler(self, request: httpx.Request) -> httpx.Response:
        path = request.url.path
        if path.endswith("/nodes"):
            return httpx.Response(200, json={"count": len(self.nodes), "nodes": self.nodes})
        if path.endswith("/resources"):
            return httpx.Response(
                200, json={"count": len(self.resources), "resources": self.resources}
            )
        if "/resources/" in path:
            rid = path.rsplit("/", 1)[1]
            for res in self.resources:
                if res["device"]["deviceID"] == rid:
                    return httpx.Response(200, json=res)
            return httpx.Response(404, text=f"resource {rid} not found")
        if path.endswith("/layout-apply") and request.method == "POST":
            self.layout_calls.append(json.loads(request.content))
            if self.post_conflict:
                return httpx.Response(409, text='{"code": "E40010", "message": "Already running"}')
            self.apply_counter += 1
            apply_id = f"apply-{self.apply_counter}"
            self.applies[apply_id] = {"script": list(self.apply_status_script), "polls": 0}
            return httpx.Response(200, json={"applyID": apply_id})
        if "/layout-apply/" in path:
            apply_id = path.rsplit("/", 1)[1]
            entry = self.applies.get(apply_id)
            if entry is None:
                return httpx.Response(404, text="no such apply")
            idx = min(entry["polls"], len(entry["script"]) - 1)
            entry["polls"] += 1
            return httpx.Response(
                200, json={"applyID": apply_id, "status": entry["script"][idx]}
            )
        return httpx.Response(404, text=f"unhandled path {path}")

    def transport(self) -> httpx.MockTransport:
        return httpx.MockTransport(self.handler)
