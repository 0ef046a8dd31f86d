"""Container Device Interface spec emission for composed MI355X GPUs.

The reference has no CDI emission (NVIDIA device plugin/DRA driver own it);
BASELINE.json's north star makes it this operator's job: after a device is
composed and visible, write the CDI spec exposing ``/dev/kfd`` + the GPU's
``/dev/dri/renderD*``/``card*`` nodes so scheduled pods see the device, and
remove it on detach.

The spec also carries what the reference never surfaces:
xGMI topology, VRAM size and gfx target per device as CDI annotations, so
workload schedulers can co-place RCCL jobs on xGMI-connected GPUs.

Specs go through the NodeExec seam so the same writer works locally (node
agent) and over pod exec (cluster mode).  One spec file per node, updated
read-modify-write under a lock.
"""

from __future__ import annotations

import json
import threading
from typing import List, Optional

from .execs import NodeExec
from .kfd import GPUDevice

CDI_VERSION = "0.6.0"
CDI_KIND = "amd.com/gpu"
DEFAULT_CDI_DIR = "/etc/cdi"


def cdi_device_id(device_id: str) -> str:
    return f"{CDI_KIND}={device_id}"


class CDISpecWriter:
    def __init__(
        self,
        execer: NodeExec,
        cdi_dir: str = DEFAULT_CDI_DIR,
        kind: str = CDI_KIND,
        root_device_nodes=None,
    ):
        """``kind`` names the CDI device class (one spec file per kind);
        ``root_device_nodes`` are class-wide device nodes injected for every
        device (GPUs: /dev/kfd; CXL.mem: none)."""
        self.execer = execer
        self.cdi_dir = cdi_dir.rstrip("/")
        self.kind = kind
        self.root_device_nodes = (
            root_device_nodes if root_device_nodes is not None else ["/dev/kfd"]
        )
        self._lock = threading.Lock()

    def _spec_path(self, node: str) -> str:
        return f"{self.cdi_dir}/{self.kind.replace('/', '-')}-cro.json"

    def _load(self, node: str) -> dict:
        try:
            return json.loads(self.execer.read_file(node, self._spec_path(node)))
        except (FileNotFoundError, json.JSONDecodeError):
            return {
                "cdiVersion": CDI_VERSION,
                "kind": self.kind,
                "containerEdits": {
                    "deviceNodes": [{"path": p} for p in self.root_device_nodes]
                },
                "devices": [],
            }

    def _store(self, node: str, spec: dict) -> None:
        self.execer.write_file(node, self._spec_path(node), json.dumps(spec, indent=2))

    def add_device(self, node: str, gpu: GPUDevice) -> str:
        """Add (or refresh) one composed GPU; returns its CDI device id."""
        with self._lock:
            spec = self._load(node)
            device_nodes = [{"path": gpu.render_path}]
            # the card node is optional (compute needs kfd+render only) and a
            # containerized agent may not have it mapped — emit only if real
            if gpu.card_path is not None and self.execer.path_exists(node, gpu.card_path):
                device_nodes.append({"path": gpu.card_path})
            entry = {
                "name": gpu.device_id,
                "containerEdits": {
                    "deviceNodes": device_nodes,
                },
                "annotations": {
                    "cro.amd.com/pci-bdf": gpu.pci_bdf,
                    "cro.amd.com/vram-bytes": str(gpu.vram_bytes),
                    "cro.amd.com/gfx-target": gpu.gfx_target,
                    "cro.amd.com/xgmi-peers": ",".join(str(p) for p in gpu.xgmi_peers),
                    "cro.amd.com/kfd-node": str(gpu.kfd_node),
                },
            }
            spec["devices"] = [d for d in spec["devices"] if d["name"] != gpu.device_id]
            spec["devices"].append(entry)
            self._store(node, spec)
            return f"{self.kind}={gpu.device_id}"

    def add_raw_device(
        self, node: str, device_id: str, device_nodes, annotations
    ) -> str:
        """Add a device entry from explicit nodes/annotations (non-GPU
        device classes); returns its CDI device id."""
        with self._lock:
            spec = self._load(node)
            entry = {
                "name": device_id,
                "containerEdits": {
                    "deviceNodes": [{"path": p} for p in device_nodes],
                },
                "annotations": dict(annotations),
            }
            spec["devices"] = [d for d in spec["devices"] if d["name"] != device_id]
            spec["devices"].append(entry)
            self._store(node, spec)
            return f"{self.kind}={device_id}"

    def remove_device(self, node: str, device_id: str) -> None:
        with self._lock:
            spec = self._load(node)
            before = len(spec["devices"])
            spec["devices"] = [d for d in spec["devices"] if d["name"] != device_id]
            if len(spec["devices"]) != before:
                self._store(node, spec)

    def devices(self, node: str) -> List[str]:
        with self._lock:
            return [d["name"] for d in self._load(node)["devices"]]

    def get_device(self, node: str, device_id: str) -> Optional[dict]:
        with self._lock:
            for d in self._load(node)["devices"]:
                if d["name"] == device_id:
                    return d
            return None
