"""CXL.mem node path: enumeration, CDI exposure and drain for composable
memory expanders.

Goes beyond the reference, whose node path is GPU-only although its CRD
accepts ``type: cxlmemory`` (composabilityrequest_types.go:41) — this build
makes the enum real on the node:

* enumeration from ``/sys/bus/cxl/devices/mem*``: ``serial`` (the device
  fuse, the identity analog of KFD unique_id), ``ram/size``, ``numa_node``,
  and the PCI endpoint via the ``device/uevent`` PCI_SLOT_NAME — same
  identity convention as GPUs: ``CXL-<16-hex-serial>`` (PCI fallback);
* CDI spec under kind ``amd.com/cxlmem`` exposing the memdev's dax device
  node when present (``/dev/daxX.Y``) with capacity/NUMA annotations;
* drain via PCI hot-remove of the endpoint (the same sysfs mechanism the
  GPU drain uses — CXL memory composition rides the same fabric).

Load checking attributes mapped dax pages to processes by scanning
``/proc/<pid>/maps`` for the memdev's ``/dev/daxX.Y`` path (the CXL
analog of the per-GPU KFD vram attribution): a process that mmap'ed the
dax device holds composed memory and blocks a non-forced detach.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

from .execs import ExecError, NodeExec

CXL_DEVICES = "/sys/bus/cxl/devices"


@dataclass
class CXLMemDevice:
    name: str  # mem0, mem1, ...
    device_id: str  # canonical "CXL-<16-hex-serial>"
    serial: int
    size_bytes: int
    numa_node: int
    pci_bdf: str
    dax_path: str = ""  # /dev/daxX.Y when a dax region is bound


def canonical_cxl_id(serial: int, pci_bdf: str) -> str:
    if serial:
        return f"CXL-{serial:016x}"
    return f"CXL-pci-{pci_bdf}"


def _read_int(execer: NodeExec, node: str, path: str, base: int = 10) -> int:
    try:
        return int(execer.read_file(node, path).strip(), base)
    except (FileNotFoundError, PermissionError, OSError, ValueError):
        return 0


def enumerate_cxl_memdevs(execer: NodeExec, node: str) -> List[CXLMemDevice]:
    """Enumerate CXL memory devices; empty list when the bus is absent."""
    try:
        entries = execer.list_dir(node, CXL_DEVICES)
    except (FileNotFoundError, PermissionError, OSError):
        return []
    devices: List[CXLMemDevice] = []
    for entry in sorted(entries):
        if not (entry.startswith("mem") and entry[3:].isdigit()):
            continue
        base = f"{CXL_DEVICES}/{entry}"
        serial = _read_int(execer, node, f"{base}/serial", base=16)
        size = _read_int(execer, node, f"{base}/ram/size", base=16) or _read_int(
            execer, node, f"{base}/ram/size"
        )
        numa = _read_int(execer, node, f"{base}/numa_node")
        bdf = ""
        try:
            uevent = execer.read_file(node, f"{base}/device/uevent")
            for line in uevent.splitlines():
                if line.startswith("PCI_SLOT_NAME="):
                    bdf = line.split("=", 1)[1].strip().lower()
        except (FileNotFoundError, PermissionError, OSError):
            pass
        dax_path = ""
        try:
            for dax in execer.list_dir(node, f"{base}/dax"):
                if dax.startswith("dax"):
                    dax_path = f"/dev/{dax}"
                    break
        except (FileNotFoundError, PermissionError, OSError):
            pass
        devices.append(
            CXLMemDevice(
                name=entry,
                device_id=canonical_cxl_id(serial, bdf),
                serial=serial,
                size_bytes=size,
                numa_node=numa,
                pci_bdf=bdf,
                dax_path=dax_path,
            )
        )
    return devices


class CxlNodeOps:
    """NodeOps surface for ``type: cxlmemory`` resources (composed through
    the CompositeNodeOps dispatcher)."""

    CDI_KIND = "amd.com/cxlmem"

    def __init__(
        self,
        execer: NodeExec,
        cdi_dir: str = "/etc/cdi",
        destructive: bool = True,
        initially_detached: Optional[List[str]] = None,
    ):
        from .cdi_spec import CDISpecWriter

        self.execer = execer
        self.destructive = destructive
        self.cdi = CDISpecWriter(execer, cdi_dir, kind=self.CDI_KIND, root_device_nodes=[])
        self._sim_detached = set(initially_detached or [])

    def ensure_driver(self, node: str) -> None:
        # the cxl core is built into mainline kernels; the bus directory is
        # the presence signal (cxl_pci/cxl_mem autoload with the device)
        if not self.execer.path_exists(node, CXL_DEVICES):
            raise ExecError(f"CXL bus not present on node {node}")

    def enumerate(self, node: str) -> List[CXLMemDevice]:
        devs = enumerate_cxl_memdevs(self.execer, node)
        if self.destructive:
            return devs
        return [d for d in devs if d.device_id not in self._sim_detached]

    def find(self, node: str, device_id: str) -> Optional[CXLMemDevice]:
        for d in self.enumerate(node):
            if d.device_id == device_id:
                return d
        return None

    def is_visible(self, node: str, device_id: str) -> bool:
        return self.find(node, device_id) is not None

    is_visible_dra = is_visible  # no CXL DRA driver exists yet

    def dax_holders(self, node: str, device_id: str) -> List[int]:
        """PIDs holding a mapping of the memdev's dax device (via
        /proc/<pid>/maps — mapped file paths include /dev/daxX.Y)."""
        dev = self.find(node, device_id)
        if dev is None or not dev.dax_path:
            return []
        holders: List[int] = []
        try:
            entries = self.execer.list_dir(node, "/proc")
        except (FileNotFoundError, PermissionError, OSError):
            return []
        for entry in entries:
            if not entry.isdigit():
                continue
            try:
                maps = self.execer.read_file(node, f"/proc/{entry}/maps")
            except (FileNotFoundError, PermissionError, OSError):
                continue  # raced exit / hidden pid — not a holder we can see
            if dev.dax_path in maps:
                holders.append(int(entry))
        return holders

    def check_no_loads(self, node: str, device_id: Optional[str] = None) -> None:
        """Block detach while processes hold composed memory mapped.

        ``device_id=None`` (whole-node, DEVICE_PLUGIN shape) checks every
        enumerated memdev; per-device otherwise. A memdev with no bound
        dax region cannot be mmap'ed and passes vacuously.
        """
        from .amdgpu import GPULoadsPresent

        targets = (
            [d.device_id for d in self.enumerate(node)]
            if device_id is None
            else [device_id]
        )
        for did in targets:
            holders = self.dax_holders(node, did)
            if holders:
                raise GPULoadsPresent(
                    f"processes hold dax mappings of {did}: {holders}"
                )

    def drain(self, node: str, device_id: str) -> None:
        dev = self.find(node, device_id)
        if dev is None:
            return
        if not self.destructive:
            self._sim_detached.add(device_id)
            return
        if not dev.pci_bdf:
            raise ExecError(f"no PCI endpoint known for {device_id}; cannot drain")
        self.execer.write_file(node, f"/sys/bus/pci/devices/{dev.pci_bdf}/remove", "1")

    def simulate_compose(self, node: str, device_id: str) -> None:
        self._sim_detached.discard(device_id)

    def refresh_after_attach(self, node: str) -> None:
        if self.destructive:
            self.execer.write_file(node, "/sys/bus/pci/rescan", "1")

    def refresh_after_detach(self, node: str) -> None:
        return None

    def write_cdi(self, node: str, device_id: str) -> str:
        dev = self.find(node, device_id)
        if dev is None:
            raise ExecError(f"cxl device {device_id} not enumerable; cannot write CDI")
        nodes = [dev.dax_path] if dev.dax_path else []
        return self.cdi.add_raw_device(
            node,
            device_id,
            device_nodes=nodes,
            annotations={
                "cro.amd.com/cxl-memdev": dev.name,
                "cro.amd.com/size-bytes": str(dev.size_bytes),
                "cro.amd.com/numa-node": str(dev.numa_node),
                "cro.amd.com/pci-bdf": dev.pci_bdf,
            },
        )

    def remove_cdi(self, node: str, device_id: str) -> None:
        self.cdi.remove_device(node, device_id)

    def health_probe(self, node: str, device_id: str) -> Optional[dict]:
        return None  # a dax write/read/verify probe is future work
