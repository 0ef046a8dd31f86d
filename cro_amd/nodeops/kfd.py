"""KFD topology enumeration — the ROCm-native source of GPU truth.

The reference enumerates devices by forking ``nvidia-smi`` over pod exec and
scanning /proc/driver/nvidia/gpus (gpus.go:207-239,1362-1448).  The amdgpu
stack has a faster, fork-free equivalent: the KFD topology tree
``/sys/class/kfd/kfd/topology/nodes/*`` that the ROCm runtime itself builds
from.  Reading it costs microseconds, so visibility checks can run at
event speed instead of CLI speed — one of the levers for beating the
reference's attach latency.

Per node directory:
  properties        ``key value`` integer lines; GPUs have simd_count > 0,
                    unique_id (device serial fuse), drm_render_minor,
                    vendor_id/device_id, location_id/domain (PCI BDF)
  gpu_id            KFD-assigned numeric id (used by /sys/class/kfd/kfd/proc)
  io_links/*/properties  links with type (11 = xGMI) and node_to — the
                    topology surfaced into CDI specs/ResourceSlices so RCCL
                    jobs can see the 7-link point-to-point xGMI fabric
                    (the reference never exposes topology).

``amd-smi``/``rocm-smi`` remain the fallback when KFD sysfs is unreadable.
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .execs import ExecError, NodeExec

KFD_NODES = "/sys/class/kfd/kfd/topology/nodes"
KFD_PROC = "/sys/class/kfd/kfd/proc"
IOLINK_TYPE_XGMI = 11


@dataclass
class GPUDevice:
    kfd_node: int
    gpu_id: int
    device_id: str  # canonical "GPU-<16-hex unique_id>" (or PCI fallback)
    unique_id: int
    render_minor: int
    pci_bdf: str  # "0000:03:00.0"
    vendor_id: int = 0
    pci_device_id: int = 0
    vram_bytes: int = 0
    gfx_target: str = ""
    xgmi_peers: List[int] = field(default_factory=list)  # peer kfd node ids
    card_index: Optional[int] = None  # resolved from /sys/class/drm by PCI

    @property
    def render_path(self) -> str:
        return f"/dev/dri/renderD{self.render_minor}"

    @property
    def card_path(self) -> Optional[str]:
        # card numbering does NOT track render minors (a node exposing one
        # GPU of many can have renderD144 + card16): only a PCI match on
        # /sys/class/drm/card*/device is authoritative
        if self.card_index is None:
            return None
        return f"/dev/dri/card{self.card_index}"


def canonical_device_id(unique_id: int, pci_bdf: str) -> str:
    if unique_id:
        return f"GPU-{unique_id:016x}"
    return f"GPU-pci-{pci_bdf}"


def _parse_properties(text: str) -> Dict[str, int]:
    props: Dict[str, int] = {}
    for line in text.splitlines():
        parts = line.split()
        if len(parts) == 2:
            try:
                props[parts[0]] = int(parts[1])
            except ValueError:
                pass
    return props


def _pci_bdf(props: Dict[str, int]) -> str:
    # location_id encodes bus/dev/fn: bus = bits 8-15, dev = bits 3-7, fn = 0-2
    loc = props.get("location_id", 0)
    domain = props.get("domain", 0)
    bus = (loc >> 8) & 0xFF
    dev = (loc >> 3) & 0x1F
    fn = loc & 0x7
    return f"{domain:04x}:{bus:02x}:{dev:02x}.{fn:x}"


def _drm_card_by_bdf(execer: NodeExec, node: str) -> Dict[str, int]:
    """Map PCI DBDF → /dev/dri/card index via /sys/class/drm/card*/device."""
    mapping: Dict[str, int] = {}
    try:
        entries = execer.list_dir(node, "/sys/class/drm")
    except (FileNotFoundError, PermissionError, OSError):
        return mapping
    for entry in entries:
        if not (entry.startswith("card") and entry[4:].isdigit()):
            continue
        try:
            uevent = execer.read_file(node, f"/sys/class/drm/{entry}/device/uevent")
        except (FileNotFoundError, PermissionError, OSError):
            continue
        for line in uevent.splitlines():
            if line.startswith("PCI_SLOT_NAME="):
                mapping[line.split("=", 1)[1].strip().lower()] = int(entry[4:])
    return mapping


def enumerate_gpus(
    execer: NodeExec, node: str, static_cache: Optional[dict] = None
) -> List[GPUDevice]:
    """Enumerate GPUs from KFD topology; raises ExecError if KFD is absent.

    ``static_cache`` (owned by the caller, e.g. AmdNodeOps) memoizes the
    per-device data that cannot change while a device keeps its identity —
    VRAM bank sizes, xGMI peers and the DRM card index — so steady-state
    re-enumeration reads only each node's ``properties``/``gpu_id``.  xGMI
    topology is chassis-fixed; a device re-composed under a new unique_id
    misses the cache and re-reads everything.
    """
    try:
        entries = execer.list_dir(node, KFD_NODES)
    except (FileNotFoundError, PermissionError, OSError):
        raise ExecError("KFD topology not present (amdgpu driver not loaded?)")
    if static_cache is None:
        static_cache = {}
    card_map = static_cache.get("_card_map")
    if card_map is None:
        card_map = _drm_card_by_bdf(execer, node)
        static_cache["_card_map"] = card_map

    gpus: List[GPUDevice] = []
    for entry in entries:
        base = f"{KFD_NODES}/{entry}"
        try:
            props = _parse_properties(execer.read_file(node, f"{base}/properties"))
        except (FileNotFoundError, PermissionError, OSError):
            # unassigned GPUs are cgroup-hidden: their node dirs exist but
            # reads fail with EPERM — they are not ours to see
            continue
        if props.get("simd_count", 0) <= 0:
            continue  # CPU node
        try:
            gpu_id = int(execer.read_file(node, f"{base}/gpu_id").strip())
        except (FileNotFoundError, PermissionError, OSError, ValueError):
            gpu_id = 0
        unique_id = props.get("unique_id", 0)
        bdf = _pci_bdf(props)
        dev = GPUDevice(
            kfd_node=int(entry),
            gpu_id=gpu_id,
            device_id=canonical_device_id(unique_id, bdf),
            unique_id=unique_id,
            render_minor=props.get("drm_render_minor", 0),
            pci_bdf=bdf,
            vendor_id=props.get("vendor_id", 0),
            pci_device_id=props.get("device_id", 0),
            gfx_target=str(props.get("gfx_target_version", "")),
            card_index=card_map.get(bdf.lower()),
        )
        cache_key = (entry, unique_id, bdf)
        cached = static_cache.get(cache_key)
        if cached is not None:
            dev.vram_bytes, dev.xgmi_peers = cached[0], list(cached[1])
            gpus.append(dev)
            continue
        # VRAM from mem_banks (heap_type 1/2 = FB public/private)
        try:
            for bank in execer.list_dir(node, f"{base}/mem_banks"):
                bprops = _parse_properties(
                    execer.read_file(node, f"{base}/mem_banks/{bank}/properties")
                )
                if bprops.get("heap_type", 0) in (1, 2):
                    dev.vram_bytes += bprops.get("size_in_bytes", 0)
        except (FileNotFoundError, PermissionError, OSError):
            pass
        # xGMI peer links
        try:
            for link in execer.list_dir(node, f"{base}/io_links"):
                lprops = _parse_properties(
                    execer.read_file(node, f"{base}/io_links/{link}/properties")
                )
                if lprops.get("type", 0) == IOLINK_TYPE_XGMI:
                    dev.xgmi_peers.append(lprops.get("node_to", -1))
        except (FileNotFoundError, PermissionError, OSError):
            pass
        static_cache[cache_key] = (dev.vram_bytes, list(dev.xgmi_peers))
        gpus.append(dev)
    return gpus


def enumerate_gpus_amdsmi(execer: NodeExec, node: str) -> List[GPUDevice]:
    """Fallback enumeration via ``amd-smi list --json`` (CLI fork — slow path).

    Output rows look like ``{"gpu": 0, "bdf": "0000:f1:00.0", "uuid":
    "d0ff75a3-…", "kfd_id": 8465, "node_id": 6, "partition_id": 0}``
    (captured on MI355X/ROCm 7.2).  Caveats vs the KFD path: the amd-smi
    uuid is NOT the KFD unique_id fuse, so device ids from this path only
    match each other (use one enumeration source consistently), and no
    render minor is reported — CDI emission needs the KFD path.
    ``kfd_id`` is the KFD gpu_id used for /sys/class/kfd/kfd/proc load
    attribution."""
    rc, out, err = execer.run(node, ["amd-smi", "list", "--json"])
    if rc != 0:
        raise ExecError(f"amd-smi list failed: {err}", rc=rc, stderr=err)
    data = json.loads(out)
    gpus: List[GPUDevice] = []
    rows = data if isinstance(data, list) else data.get("gpus", [])
    for row in rows:
        bdf = row.get("bdf", "")
        uuid = row.get("uuid", "") or ""
        uid = 0
        hexpart = uuid.replace("GPU-", "").replace("-", "")
        try:
            uid = int(hexpart[:16], 16) if hexpart else 0
        except ValueError:
            uid = 0
        gpus.append(
            GPUDevice(
                kfd_node=row.get("node_id", -1),
                gpu_id=row.get("kfd_id", -1),
                device_id=canonical_device_id(uid, bdf),
                unique_id=uid,
                render_minor=0,
                pci_bdf=bdf,
            )
        )
    return gpus


def self_host_pid() -> int:
    """Best-effort host-namespace pid of this process (see
    :func:`resolve_self_kfd_pid` for the authoritative method).

    /sys/class/kfd/kfd/proc is keyed by HOST pids; a containerized node
    agent sees namespaced pids from os.getpid().  CRO_SELF_KFD_PID
    overrides; /proc/self/sched leaks the host pid on some kernels (modern
    ones show the namespaced pid — then this falls back to os.getpid()).
    """
    import os
    import re

    env = os.environ.get("CRO_SELF_KFD_PID", "")
    if env.isdigit():
        return int(env)
    try:
        with open("/proc/self/sched") as f:
            m = re.search(r"\((\d+),", f.readline())
        if m:
            return int(m.group(1))
    except OSError:
        pass
    return os.getpid()


def _read_vram_map(execer: NodeExec, node: str, gpu_id: int) -> Dict[int, int]:
    out: Dict[int, int] = {}
    try:
        pids = execer.list_dir(node, KFD_PROC)
    except (FileNotFoundError, PermissionError, OSError):
        return out
    for pid in pids:
        if not pid.isdigit():
            continue
        try:
            out[int(pid)] = int(
                execer.read_file(node, f"{KFD_PROC}/{pid}/vram_{gpu_id}").strip()
            )
        except (FileNotFoundError, PermissionError, OSError, ValueError):
            pass
    return out


def resolve_self_kfd_pid(
    execer: NodeExec, node: str, gpu_id: int, hip_device: int
) -> Optional[int]:
    """Authoritative host-pid self-identification via a VRAM fingerprint.

    Allocates a marker-sized VRAM buffer on the device and returns the pid
    of the single /sys/class/kfd/kfd/proc entry whose ``vram_<gpu_id>``
    grew by at least the marker — that entry is this process (cgroup device
    isolation guarantees no other container can touch our GPU, and no other
    process of ours allocates concurrently).  Retries with distinct marker
    sizes to reject coincidental growth.  Returns None when no probe
    library / GPU is available.
    """
    import os

    try:
        from .probe import vram_alloc, vram_free
    except Exception:
        return None

    for attempt in range(3):
        marker = (48 << 20) + ((os.getpid() + attempt * 7919) % 4096) * 8192
        before = _read_vram_map(execer, node, gpu_id)
        handle = None
        try:
            handle = vram_alloc(hip_device, marker)
            if not handle:
                return None
            after = _read_vram_map(execer, node, gpu_id)
        finally:
            if handle:
                vram_free(handle)
        grown = [
            pid
            for pid, v in after.items()
            if v - before.get(pid, 0) >= int(marker * 0.9)
        ]
        if len(grown) == 1:
            return grown[0]
    return None


def gpu_compute_pids(execer: NodeExec, node: str, gpu_id: Optional[int] = None) -> List[int]:
    """PIDs with open KFD compute contexts (the amdgpu-native analog of
    ``nvidia-smi --query-compute-apps``, gpus.go:241-350).

    /sys/class/kfd/kfd/proc/<pid>/ exists per process with a KFD context;
    per-GPU attribution uses the vram_<gpu_id> usage files when present.
    """
    try:
        pids = execer.list_dir(node, KFD_PROC)
    except (FileNotFoundError, PermissionError, OSError):
        return []
    result: List[int] = []
    for pid in pids:
        if not pid.isdigit():
            continue
        if gpu_id is None:
            result.append(int(pid))
            continue
        if isinstance(gpu_id, (list, tuple, set)):
            if any(_pid_vram(execer, node, pid, g) > 0 for g in gpu_id):
                result.append(int(pid))
            continue
        # per-device attribution: a pid loads THIS gpu iff its vram_<gpu_id>
        # file reads > 0.  A missing/unreadable file means the process has no
        # context on this device (other tenants' processes on a shared node
        # are visible in the proc dir but their per-GPU files are not ours to
        # read) — do NOT count those, or detach wedges forever on shared
        # machines.  The node-wide check (gpu_id=None) stays conservative.
        if _pid_vram(execer, node, pid, gpu_id) > 0:
            result.append(int(pid))
    return result


def _pid_vram(execer: NodeExec, node: str, pid: str, gpu_id: int) -> int:
    try:
        return int(execer.read_file(node, f"{KFD_PROC}/{pid}/vram_{gpu_id}").strip())
    except (FileNotFoundError, PermissionError, OSError, ValueError):
        return 0
