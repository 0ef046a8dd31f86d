"""ctypes wrapper for the gfx950 health probe (cro_amd/hip/probe.hip).

Fails LOUDLY when running on a GPU node without the compiled extension —
a silent fallback would let GPU tests pass without the native path
(the operator must never advertise a device it could not verify).
"""

from __future__ import annotations

import ctypes
import os
from typing import Optional

_LIB_PATH = os.path.join(os.path.dirname(__file__), "..", "hip", "libcroprobe.so")


class CroProbeResult(ctypes.Structure):
    _fields_ = [
        ("ok", ctypes.c_int),
        ("mfma_f32_exact", ctypes.c_int),
        ("hbm_gbps", ctypes.c_double),
        ("bf16_tflops", ctypes.c_double),
        ("vram_total", ctypes.c_longlong),
        ("vram_free", ctypes.c_longlong),
        ("t_setup_ms", ctypes.c_double),
        ("t_mfma_ms", ctypes.c_double),
        ("t_bw_ms", ctypes.c_double),
        ("t_bf16_ms", ctypes.c_double),
        ("gcn_arch", ctypes.c_char * 64),
        ("msg", ctypes.c_char * 256),
    ]


_lib = None


def _gpu_present() -> bool:
    return os.path.exists("/dev/kfd")


def load_library(required: Optional[bool] = None) -> Optional[ctypes.CDLL]:
    """Load libcroprobe.so.  required=None → required iff a GPU is present."""
    global _lib
    if _lib is not None:
        return _lib
    if required is None:
        required = _gpu_present()
    path = os.path.abspath(_LIB_PATH)
    if not os.path.exists(path):
        if required:
            raise RuntimeError(
                f"libcroprobe.so missing at {path} on a GPU node — build it with "
                "`python -m cro_amd.hip.build` (hipcc --offload-arch=gfx950)"
            )
        return None
    lib = ctypes.CDLL(path)
    lib.cro_probe_run.argtypes = [ctypes.c_int, ctypes.POINTER(CroProbeResult)]
    lib.cro_probe_run.restype = ctypes.c_int
    lib.cro_probe_device_count.restype = ctypes.c_int
    lib.cro_probe_pci_bus_id.argtypes = [ctypes.c_int, ctypes.c_char_p, ctypes.c_int]
    lib.cro_probe_pci_bus_id.restype = ctypes.c_int
    lib.cro_probe_alloc.argtypes = [ctypes.c_int, ctypes.c_longlong]
    lib.cro_probe_alloc.restype = ctypes.c_void_p
    lib.cro_probe_free.argtypes = [ctypes.c_void_p]
    lib.cro_probe_free.restype = None
    _lib = lib
    return lib


def vram_alloc(device: int, nbytes: int):
    """Allocate+touch VRAM on a device; returns an opaque handle (None on
    failure).  Used by the self-pid fingerprint."""
    lib = load_library(required=True)
    return lib.cro_probe_alloc(device, nbytes)


def vram_free(handle) -> None:
    if handle:
        load_library(required=True).cro_probe_free(handle)


def device_count() -> int:
    lib = load_library()
    if lib is None:
        return 0
    return max(lib.cro_probe_device_count(), 0)


_bdf_cache = {}


def hip_device_for_bdf(pci_bdf: str) -> Optional[int]:
    """Map a PCI DBDF (from KFD topology) to a HIP device ordinal (cached —
    hipDeviceGetPCIBusId is sysfs-backed and the probe runs per attach)."""
    want = pci_bdf.lower()
    if want in _bdf_cache:
        return _bdf_cache[want]
    lib = load_library()
    if lib is None:
        return None
    buf = ctypes.create_string_buffer(64)
    for dev in range(device_count()):
        if lib.cro_probe_pci_bus_id(dev, buf, 64) == 0:
            if buf.value.decode().lower().startswith(want.rsplit(".", 1)[0]):
                _bdf_cache[want] = dev
                return dev
    return None


def run_probe(device: int = 0) -> dict:
    lib = load_library(required=True)
    res = CroProbeResult()
    rc = lib.cro_probe_run(device, ctypes.byref(res))
    return {
        "ok": bool(res.ok) and rc == 0,
        "rc": rc,
        "mfma_f32_exact": bool(res.mfma_f32_exact),
        "hbm_gbps": res.hbm_gbps,
        "bf16_tflops": res.bf16_tflops,
        "vram_total": res.vram_total,
        "vram_free": res.vram_free,
        "t_setup_ms": res.t_setup_ms,
        "t_mfma_ms": res.t_mfma_ms,
        "t_bw_ms": res.t_bw_ms,
        "t_bf16_ms": res.t_bf16_ms,
        "gcn_arch": res.gcn_arch.decode(errors="replace"),
        "msg": res.msg.decode(errors="replace"),
    }


def probe_fn_for_nodeops(gpu) -> dict:
    """AmdNodeOps probe hook: GPUDevice → probe result on the right ordinal."""
    dev = hip_device_for_bdf(gpu.pci_bdf)
    if dev is None:
        dev = 0
    return run_probe(dev)


def probe_via_exec(execer, node: str, gpu, argv_prefix=None) -> dict:
    """Health probe through the node agent (``croagent probe --bdf``) — the
    cluster shape where controllers run off-node and reach hardware only
    through the NodeExec seam.  Returns the same dict shape as run_probe.

    ``argv_prefix`` wraps the command for the container-driver arm (e.g.
    ``["chroot", "/run/amdgpu-driver"]`` so the probe runs against the
    driver container's ROCm userspace — the nvidia-smi-through-chroot
    analog, gpus.go:566-749).
    """
    import json as _json

    argv = list(argv_prefix or []) + ["croagent", "probe", "--bdf", gpu.pci_bdf]
    rc, out, err = execer.run(node, argv, timeout=300)
    if rc != 0 and not out.strip():
        return {"ok": False, "rc": rc, "msg": err.strip() or "croagent probe failed"}
    try:
        return _json.loads(out)
    except ValueError:
        return {"ok": False, "rc": rc, "msg": f"unparseable probe output: {out[:200]}"}


def make_exec_probe_fn(execer, node: str, argv_prefix_fn=None):
    """AmdNodeOps probe hook bound to a NodeExec (local or remote agent).

    ``argv_prefix_fn()`` is resolved per probe so a driver-mode change
    (DeviceConfig applied mid-flight) switches the chroot arm without a
    rebuild — pass ``lambda: ops.probe_argv_prefix(node)``.
    """

    def probe(gpu):
        prefix = argv_prefix_fn() if argv_prefix_fn else None
        return probe_via_exec(execer, node, gpu, argv_prefix=prefix)

    return probe
