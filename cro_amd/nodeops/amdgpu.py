"""amdgpu node operations — the MI355X-native device lifecycle.

Re-design of the reference's nvidia node path (internal/utils/gpus.go,
SURVEY.md §2.8) for the amdgpu/ROCm stack:

| reference (nvidia)                         | here (amdgpu)                              |
|--------------------------------------------|--------------------------------------------|
| modinfo nvidia / ClusterPolicy driver check | /sys/module/amdgpu presence (gpus.go:97-127)|
| nvidia-smi --query-gpu=gpu_uuid visibility  | KFD topology unique_id scan (gpus.go:207-239)|
| nvidia-smi --query-compute-apps load check  | /sys/class/kfd/kfd/proc/<pid> scan (gpus.go:241-350)|
| nvidia-smi drain + sysfs remove + modprobe  | sysfs PCI remove; modprobe -r amdgpu only for last device (gpus.go:352-865)|
| device-plugin/DCGM daemonset restart        | device-plugin/metrics-exporter restart hooks + ResourceSlice publish (gpus.go:1109-1146)|
| (no CDI emission)                           | CDI spec write incl. xGMI topology          |
| (no functional verification)                | HIP health probe on gfx950 (MFMA + HBM), optional |

Two implementations of the same NodeOps surface:

* :class:`AmdNodeOps` — real sysfs/KFD via a NodeExec seam.  Because hot
  PCI removal on a machine we don't own is destructive, ``destructive=False``
  redirects the *writes* of the lifecycle (pci remove / rescan / modprobe)
  into a simulated-detach set while every *read* path (enumeration, loads,
  CDI) stays real — the mode the single-node GPU bench runs in, where no
  physical CXL fabric exists to actually compose devices.
* :class:`MockNodeOps` — pure in-memory double for control-plane tests
  (the injected-interface analog of the reference's gomonkey MockExecutor,
  suite_test.go:296-307).
"""

from __future__ import annotations

import logging
import threading
import time
from typing import Callable, Dict, List, Optional, Set

from ..api.v1alpha1.types import ResourceSlice, ResourceSliceDevice, ResourceSliceSpec
from .cdi_spec import CDISpecWriter
from .execs import ExecError, NodeExec
from .kfd import GPUDevice, enumerate_gpus, enumerate_gpus_amdsmi, gpu_compute_pids

log = logging.getLogger(__name__)


class DriverMissing(Exception):
    pass


class GPULoadsPresent(Exception):
    pass


class DrainInProgress(Exception):
    """Last-device drain is running asynchronously; re-check shortly."""


class NodeOps:
    """Surface the ComposableResource controller programs against."""

    def for_type(self, resource_type: str) -> "NodeOps":
        """Per-resource-type dispatch hook (CompositeNodeOps overrides);
        single-type implementations serve every type themselves."""
        return self

    def ensure_driver(self, node: str) -> None:
        raise NotImplementedError  # pragma: no cover

    def enumerate(self, node: str) -> List[GPUDevice]:
        raise NotImplementedError  # pragma: no cover

    def is_visible(self, node: str, device_id: str) -> bool:
        raise NotImplementedError  # pragma: no cover

    def check_no_loads(self, node: str, device_id: Optional[str] = None) -> None:
        raise NotImplementedError  # pragma: no cover

    def drain(self, node: str, device_id: str) -> None:
        raise NotImplementedError  # pragma: no cover

    def refresh_after_attach(self, node: str) -> None:
        raise NotImplementedError  # pragma: no cover

    def refresh_after_detach(self, node: str) -> None:
        raise NotImplementedError  # pragma: no cover

    def write_cdi(self, node: str, device_id: str) -> str:
        raise NotImplementedError  # pragma: no cover

    def remove_cdi(self, node: str, device_id: str) -> None:
        raise NotImplementedError  # pragma: no cover

    def health_probe(self, node: str, device_id: str) -> Optional[dict]:
        return None  # optional capability


class AmdNodeOps(NodeOps):
    def __init__(
        self,
        execer: NodeExec,
        client=None,
        cdi_dir: str = "/etc/cdi",
        destructive: bool = True,
        initially_detached: Optional[List[str]] = None,
        restart_hooks: Optional[Dict[str, Callable[[str], None]]] = None,
        probe_fn: Optional[Callable[[int], dict]] = None,
        publish_resource_slices: bool = True,
    ):
        self.execer = execer
        self.client = client  # runtime Client for ResourceSlice publication
        self.cdi = CDISpecWriter(execer, cdi_dir)
        self.destructive = destructive
        self.publish_slices = publish_resource_slices
        # device_ids whose PCI removal is simulated (non-destructive mode)
        self._sim_detached: Set[str] = set(initially_detached or [])
        self._sim_lock = threading.Lock()
        # daemonset-restart analogs, keyed by component name
        self.restart_hooks = restart_hooks or {}
        self.probe_fn = probe_fn  # probe_fn(gpu: GPUDevice) -> result dict
        # gpu_id -> our own host pid holding a KFD context there (or None)
        self._self_pid_cache: Dict[int, Optional[int]] = {}
        # async last-device drains: device_id -> worker thread / error text
        self._drain_lock = threading.Lock()
        self._drain_threads: Dict[str, threading.Thread] = {}
        self._drain_errors: Dict[str, str] = {}
        # short-TTL enumeration cache: one attach touches the topology 4+
        # times (refresh/visibility/CDI/probe) and each full KFD sysfs scan
        # costs ~2 ms; lifecycle mutations invalidate explicitly
        self.enum_cache_ttl = 0.025
        # node -> (monotonic, gpus, generation_id) — TTL bounds staleness,
        # generation_id revalidation avoids full topology walks after it
        self._enum_cache: Dict[str, tuple] = {}
        self._enum_lock = threading.Lock()
        # per-device static metadata (VRAM/xGMI/card map) survives the TTL
        # cache — keyed by device identity, see kfd.enumerate_gpus
        self._static_meta: dict = {}

    # -- driver ------------------------------------------------------------

    def driver_mode(self, node: str) -> str:
        """Driver-detection chain (gpus.go:97-127 analog): an AMD GPU
        operator DeviceConfig with the containerized driver enabled →
        ``container`` (the ClusterPolicy arm); else /sys/module/amdgpu on
        the node → ``host`` (the modinfo arm); else ``none``."""
        if self.client is not None:
            from ..api.v1alpha1.types import DeviceConfig

            try:
                configs = self.client.list(DeviceConfig)
            except Exception:
                configs = []
            if any(dc.spec.driver.enable for dc in configs):
                return "container"
        if self.execer.path_exists(node, "/sys/module/amdgpu"):
            return "host"
        return "none"

    def driver_root(self, node: str) -> str:
        """Driver-container rootfs for module operations, or "" in host
        mode.  Container mode resolves DeviceConfig.spec.driver.driver_root
        (CRO_DRIVER_ROOT env overrides) — the chroot target of the
        reference's drain path B (gpus.go:566-749: chroot
        /run/nvidia/driver before module/device ops)."""
        if self.driver_mode(node) != "container":
            return ""
        import os

        env_root = os.environ.get("CRO_DRIVER_ROOT", "")
        if env_root:
            return env_root
        from ..api.v1alpha1.types import DeviceConfig

        for dc in self.client.list(DeviceConfig):
            if dc.spec.driver.enable:
                return dc.spec.driver.driver_root
        return "/run/amdgpu-driver"

    def _module_argv(self, node: str, argv: List[str]) -> List[str]:
        """Wrap a module-level command (modprobe, croagent probe) with the
        driver-container chroot when the driver is containerized — the
        kernel modules and ROCm userspace live in the driver container's
        rootfs there, not on the host."""
        root = self.driver_root(node)
        if root:
            return ["chroot", root] + argv
        return argv

    def probe_argv_prefix(self, node: str) -> List[str]:
        """Chroot prefix for the exec-probe hook (probe.make_exec_probe_fn)
        — [] in host mode, ["chroot", <driver_root>] in container mode."""
        root = self.driver_root(node)
        return ["chroot", root] if root else []

    def ensure_driver(self, node: str) -> None:
        """Driver gate before any attach work.

        Host mode: the amdgpu module directory must exist. Container mode:
        the driver daemonset must be fully ready (the driver-pod readiness
        gate, gpus.go:161-193) AND the module loaded — a containerized
        driver still surfaces /sys/module/amdgpu on the host.
        """
        mode = self.driver_mode(node)
        if mode == "none":
            raise DriverMissing(f"amdgpu kernel module not loaded on node {node}")
        if mode == "container":
            import os

            from ..api.v1alpha1.types import DaemonSet, DeviceConfig
            from ..runtime.errors import NotFoundError

            ns = os.environ.get("CRO_AMD_GPU_OPERATOR_NAMESPACE", "amd-gpu-operator")
            ds_name = "amd-gpu-driver"
            for dc in self.client.list(DeviceConfig):
                if dc.spec.driver.enable and dc.spec.driver.daemonset_name:
                    ds_name = dc.spec.driver.daemonset_name
                    break
            try:
                ds = self.client.get(DaemonSet, f"{ns}/{ds_name}")
            except NotFoundError:
                raise DriverMissing(
                    f"containerized driver enabled but daemonset "
                    f"{ns}/{ds_name} not found"
                )
            st = ds.status
            if st.number_ready < st.desired_number_scheduled:
                raise DriverMissing(
                    f"driver daemonset {ns}/{ds_name} not ready "
                    f"({st.number_ready}/{st.desired_number_scheduled})"
                )
            if not self.execer.path_exists(node, "/sys/module/amdgpu"):
                raise DriverMissing(
                    f"driver daemonset ready but amdgpu not loaded on {node}"
                )

    # -- enumeration / visibility -----------------------------------------

    def _topology_generation(self, node: str) -> Optional[str]:
        """KFD bumps /sys/class/kfd/kfd/topology/generation_id on every
        topology change (hot-plug either way) — one tiny sysfs read tells
        whether a cached enumeration is still exact. None when unreadable
        (mock fixtures, hidden sysfs) → callers fall back to a full walk."""
        try:
            gen = self.execer.read_file(
                node, "/sys/class/kfd/kfd/topology/generation_id"
            ).strip()
            return gen or None
        except (FileNotFoundError, PermissionError, OSError, ExecError):
            return None

    def enumerate(self, node: str) -> List[GPUDevice]:
        now = time.monotonic()
        with self._enum_lock:
            cached = self._enum_cache.get(node)
            if cached is not None and now - cached[0] < self.enum_cache_ttl:
                gpus = cached[1]
            else:
                gpus = None
        if gpus is None and cached is not None and cached[2] is not None:
            # TTL expired, but an unchanged generation_id proves the
            # topology did not move — revalidate instead of re-walking
            if self._topology_generation(node) == cached[2]:
                gpus = cached[1]
                with self._enum_lock:
                    self._enum_cache[node] = (now, gpus, cached[2])
        if gpus is None:
            try:
                gpus = enumerate_gpus(self.execer, node, self._static_meta)
            except ExecError:
                gpus = enumerate_gpus_amdsmi(self.execer, node)
            gen = self._topology_generation(node)
            with self._enum_lock:
                self._enum_cache[node] = (now, gpus, gen)
        if self.destructive:
            return list(gpus)
        with self._sim_lock:
            return [g for g in gpus if g.device_id not in self._sim_detached]

    def _invalidate_enum(self, node: str) -> None:
        with self._enum_lock:
            self._enum_cache.pop(node, None)

    def find_gpu(self, node: str, device_id: str) -> Optional[GPUDevice]:
        for g in self.enumerate(node):
            if g.device_id == device_id:
                return g
        return None

    def is_visible(self, node: str, device_id: str) -> bool:
        return self.find_gpu(node, device_id) is not None

    def is_visible_dra(self, node: str, device_id: str) -> bool:
        """DRA visibility: the published ResourceSlice lists the uuid
        (gpus.go:207-239 ResourceSlice branch)."""
        if self.client is None:
            return self.is_visible(node, device_id)
        for sl in self.client.list(ResourceSlice):
            if sl.spec.node_name == node:
                if any(d.uuid == device_id for d in sl.spec.devices):
                    return True
        return False

    # -- loads -------------------------------------------------------------

    def check_no_loads(self, node: str, device_id: Optional[str] = None) -> None:
        """Per-device (DRA) or node-wide (DEVICE_PLUGIN) load check.

        "Node-wide" means GPUs *this node can enumerate*: under cgroup
        device isolation /sys/class/kfd/kfd/proc shows every host process
        with any KFD context, but processes whose VRAM sits on devices
        invisible to this node cannot possibly hold ours — counting them
        (as a naive whole-dir scan would) wedges detach on shared machines.
        """
        if device_id is not None:
            gpu = self.find_gpu(node, device_id)
            if gpu is None:
                return  # device already gone — nothing can be loading it
            gpu_ids = [gpu.gpu_id]
        else:
            gpu_ids = [g.gpu_id for g in self.enumerate(node)]
            if not gpu_ids:
                return
        pids = gpu_compute_pids(self.execer, node, gpu_ids)
        own = set()
        for gid in gpu_ids:
            own |= self._own_pids(node, gid)
        pids = [p for p in pids if p not in own]
        if pids:
            scope = f"device {device_id}" if device_id else f"node {node}"
            raise GPULoadsPresent(f"{scope} has active KFD compute processes: {pids}")

    def _own_pids(self, node: str, gpu_id: Optional[int]) -> Set[int]:
        """Node-agent self-exemption for the load check: the agent's own
        process holds a KFD context (health probe / torch) and must not
        block the detach it is itself orchestrating.

        /sys/class/kfd/kfd/proc is keyed by HOST pids; a containerized
        agent's os.getpid() is namespaced, so the authoritative mapping is
        the VRAM-fingerprint resolver (kfd.resolve_self_kfd_pid), cached per
        device.  Applies only when operating on the local node.
        """
        from .execs import LocalNodeExec as _Local
        from .kfd import resolve_self_kfd_pid, self_host_pid

        if not isinstance(self.execer, _Local):
            return set()
        import os as _os

        own: Set[int] = {_os.getpid(), self_host_pid()}
        if gpu_id is not None:
            cached = self._self_pid_cache.get(gpu_id)
            if cached is None and gpu_id not in self._self_pid_cache:
                hip_dev = None
                try:
                    from .probe import hip_device_for_bdf, load_library

                    if load_library(required=False) is not None:
                        for g in self.enumerate(node):
                            if g.gpu_id == gpu_id:
                                hip_dev = hip_device_for_bdf(g.pci_bdf)
                                break
                        if hip_dev is not None:
                            cached = resolve_self_kfd_pid(
                                self.execer, node, gpu_id, hip_dev
                            )
                except Exception as exc:
                    log.debug("self-pid resolution failed: %s", exc)
                self._self_pid_cache[gpu_id] = cached
            if cached is not None:
                own.add(cached)
        return own

    # -- drain / attach refresh -------------------------------------------

    def drain(self, node: str, device_id: str) -> None:
        """Hot-remove one composed GPU from the node.

        Sequence (gpus.go:352-565 re-designed for amdgpu):
          1. locate the device (already-gone → idempotent no-op);
          2. if it is the LAST visible GPU, unload amdgpu first — KFD keeps
             /dev/kfd open per-process and the module cannot release a lone
             device while bound;
          3. write 1 to /sys/bus/pci/devices/<bdf>/remove.
        No per-GPU persistence daemon exists on AMD (nvidia-persistenced has
        no analog), and there is no drain-status query — drain progress is
        tracked in CR status instead.
        """
        # a previously started async drain for this device?
        with self._drain_lock:
            thread = self._drain_threads.get(device_id)
            if thread is not None:
                if thread.is_alive():
                    raise DrainInProgress(f"drain of {device_id} still running")
                del self._drain_threads[device_id]
                err = self._drain_errors.pop(device_id, None)
                if err is not None:
                    raise ExecError(err)
                return  # completed successfully

        gpu = self.find_gpu(node, device_id)
        if gpu is None:
            return
        if not self.destructive:
            with self._sim_lock:
                self._sim_detached.add(device_id)
            self._invalidate_enum(node)
            return
        remaining = [g for g in self.enumerate(node) if g.device_id != device_id]
        if remaining:
            self.execer.write_file(node, f"/sys/bus/pci/devices/{gpu.pci_bdf}/remove", "1")
            self._invalidate_enum(node)
            return

        # LAST device: module unload + sysfs remove can block for a long
        # time while KFD tears down, and the remove can stall the writer —
        # run asynchronously and report progress on re-checks, the pattern
        # the reference uses for its async sysfs remove (gpus.go:1534-1585)
        # resolved before the thread starts: the chroot decision must not
        # race a DeviceConfig change mid-drain
        modprobe_argv = self._module_argv(node, ["modprobe", "-r", "amdgpu"])

        def unload_and_remove():
            try:
                rc, _, err = self.execer.run(node, modprobe_argv, timeout=120)
                if rc != 0:
                    raise ExecError(f"modprobe -r amdgpu failed: {err}", rc=rc, stderr=err)
                self.execer.write_file(
                    node, f"/sys/bus/pci/devices/{gpu.pci_bdf}/remove", "1"
                )
                self._invalidate_enum(node)
            except Exception as exc:
                with self._drain_lock:
                    self._drain_errors[device_id] = str(exc)

        t = threading.Thread(target=unload_and_remove, name=f"drain-{device_id}", daemon=True)
        with self._drain_lock:
            self._drain_threads[device_id] = t
        t.start()
        raise DrainInProgress(f"last-device drain of {device_id} started")

    def refresh_after_attach(self, node: str) -> None:
        """Make a newly composed device enumerable and published.

        PCI rescan binds the amdgpu driver to the hot-added function; then
        the node's ResourceSlice is re-published (the standalone analog of
        restarting the DRA kubelet-plugin pod, gpus.go:1109-1146) and the
        device-plugin/metrics restart hooks fire (nvidia-device-plugin /
        nvidia-dcgm parity, composableresource_controller.go:257-269).
        """
        if self.destructive:
            self.execer.write_file(node, "/sys/bus/pci/rescan", "1")
        # non-destructive mode: simulate_compose() already restored the
        # device into the enumerable set; rescan is a no-op by design
        self._invalidate_enum(node)  # topology just changed
        for name, hook in self.restart_hooks.items():
            try:
                hook(node)
            except Exception as exc:  # hook failures are non-fatal, like
                log.warning("restart hook %s failed on %s: %s", name, node, exc)
        self._publish_slice(node)

    def simulate_compose(self, node: str, device_id: str) -> None:
        """Non-destructive mode: mark a device as fabric-composed so the next
        rescan+enumeration sees it (models hot-add on a box with no fabric)."""
        with self._sim_lock:
            self._sim_detached.discard(device_id)
        self._invalidate_enum(node)

    def refresh_after_detach(self, node: str) -> None:
        for name, hook in self.restart_hooks.items():
            try:
                hook(node)
            except Exception as exc:
                log.warning("restart hook %s failed on %s: %s", name, node, exc)
        self._publish_slice(node)

    def _publish_slice(self, node: str) -> None:
        if self.client is None or not self.publish_slices:
            return
        devices = [
            ResourceSliceDevice(
                name=f"gpu-{i}",
                uuid=g.device_id,
                model="mi355x",
                node=node,
                attributes={
                    "pci-bdf": g.pci_bdf,
                    "vram-bytes": str(g.vram_bytes),
                    "xgmi-peers": ",".join(map(str, g.xgmi_peers)),
                },
            )
            for i, g in enumerate(self.enumerate(node))
        ]
        name = f"{node}-gpu-pool"
        existing = self.client.try_get(ResourceSlice, name)
        if existing is None:
            sl = ResourceSlice()
            sl.metadata.name = name
            sl.spec = ResourceSliceSpec(node_name=node, pool=name, devices=devices)
            self.client.create(sl)
        else:
            existing.spec.devices = devices
            self.client.update(existing)

    # -- CDI ---------------------------------------------------------------

    def write_cdi(self, node: str, device_id: str) -> str:
        gpu = self.find_gpu(node, device_id)
        if gpu is None:
            raise ExecError(f"device {device_id} not enumerable; cannot write CDI spec")
        return self.cdi.add_device(node, gpu)

    def remove_cdi(self, node: str, device_id: str) -> None:
        self.cdi.remove_device(node, device_id)

    # -- health ------------------------------------------------------------

    def health_probe(self, node: str, device_id: str) -> Optional[dict]:
        if self.probe_fn is None:
            return None
        gpu = self.find_gpu(node, device_id)
        if gpu is None:
            return None
        return self.probe_fn(gpu)


class MockNodeOps(NodeOps):
    """In-memory NodeOps double for control-plane tests and CPU bench.

    Models per-node visible-device sets with explicit fabric→node latency:
    a device becomes visible only after ``refresh_after_attach`` (the PCI
    rescan analog) once the mock fabric composed it.
    """

    def __init__(self, client=None, attach_visible_delay: float = 0.0):
        self.client = client
        self.driver_present: Dict[str, bool] = {}
        self.composed: Dict[str, Set[str]] = {}  # fabric-composed, pre-rescan
        self.visible: Dict[str, Set[str]] = {}  # enumerable after rescan
        self.loads: Dict[str, Set[str]] = {}  # device ids under load ("*" = node)
        self.cdi_written: Dict[str, Set[str]] = {}
        # bounded call log: long CPU soaks/benches run through this mock
        # (a 16k-cycle leak check traced its only RSS growth here)
        from collections import deque

        self.calls: "deque[tuple]" = deque(maxlen=10000)
        self.attach_visible_delay = attach_visible_delay
        self._visible_at: Dict[tuple, float] = {}
        self._lock = threading.RLock()

    # test setup helpers
    def set_driver(self, node: str, present: bool = True) -> None:
        self.driver_present[node] = present

    def fabric_composed(self, node: str, device_id: str) -> None:
        with self._lock:
            self.composed.setdefault(node, set()).add(device_id)

    def fabric_removed(self, node: str, device_id: str) -> None:
        with self._lock:
            self.composed.get(node, set()).discard(device_id)

    def add_load(self, node: str, device_id: str = "*") -> None:
        with self._lock:
            self.loads.setdefault(node, set()).add(device_id)

    def clear_loads(self, node: str) -> None:
        with self._lock:
            self.loads.pop(node, None)

    # NodeOps surface
    def ensure_driver(self, node: str) -> None:
        self.calls.append(("ensure_driver", node))
        if not self.driver_present.get(node, True):
            raise DriverMissing(f"amdgpu not loaded on {node}")

    def enumerate(self, node: str):
        with self._lock:
            return sorted(self.visible.get(node, set()))

    def is_visible(self, node: str, device_id: str) -> bool:
        with self._lock:
            ready_at = self._visible_at.get((node, device_id))
            if ready_at is not None and time.monotonic() >= ready_at:
                self.visible.setdefault(node, set()).add(device_id)
                del self._visible_at[(node, device_id)]
            return device_id in self.visible.get(node, set())

    def is_visible_dra(self, node: str, device_id: str) -> bool:
        return self.is_visible(node, device_id)

    def check_no_loads(self, node: str, device_id: Optional[str] = None) -> None:
        self.calls.append(("check_no_loads", node, device_id))
        with self._lock:
            loads = self.loads.get(node, set())
            if device_id is None:
                if loads:
                    raise GPULoadsPresent(f"node {node} busy: {sorted(loads)}")
            elif device_id in loads or "*" in loads:
                raise GPULoadsPresent(f"device {device_id} busy")

    def drain(self, node: str, device_id: str) -> None:
        self.calls.append(("drain", node, device_id))
        with self._lock:
            self.visible.get(node, set()).discard(device_id)

    def refresh_after_attach(self, node: str) -> None:
        self.calls.append(("refresh_after_attach", node))
        with self._lock:
            for d in self.composed.get(node, set()) - self.visible.get(node, set()):
                if self.attach_visible_delay > 0:
                    self._visible_at.setdefault(
                        (node, d), time.monotonic() + self.attach_visible_delay
                    )
                else:
                    self.visible.setdefault(node, set()).add(d)

    def refresh_after_detach(self, node: str) -> None:
        self.calls.append(("refresh_after_detach", node))

    def write_cdi(self, node: str, device_id: str) -> str:
        self.calls.append(("write_cdi", node, device_id))
        with self._lock:
            self.cdi_written.setdefault(node, set()).add(device_id)
        return f"amd.com/gpu={device_id}"

    def remove_cdi(self, node: str, device_id: str) -> None:
        self.calls.append(("remove_cdi", node, device_id))
        with self._lock:
            self.cdi_written.get(node, set()).discard(device_id)
