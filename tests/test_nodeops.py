"""Node-ops unit tests: KFD topology parsing (fixture sysfs tree), compute
process scan, CDI spec golden output, capacity checks, debouncer, taints."""

import json

import pytest

from cro_amd.api.v1alpha1.types import DeviceTaintRule, NodeSpecRequirements
from cro_amd.nodeops.amdgpu import AmdNodeOps, DriverMissing
from cro_amd.nodeops.cdi_spec import CDISpecWriter
from cro_amd.nodeops.execs import ExecError, MockNodeExec
from cro_amd.nodeops.kfd import (
    KFD_NODES,
    KFD_PROC,
    canonical_device_id,
    enumerate_gpus,
    gpu_compute_pids,
)
from cro_amd.nodeops.nodes import Debouncer, check_node_capacity_sufficient
from cro_amd.nodeops import taints
from tests.conftest import make_node, make_resource

NODE = "node0"


def kfd_fixture(execer: MockNodeExec, n_gpus: int = 2, node: str = NODE):
    """Builds a faithful KFD topology tree: one CPU node + n GPU nodes with
    unique_id, render minors, xGMI links between the GPUs."""
    execer.set_file(node, f"{KFD_NODES}/0/properties", "cpu_cores_count 96\nsimd_count 0\n")
    ids = []
    for i in range(n_gpus):
        base = f"{KFD_NODES}/{i + 1}"
        uid = 0xABC0000 + i
        loc = ((3 + i) << 8)  # bus 03, 04, ...
        execer.set_file(
            node,
            f"{base}/properties",
            f"simd_count 1024\nunique_id {uid}\ndrm_render_minor {128 + i}\n"
            f"vendor_id 4098\ndevice_id 29857\nlocation_id {loc}\ndomain 0\n"
            "gfx_target_version 90500\n",
        )
        execer.set_file(node, f"{base}/gpu_id", str(1000 + i))
        # DRM card mapping is by PCI slot, deliberately NOT render-128
        # (real boxes renumber: renderD144 ↔ card16)
        execer.set_file(
            node, f"/sys/class/drm/card{i}/device/uevent",
            f"DRIVER=amdgpu\nPCI_SLOT_NAME=0000:{3 + i:02x}:00.0\n",
        )
        execer.set_file(node, f"/dev/dri/card{i}", "")  # device file present
        execer.set_file(
            node, f"{base}/mem_banks/0/properties",
            "heap_type 1\nsize_in_bytes 309237645312\n",
        )
        for j in range(n_gpus):
            if j != i:
                execer.set_file(
                    node, f"{base}/io_links/{j}/properties",
                    f"type 11\nnode_to {j + 1}\n",
                )
        ids.append(canonical_device_id(uid, f"0000:{3 + i:02x}:00.0"))
    execer.set_file(node, "/sys/module/amdgpu/version", "6.x")
    return ids


def test_enumerate_gpus_from_kfd():
    ex = MockNodeExec()
    ids = kfd_fixture(ex, 2)
    gpus = enumerate_gpus(ex, NODE)
    assert len(gpus) == 2
    g0 = gpus[0]
    assert g0.device_id == ids[0]
    assert g0.device_id == f"GPU-{0xABC0000:016x}"
    assert g0.render_minor == 128
    assert g0.render_path == "/dev/dri/renderD128"
    assert g0.card_path == "/dev/dri/card0"
    assert g0.pci_bdf == "0000:03:00.0"
    assert g0.vram_bytes == 309237645312  # 288 GB HBM3E
    assert g0.xgmi_peers == [2]
    assert g0.gpu_id == 1000


def test_enumerate_skips_cpu_nodes():
    ex = MockNodeExec()
    kfd_fixture(ex, 1)
    gpus = enumerate_gpus(ex, NODE)
    assert len(gpus) == 1  # node 0 (CPU) skipped


def test_enumerate_without_kfd_raises():
    ex = MockNodeExec()
    with pytest.raises(ExecError):
        enumerate_gpus(ex, NODE)


def test_pci_fallback_device_id():
    assert canonical_device_id(0, "0000:03:00.0") == "GPU-pci-0000:03:00.0"


def test_gpu_compute_pids_per_device():
    ex = MockNodeExec()
    ex.set_file(NODE, f"{KFD_PROC}/1234/vram_1000", "1048576")
    ex.set_file(NODE, f"{KFD_PROC}/1234/vram_1001", "0")
    ex.set_file(NODE, f"{KFD_PROC}/5678/vram_1000", "0")
    ex.set_file(NODE, f"{KFD_PROC}/5678/vram_1001", "4096")
    assert gpu_compute_pids(ex, NODE, 1000) == [1234]
    assert gpu_compute_pids(ex, NODE, 1001) == [5678]
    assert sorted(gpu_compute_pids(ex, NODE)) == [1234, 5678]


def test_gpu_compute_pids_empty_when_no_proc():
    ex = MockNodeExec()
    assert gpu_compute_pids(ex, NODE) == []


def test_cdi_writer_add_remove(tmp_path):
    ex = MockNodeExec()
    kfd_fixture(ex, 2)
    gpus = enumerate_gpus(ex, NODE)
    writer = CDISpecWriter(ex, cdi_dir="/etc/cdi")
    cdi_id = writer.add_device(NODE, gpus[0])
    assert cdi_id == f"amd.com/gpu={gpus[0].device_id}"
    spec = json.loads(ex.files[(NODE, "/etc/cdi/amd.com-gpu-cro.json")])
    assert spec["cdiVersion"] == "0.6.0"
    assert spec["kind"] == "amd.com/gpu"
    assert {"path": "/dev/kfd"} in spec["containerEdits"]["deviceNodes"]
    dev = spec["devices"][0]
    assert dev["name"] == gpus[0].device_id
    paths = [d["path"] for d in dev["containerEdits"]["deviceNodes"]]
    assert "/dev/dri/renderD128" in paths and "/dev/dri/card0" in paths
    assert dev["annotations"]["cro.amd.com/xgmi-peers"] == "2"
    assert dev["annotations"]["cro.amd.com/vram-bytes"] == "309237645312"
    # add second, remove first
    writer.add_device(NODE, gpus[1])
    writer.remove_device(NODE, gpus[0].device_id)
    assert writer.devices(NODE) == [gpus[1].device_id]


def test_cdi_add_is_idempotent():
    ex = MockNodeExec()
    kfd_fixture(ex, 1)
    gpus = enumerate_gpus(ex, NODE)
    writer = CDISpecWriter(ex)
    writer.add_device(NODE, gpus[0])
    writer.add_device(NODE, gpus[0])
    assert len(writer.devices(NODE)) == 1


def test_amd_node_ops_driver_and_visibility():
    ex = MockNodeExec()
    ids = kfd_fixture(ex, 2)
    ops = AmdNodeOps(ex, cdi_dir="/etc/cdi")
    ops.ensure_driver(NODE)
    assert ops.is_visible(NODE, ids[0])
    assert not ops.is_visible(NODE, "GPU-nope")
    ex2 = MockNodeExec()  # no amdgpu module
    ops2 = AmdNodeOps(ex2)
    with pytest.raises(DriverMissing):
        ops2.ensure_driver(NODE)


def test_amd_node_ops_simulated_lifecycle():
    """Non-destructive mode: drain hides the device, compose+rescan restores
    it — the bench-box lifecycle."""
    ex = MockNodeExec()
    ids = kfd_fixture(ex, 2)
    ops = AmdNodeOps(ex, destructive=False, initially_detached=[ids[0]])
    assert not ops.is_visible(NODE, ids[0])
    assert ops.is_visible(NODE, ids[1])
    ops.simulate_compose(NODE, ids[0])
    ops.refresh_after_attach(NODE)
    assert ops.is_visible(NODE, ids[0])
    ops.drain(NODE, ids[0])
    assert not ops.is_visible(NODE, ids[0])


def test_amd_node_ops_destructive_drain_writes_sysfs():
    ex = MockNodeExec()
    ids = kfd_fixture(ex, 2)
    ops = AmdNodeOps(ex, destructive=True)
    ops.drain(NODE, ids[0])
    assert ex.files[(NODE, "/sys/bus/pci/devices/0000:03:00.0/remove")] == "1"


def test_amd_node_ops_last_gpu_drain_unloads_module():
    import time

    from cro_amd.nodeops.amdgpu import DrainInProgress

    ex = MockNodeExec()
    ids = kfd_fixture(ex, 1)
    ex.set_command(("modprobe", "-r", "amdgpu"), (0, "", ""))
    ops = AmdNodeOps(ex, destructive=True)
    # last device → asynchronous module unload + remove
    with pytest.raises(DrainInProgress):
        ops.drain(NODE, ids[0])
    time.sleep(0.05)
    ops.drain(NODE, ids[0])  # completion check
    assert ("run", NODE, ("modprobe", "-r", "amdgpu")) in ex.calls
    assert ex.files[(NODE, "/sys/bus/pci/devices/0000:03:00.0/remove")] == "1"


def test_capacity_check(client):
    make_node(client, "node0", milli_cpu=8000, memory=1 << 30, pods=10)
    assert check_node_capacity_sufficient(
        client, "node0", NodeSpecRequirements(milli_cpu=4000)
    )
    assert not check_node_capacity_sufficient(
        client, "node0", NodeSpecRequirements(milli_cpu=16000)
    )
    assert not check_node_capacity_sufficient(
        client, "node0", NodeSpecRequirements(allowed_pod_number=100)
    )


def test_debouncer():
    import time

    d = Debouncer(interval=0.1)
    calls = []
    assert d("k", lambda: calls.append(1)) is True
    assert d("k", lambda: calls.append(2)) is False  # debounced
    assert d("other", lambda: calls.append(3)) is True  # separate key
    time.sleep(0.11)
    assert d("k", lambda: calls.append(4)) is True
    assert calls == [1, 3, 4]


def test_taint_helpers(client):
    r = make_resource("gpu-1")
    r.status.device_id = "GPU-x"
    taints.create_device_taint(client, r)
    taints.create_device_taint(client, r)  # idempotent
    assert taints.has_device_taint(client, r)
    rules = client.list(DeviceTaintRule)
    assert len(rules) == 1
    assert rules[0].spec.device_uuid == "GPU-x"
    assert rules[0].spec.effect == "NoSchedule"
    taints.delete_device_taint(client, r)
    taints.delete_device_taint(client, r)  # idempotent
    assert not taints.has_device_taint(client, r)


def test_node_wide_loads_only_count_visible_gpus():
    """Node-wide (DEVICE_PLUGIN) check must ignore other tenants' processes
    whose VRAM sits on GPUs this node cannot enumerate."""
    ex = MockNodeExec()
    kfd_fixture(ex, 1)  # our gpu_id is 1000
    ex.set_file(NODE, f"{KFD_PROC}/5555/vram_9999", "1048576")  # foreign GPU
    ex.set_file(NODE, f"{KFD_PROC}/6666/vram_1000", "0")
    ops = AmdNodeOps(ex)
    ops.check_no_loads(NODE)  # no load on OUR gpus → passes
    ex.set_file(NODE, f"{KFD_PROC}/7777/vram_1000", "2048")
    with pytest.raises(Exception, match="7777"):
        ops.check_no_loads(NODE)


def test_last_gpu_drain_is_async():
    """Last-device drain runs asynchronously: first call raises
    DrainInProgress, completion is reported on re-check (the reference's
    async sysfs-remove pattern, gpus.go:1534-1585)."""
    import time

    from cro_amd.nodeops.amdgpu import DrainInProgress

    ex = MockNodeExec()
    ids = kfd_fixture(ex, 1)
    ex.set_command(("modprobe", "-r", "amdgpu"), (0, "", ""), delay=0.1)
    ops = AmdNodeOps(ex, destructive=True)
    with pytest.raises(DrainInProgress):
        ops.drain(NODE, ids[0])
    with pytest.raises(DrainInProgress):
        ops.drain(NODE, ids[0])  # still unloading
    time.sleep(0.15)
    ops.drain(NODE, ids[0])  # completed
    assert ex.files[(NODE, "/sys/bus/pci/devices/0000:03:00.0/remove")] == "1"


def test_last_gpu_drain_async_error_surfaces():
    import time

    from cro_amd.nodeops.amdgpu import DrainInProgress
    from cro_amd.nodeops.execs import ExecError

    ex = MockNodeExec()
    ids = kfd_fixture(ex, 1)
    ex.set_command(("modprobe", "-r", "amdgpu"), (1, "", "module in use"), delay=0.05)
    ops = AmdNodeOps(ex, destructive=True)
    with pytest.raises(DrainInProgress):
        ops.drain(NODE, ids[0])
    time.sleep(0.1)
    with pytest.raises(ExecError, match="module in use"):
        ops.drain(NODE, ids[0])
    # after the error is consumed, drain can be retried from scratch
    ex.set_command(("modprobe", "-r", "amdgpu"), (0, "", ""))
    with pytest.raises(DrainInProgress):
        ops.drain(NODE, ids[0])
    time.sleep(0.05)
    ops.drain(NODE, ids[0])


def test_restart_daemonset_annotation_and_debounce(client):
    import time as _time

    from cro_amd.api.v1alpha1.types import DaemonSet
    from cro_amd.nodeops.nodes import (
        RESTARTED_AT_ANNOTATION,
        restart_daemonset,
    )
    from cro_amd.runtime.errors import NotFoundError

    ds = DaemonSet()
    ds.metadata.name = "amd-gpu-operator/amd-device-plugin"
    ds.status.desired_number_scheduled = 1
    ds.status.number_ready = 1
    ds.status.current_number_scheduled = 1
    client.create(ds)
    assert restart_daemonset(client, "amd-gpu-operator", "amd-device-plugin") is True
    got = client.get(DaemonSet, "amd-gpu-operator/amd-device-plugin")
    stamp = got.spec.template_annotations[RESTARTED_AT_ANNOTATION]
    assert stamp.endswith("Z")
    # immediate second restart is debounced (nodes.go:56-67)
    assert restart_daemonset(client, "amd-gpu-operator", "amd-device-plugin") is False
    got2 = client.get(DaemonSet, "amd-gpu-operator/amd-device-plugin")
    assert got2.spec.template_annotations[RESTARTED_AT_ANNOTATION] == stamp
    with pytest.raises(NotFoundError):
        restart_daemonset(client, "amd-gpu-operator", "nope")


def test_device_plugin_mode_restarts_daemonsets(mock_world):
    from cro_amd.api.v1alpha1.types import ComposableResource, DaemonSet
    from cro_amd.nodeops.nodes import RESTARTED_AT_ANNOTATION

    mock_world.adapter.device_resource_type = "DEVICE_PLUGIN"
    for name in ("amd-device-plugin", "amd-metrics-exporter"):
        ds = DaemonSet()
        ds.metadata.name = f"amd-gpu-operator/{name}"
        ds.status.desired_number_scheduled = 1
        ds.status.number_ready = 1
        ds.status.current_number_scheduled = 1
        mock_world.client.create(ds)
    make_node(mock_world.client, "node0")
    from tests.conftest import make_resource

    mock_world.client.create(make_resource("gpu-1"))
    mock_world.resource_rec.reconcile("gpu-1")  # None → Attaching
    mock_world.resource_rec.reconcile("gpu-1")  # Attaching → Online
    assert (
        mock_world.client.get(ComposableResource, "gpu-1").status.state == "Online"
    )
    for name in ("amd-device-plugin", "amd-metrics-exporter"):
        got = mock_world.client.get(DaemonSet, f"amd-gpu-operator/{name}")
        assert RESTARTED_AT_ANNOTATION in got.spec.template_annotations


def test_amdsmi_fallback_parses_real_format():
    """Parser vs a captured real `amd-smi list --json` payload
    (MI355X / ROCm 7.2)."""
    from cro_amd.nodeops.kfd import enumerate_gpus_amdsmi

    ex = MockNodeExec()
    ex.set_command(
        ("amd-smi", "list", "--json"),
        (0, '[{"gpu": 0, "bdf": "0000:f1:00.0", '
            '"uuid": "d0ff75a3-0000-1000-805e-ceb808463c96", '
            '"kfd_id": 8465, "node_id": 6, "partition_id": 0}]', ""),
    )
    gpus = enumerate_gpus_amdsmi(ex, NODE)
    assert len(gpus) == 1
    g = gpus[0]
    assert g.pci_bdf == "0000:f1:00.0"
    assert g.gpu_id == 8465  # kfd_id drives /sys/class/kfd/kfd/proc attribution
    assert g.kfd_node == 6
    assert g.device_id.startswith("GPU-")


def test_restart_daemonset_stability_guards(client):
    """Stability guards (nodes.go:40-50): no restart when nothing is
    scheduled or a rollout is in flight; unparseable stamp errors."""
    from cro_amd.api.v1alpha1.types import DaemonSet
    from cro_amd.nodeops.nodes import RESTARTED_AT_ANNOTATION, restart_daemonset

    ds = DaemonSet()
    ds.metadata.name = "ns/ds"
    client.create(ds)  # desired == 0
    assert restart_daemonset(client, "ns", "ds") is False

    got = client.get(DaemonSet, "ns/ds")
    got.status.desired_number_scheduled = 2
    got.status.number_ready = 1  # rollout in flight
    got.status.current_number_scheduled = 2
    client.update_status(got)
    assert restart_daemonset(client, "ns", "ds") is False

    got = client.get(DaemonSet, "ns/ds")
    got.status.number_ready = 2
    client.update_status(got)
    assert restart_daemonset(client, "ns", "ds") is True

    got = client.get(DaemonSet, "ns/ds")
    got.spec.template_annotations[RESTARTED_AT_ANNOTATION] = "not-a-time"
    client.update(got)
    with pytest.raises(ValueError, match="restartedAt"):
        restart_daemonset(client, "ns", "ds")


def test_trusted_binary_resolution(tmp_path):
    """LocalNodeExec resolves binaries only from the trusted path list and
    rejects path-qualified names (gpus.go:996-1038 trust model)."""
    from cro_amd.nodeops.execs import ExecError, LocalNodeExec

    ex = LocalNodeExec()
    assert ex.resolve_binary("modprobe").startswith(("/usr/sbin", "/sbin", "/usr/bin"))
    with pytest.raises(ExecError, match="bare"):
        ex.resolve_binary("/tmp/evil/nvidia-smi")
    with pytest.raises(ExecError, match="trusted"):
        ex.resolve_binary("definitely-not-a-binary-xyz")


def test_amdsmi_fallback_via_node_ops():
    """KFD missing entirely → AmdNodeOps falls back to amd-smi enumeration."""
    ex = MockNodeExec()
    ex.set_file(NODE, "/sys/module/amdgpu/version", "6.x")
    ex.set_command(
        ("amd-smi", "list", "--json"),
        (0, '[{"gpu": 0, "bdf": "0000:0a:00.0", '
            '"uuid": "deadbeef-0000-1000-8000-000000000000", '
            '"kfd_id": 111, "node_id": 2, "partition_id": 0}]', ""),
    )
    ops = AmdNodeOps(ex)
    gpus = ops.enumerate(NODE)
    assert len(gpus) == 1 and gpus[0].pci_bdf == "0000:0a:00.0"


def test_enum_ttl_cache_and_invalidation():
    ex = MockNodeExec()
    ids = kfd_fixture(ex, 1)
    ops = AmdNodeOps(ex, destructive=False)
    ops.enum_cache_ttl = 10.0  # effectively permanent for this test
    assert ops.is_visible(NODE, ids[0])
    reads_before = len(ex.calls)
    ops.is_visible(NODE, ids[0])  # served from cache
    assert len(ex.calls) == reads_before
    ops.drain(NODE, ids[0])  # lifecycle mutation invalidates
    assert not ops.is_visible(NODE, ids[0])
    assert len(ex.calls) > reads_before  # re-read happened


def test_static_meta_survives_invalidation():
    ex = MockNodeExec()
    ids = kfd_fixture(ex, 1)
    ops = AmdNodeOps(ex, destructive=False)
    ops.enumerate(NODE)
    mem_reads = [c for c in ex.calls if c[0] == "read" and "mem_banks" in c[2]]
    ops._invalidate_enum(NODE)
    ops.enumerate(NODE)
    mem_reads_after = [c for c in ex.calls if c[0] == "read" and "mem_banks" in c[2]]
    # vram banks were NOT re-read (static cache), yet values are present
    assert len(mem_reads_after) == len(mem_reads)
    assert ops.enumerate(NODE)[0].vram_bytes == 309237645312


def test_enum_generation_id_revalidation():
    """After the TTL, an unchanged /sys/class/kfd/kfd/topology/generation_id
    revalidates the cached enumeration (one sysfs read, no topology walk);
    a bumped generation_id forces the full walk."""
    ex = MockNodeExec()
    ids = kfd_fixture(ex, 2)
    gen_path = "/sys/class/kfd/kfd/topology/generation_id"
    ex.set_file(NODE, gen_path, "7\n")
    ops = AmdNodeOps(ex, cdi_dir="/etc/cdi", destructive=True)
    ops.enum_cache_ttl = 0.0  # every call is past the TTL

    assert [g.device_id for g in ops.enumerate(NODE)] == ids
    # break the topology files; unchanged generation_id → cache still served
    saved = {k: v for k, v in ex.files.items() if "topology/nodes" in k[1]}
    for k in saved:
        ex.files.pop(k)
    assert [g.device_id for g in ops.enumerate(NODE)] == ids

    # rebuild with one fewer GPU and bump the generation → full re-walk sees it
    kfd_fixture(ex, 1)
    ex.set_file(NODE, gen_path, "8\n")
    # clear per-device static caches tied to the old tree
    ops._static_meta.clear()
    assert len(ops.enumerate(NODE)) == 1

    # explicit invalidation always forces a walk even with stable gen
    ops._invalidate_enum(NODE)
    assert len(ops.enumerate(NODE)) == 1


def test_driver_detection_chain():
    """DeviceConfig (container driver) → daemonset readiness gate;
    no DeviceConfig → host /sys/module/amdgpu; neither → DriverMissing
    (gpus.go:97-193 chain)."""
    from cro_amd.api.v1alpha1.types import DaemonSet, DeviceConfig
    from cro_amd.runtime.client import Client
    from cro_amd.runtime.store import InMemoryStore

    client = Client(InMemoryStore())
    ex = MockNodeExec()
    kfd_fixture(ex, 1)
    ops = AmdNodeOps(ex, client=client, cdi_dir="/etc/cdi")

    # host mode: module present, no DeviceConfig
    assert ops.driver_mode(NODE) == "host"
    ops.ensure_driver(NODE)

    # container mode: DeviceConfig enables the driver → daemonset must exist
    dc = DeviceConfig()
    dc.metadata.name = "default"
    dc.spec.driver.enable = True
    client.create(dc)
    assert ops.driver_mode(NODE) == "container"
    with pytest.raises(DriverMissing, match="not found"):
        ops.ensure_driver(NODE)

    ds = DaemonSet()
    ds.metadata.name = "amd-gpu-operator/amd-gpu-driver"
    ds.status.desired_number_scheduled = 1
    ds.status.number_ready = 0
    client.create(ds)
    with pytest.raises(DriverMissing, match="not ready"):
        ops.ensure_driver(NODE)

    ds = client.get(DaemonSet, "amd-gpu-operator/amd-gpu-driver")
    ds.status.number_ready = 1
    client.update_status(ds)
    ops.ensure_driver(NODE)  # ready + module loaded → passes

    # module gone ⇒ even a ready containerized driver fails the gate
    ex.files.pop((NODE, "/sys/module/amdgpu/version"), None)
    for k in [k for k in list(ex.dirs) if "module/amdgpu" in k[1]] if hasattr(ex, "dirs") else []:
        ex.dirs.pop(k)
    if not ex.path_exists(NODE, "/sys/module/amdgpu"):
        with pytest.raises(DriverMissing, match="not loaded"):
            ops.ensure_driver(NODE)

    # none: fresh ops without module or DeviceConfig
    bare = AmdNodeOps(MockNodeExec(), cdi_dir="/etc/cdi")
    assert bare.driver_mode(NODE) == "none"


def _container_driver_client(driver_root="/run/amdgpu-driver"):
    from cro_amd.api.v1alpha1.types import DaemonSet, DeviceConfig
    from cro_amd.runtime.client import Client
    from cro_amd.runtime.store import InMemoryStore

    client = Client(InMemoryStore())
    dc = DeviceConfig()
    dc.metadata.name = "default"
    dc.spec.driver.enable = True
    dc.spec.driver.driver_root = driver_root
    client.create(dc)
    ds = DaemonSet()
    ds.metadata.name = "amd-gpu-operator/amd-gpu-driver"
    ds.status.desired_number_scheduled = 1
    ds.status.number_ready = 1
    client.create(ds)
    return client


def test_last_gpu_drain_container_arm_chroots_modprobe(monkeypatch):
    """Drain path B (gpus.go:566-749 parity): with a containerized driver
    the module unload runs through the driver container's rootfs —
    ``chroot <driver_root> modprobe -r amdgpu`` — because the kernel
    modules live there, not on the host."""
    import time

    from cro_amd.nodeops.amdgpu import DrainInProgress

    monkeypatch.delenv("CRO_DRIVER_ROOT", raising=False)
    ex = MockNodeExec()
    ids = kfd_fixture(ex, 1)
    ex.set_command(("chroot", "/run/amdgpu-driver", "modprobe", "-r", "amdgpu"), (0, "", ""))
    ops = AmdNodeOps(ex, client=_container_driver_client(), destructive=True)
    assert ops.driver_mode(NODE) == "container"
    assert ops.driver_root(NODE) == "/run/amdgpu-driver"

    with pytest.raises(DrainInProgress):
        ops.drain(NODE, ids[0])
    time.sleep(0.05)
    ops.drain(NODE, ids[0])  # completion check
    assert ("run", NODE, ("chroot", "/run/amdgpu-driver", "modprobe", "-r", "amdgpu")) in ex.calls
    # bare modprobe must NOT have run on the host
    assert ("run", NODE, ("modprobe", "-r", "amdgpu")) not in ex.calls
    # sysfs hot-remove stays host-global (sysfs is shared with the container)
    assert ex.files[(NODE, "/sys/bus/pci/devices/0000:03:00.0/remove")] == "1"


def test_last_gpu_drain_host_arm_plain_modprobe(monkeypatch):
    """Host-driver arm: no chroot wrapping (drain path A)."""
    import time

    from cro_amd.nodeops.amdgpu import DrainInProgress

    monkeypatch.delenv("CRO_DRIVER_ROOT", raising=False)
    ex = MockNodeExec()
    ids = kfd_fixture(ex, 1)
    ex.set_command(("modprobe", "-r", "amdgpu"), (0, "", ""))
    ops = AmdNodeOps(ex, destructive=True)
    assert ops.driver_mode(NODE) == "host"
    assert ops.driver_root(NODE) == ""
    with pytest.raises(DrainInProgress):
        ops.drain(NODE, ids[0])
    time.sleep(0.05)
    ops.drain(NODE, ids[0])
    assert ("run", NODE, ("modprobe", "-r", "amdgpu")) in ex.calls


def test_driver_root_env_override(monkeypatch):
    monkeypatch.setenv("CRO_DRIVER_ROOT", "/custom/driver/root")
    ex = MockNodeExec()
    kfd_fixture(ex, 1)
    ops = AmdNodeOps(ex, client=_container_driver_client("/ignored"))
    assert ops.driver_root(NODE) == "/custom/driver/root"
    assert ops._module_argv(NODE, ["modprobe", "-r", "amdgpu"]) == [
        "chroot", "/custom/driver/root", "modprobe", "-r", "amdgpu"]


def test_exec_probe_container_arm(monkeypatch):
    """The gfx950 exec-probe runs through the driver-container chroot in
    container mode (nvidia-smi-through-chroot analog)."""
    import json

    from cro_amd.nodeops.probe import make_exec_probe_fn

    monkeypatch.delenv("CRO_DRIVER_ROOT", raising=False)
    ex = MockNodeExec()
    ids = kfd_fixture(ex, 1)
    ops = AmdNodeOps(ex, client=_container_driver_client())
    probe_result = {"ok": True, "mfma_f32_exact": True}
    ex.set_command(
        ("chroot", "/run/amdgpu-driver", "croagent", "probe", "--bdf", "0000:03:00.0"),
        (0, json.dumps(probe_result), ""),
    )
    probe = make_exec_probe_fn(
        ex, NODE, argv_prefix_fn=lambda: ops.probe_argv_prefix(NODE))
    gpu = ops.find_gpu(NODE, ids[0])
    assert probe(gpu)["ok"] is True
    assert ("run", NODE,
            ("chroot", "/run/amdgpu-driver", "croagent", "probe", "--bdf", "0000:03:00.0")
            ) in ex.calls
