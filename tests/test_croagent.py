"""croagent (native node-agent CLI) tests against a fixture sysroot.

The binary is gfx950-targeted but its sysfs surface (list/pids/drain/rescan)
runs host-side, so everything except `probe` is CPU-testable here.
"""

import json
import os
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
AGENT = os.path.join(REPO, "cro_amd", "agent", "croagent")


@pytest.fixture(scope="module")
def agent():
    if not os.path.exists(AGENT):
        subprocess.run(
            ["python3", "-m", "cro_amd.hip.build"], cwd=REPO, check=True
        )
    return AGENT


@pytest.fixture
def sysroot(tmp_path):
    base = tmp_path / "sys" / "class" / "kfd" / "kfd" / "topology" / "nodes"
    (base / "0").mkdir(parents=True)
    (base / "0" / "properties").write_text("cpu_cores_count 96\nsimd_count 0\n")
    g = base / "4"
    g.mkdir()
    g.joinpath("properties").write_text(
        "simd_count 1024\nunique_id 12600702146510664649\n"
        "drm_render_minor 144\nlocation_id 23040\ndomain 0\n"
        "gfx_target_version 90500\n"
    )
    g.joinpath("gpu_id").write_text("28206\n")
    (g / "mem_banks" / "0").mkdir(parents=True)
    (g / "mem_banks" / "0" / "properties").write_text(
        "heap_type 1\nsize_in_bytes 309237645312\n"
    )
    (g / "io_links" / "0").mkdir(parents=True)
    (g / "io_links" / "0" / "properties").write_text("type 11\nnode_to 5\n")
    drm = tmp_path / "sys" / "class" / "drm" / "card8" / "device"
    drm.mkdir(parents=True)
    (drm / "uevent").write_text("DRIVER=amdgpu\nPCI_SLOT_NAME=0000:5a:00.0\n")
    proc = tmp_path / "sys" / "class" / "kfd" / "kfd" / "proc"
    (proc / "1234").mkdir(parents=True)
    (proc / "1234" / "vram_28206").write_text("1048576\n")
    (proc / "777").mkdir()
    (proc / "777" / "vram_28206").write_text("0\n")
    pci = tmp_path / "sys" / "bus" / "pci" / "devices" / "0000:5a:00.0"
    pci.mkdir(parents=True)
    (tmp_path / "sys" / "bus" / "pci" / "rescan").write_text("")
    (pci / "remove").write_text("")
    return str(tmp_path)


def run_agent(agent, *args):
    proc = subprocess.run([agent, *args], capture_output=True, text=True, timeout=30)
    return proc


def test_list_inventory(agent, sysroot):
    proc = run_agent(agent, "list", "--sysroot", sysroot)
    assert proc.returncode == 0, proc.stderr
    data = json.loads(proc.stdout)
    assert len(data["gpus"]) == 1
    g = data["gpus"][0]
    assert g["device_id"] == f"GPU-{12600702146510664649:016x}"
    assert g["render_minor"] == 144
    assert g["card_index"] == 8  # PCI-matched, not render-128
    assert g["pci_bdf"] == "0000:5a:00.0"
    assert g["vram_bytes"] == 309237645312
    assert g["gfx_target"] == 90500
    assert g["xgmi_peers"] == [5]
    assert g["gpu_id"] == 28206


def test_pids_attribution(agent, sysroot):
    proc = run_agent(agent, "pids", "--gpu-id", "28206", "--sysroot", sysroot)
    assert json.loads(proc.stdout)["pids"] == [1234]  # 777 has vram 0
    proc = run_agent(agent, "pids", "--sysroot", sysroot)
    assert sorted(json.loads(proc.stdout)["pids"]) == [777, 1234]


def test_drain_and_rescan_write_sysfs(agent, sysroot):
    proc = run_agent(agent, "drain", "--bdf", "0000:5a:00.0", "--sysroot", sysroot)
    assert proc.returncode == 0
    with open(os.path.join(sysroot, "sys/bus/pci/devices/0000:5a:00.0/remove")) as f:
        assert f.read() == "1"
    proc = run_agent(agent, "rescan", "--sysroot", sysroot)
    assert proc.returncode == 0
    with open(os.path.join(sysroot, "sys/bus/pci/rescan")) as f:
        assert f.read() == "1"


def test_drain_requires_bdf(agent):
    assert run_agent(agent, "drain").returncode == 2


def test_unknown_command(agent):
    assert run_agent(agent, "frobnicate").returncode == 2


@pytest.mark.gpu
def test_probe_on_real_gpu(agent):
    if not os.path.exists("/dev/kfd"):
        pytest.skip("no GPU")
    proc = run_agent(agent, "probe", "--device", "0")
    assert proc.returncode == 0, proc.stdout + proc.stderr
    data = json.loads(proc.stdout)
    assert data["ok"] and data["mfma_f32_exact"]
    assert "gfx950" in data["gcn_arch"]


def test_cxl_inventory(agent, tmp_path):
    base = tmp_path / "sys" / "bus" / "cxl" / "devices" / "mem0"
    base.mkdir(parents=True)
    (base / "serial").write_text("0xc0ffee00\n")
    ram = base / "ram"
    ram.mkdir()
    (ram / "size").write_text("0x4000000000\n")
    (base / "numa_node").write_text("2\n")
    dev = base / "device"
    dev.mkdir()
    (dev / "uevent").write_text("DRIVER=cxl_pci\nPCI_SLOT_NAME=0000:60:00.0\n")
    proc = run_agent(agent, "cxl", "--sysroot", str(tmp_path))
    assert proc.returncode == 0, proc.stderr
    data = json.loads(proc.stdout)
    assert data["memdevs"] == [
        {
            "device_id": f"CXL-{0xC0FFEE00:016x}",
            "memdev": "mem0",
            "size_bytes": 0x4000000000,
            "numa_node": 2,
            "pci_bdf": "0000:60:00.0",
        }
    ]


@pytest.mark.gpu
def test_probe_by_bdf_on_real_gpu(agent):
    if not os.path.exists("/dev/kfd"):
        pytest.skip("no GPU")
    from cro_amd.nodeops.execs import LocalNodeExec
    from cro_amd.nodeops.kfd import enumerate_gpus

    gpus = enumerate_gpus(LocalNodeExec(), "local")
    proc = run_agent(agent, "probe", "--bdf", gpus[0].pci_bdf)
    assert proc.returncode == 0, proc.stdout + proc.stderr
    data = json.loads(proc.stdout)
    assert data["ok"] and data["mfma_f32_exact"]
