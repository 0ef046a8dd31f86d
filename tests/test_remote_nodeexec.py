"""Remote node access: the agent HTTP API driven through RemoteNodeExec —
the whole amdgpu node path (KFD enumeration, CDI writes, drain) running
against a node reached only over HTTP."""

import json
import os

import pytest
from fastapi.testclient import TestClient

from cro_amd.nodeops.amdgpu import AmdNodeOps
from cro_amd.nodeops.execs import ExecError, LocalNodeExec, MockNodeExec
from cro_amd.nodeops.kfd import enumerate_gpus
from cro_amd.nodeops.remote_exec import RemoteNodeExec
from cro_amd.server.agent_api import build_agent_app
from tests.test_nodeops import kfd_fixture

NODE = "node0"


@pytest.fixture
def remote_exec():
    """Agent serving a MockNodeExec-backed node, reached over ASGI HTTP."""
    backend = MockNodeExec()
    ids = kfd_fixture(backend, 2, node=NODE)
    app = build_agent_app(backend, node_name=NODE, allow_insecure=True)
    http = TestClient(app)

    class TestClientTransport:
        pass

    remote = RemoteNodeExec({NODE: "http://agent"})
    remote._http = http  # TestClient implements the httpx.Client surface
    return remote, backend, ids


def test_remote_file_and_dir_ops(remote_exec):
    remote, backend, _ = remote_exec
    assert "simd_count 1024" in remote.read_file(
        NODE, "/sys/class/kfd/kfd/topology/nodes/1/properties"
    )
    with pytest.raises(FileNotFoundError):
        remote.read_file(NODE, "/nope")
    remote.write_file(NODE, "/sys/bus/pci/rescan", "1")
    assert backend.files[(NODE, "/sys/bus/pci/rescan")] == "1"
    entries = remote.list_dir(NODE, "/sys/class/kfd/kfd/topology/nodes")
    assert set(entries) == {"0", "1", "2"}
    assert remote.path_exists(NODE, "/sys/module/amdgpu")
    assert not remote.path_exists(NODE, "/sys/module/nvidia")


def test_remote_run(remote_exec):
    remote, backend, _ = remote_exec
    backend.set_command(("modprobe", "-r", "amdgpu"), (0, "ok", ""))
    rc, out, err = remote.run(NODE, ["modprobe", "-r", "amdgpu"])
    assert (rc, out) == (0, "ok")
    with pytest.raises(ExecError):
        remote.run(NODE, ["not-canned"])


def test_kfd_enumeration_over_http(remote_exec):
    remote, _, ids = remote_exec
    gpus = enumerate_gpus(remote, NODE)
    assert [g.device_id for g in gpus] == ids
    assert gpus[0].xgmi_peers == [2]


def test_full_node_ops_over_http(remote_exec):
    """AmdNodeOps (driver gate, visibility, CDI write, drain) unchanged on
    a remote node."""
    remote, backend, ids = remote_exec
    ops = AmdNodeOps(remote, cdi_dir="/etc/cdi", destructive=True)
    ops.enum_cache_ttl = 0.0  # deterministic across mutations in this test
    ops.ensure_driver(NODE)
    assert ops.is_visible(NODE, ids[0])
    cdi_id = ops.write_cdi(NODE, ids[0])
    assert cdi_id == f"amd.com/gpu={ids[0]}"
    spec = json.loads(backend.files[(NODE, "/etc/cdi/amd.com-gpu-cro.json")])
    assert spec["devices"][0]["name"] == ids[0]
    ops.drain(NODE, ids[0])  # not last device → synchronous sysfs remove
    assert backend.files[(NODE, "/sys/bus/pci/devices/0000:03:00.0/remove")] == "1"


def test_resolver_miss():
    remote = RemoteNodeExec({})
    with pytest.raises(ExecError, match="no agent endpoint"):
        remote.read_file("ghost", "/etc/hostname")


def test_agent_api_against_real_fs(tmp_path):
    """LocalNodeExec with a sysroot behind the agent API: file IO hits disk."""
    os.makedirs(tmp_path / "sys" / "module" / "amdgpu")
    app = build_agent_app(LocalNodeExec(sysroot=str(tmp_path)), node_name="n", allow_insecure=True)
    http = TestClient(app)
    remote = RemoteNodeExec({"n": "http://agent"})
    remote._http = http
    remote.write_file("n", "/etc/cdi/spec.json", "{}")
    assert (tmp_path / "etc" / "cdi" / "spec.json").read_text() == "{}"
    assert remote.path_exists("n", "/sys/module/amdgpu")


def test_probe_via_exec_parses_croagent_output(remote_exec):
    """Exec-based probe hook: croagent JSON over the agent API."""
    from cro_amd.nodeops.kfd import enumerate_gpus
    from cro_amd.nodeops.probe import make_exec_probe_fn

    remote, backend, ids = remote_exec
    backend.set_command(
        ("croagent", "probe", "--bdf", "0000:03:00.0"),
        (0, '{"ok":true,"rc":0,"mfma_f32_exact":true,"hbm_gbps":4500.0,'
            '"bf16_tflops":2000.0,"vram_total":309220868096,"vram_free":1,'
            '"gcn_arch":"gfx950","msg":"ok"}', ""),
    )
    gpus = enumerate_gpus(remote, NODE)
    probe = make_exec_probe_fn(remote, NODE)
    result = probe(gpus[0])
    assert result["ok"] and result["mfma_f32_exact"]
    assert result["gcn_arch"] == "gfx950"


def test_probe_via_exec_failure_shapes(remote_exec):
    from cro_amd.nodeops.kfd import GPUDevice
    from cro_amd.nodeops.probe import probe_via_exec

    remote, backend, _ = remote_exec
    gpu = GPUDevice(kfd_node=1, gpu_id=1, device_id="GPU-x", unique_id=1,
                    render_minor=128, pci_bdf="0000:99:00.0")
    backend.set_command(("croagent", "probe", "--bdf", "0000:99:00.0"),
                        (1, "", "no HIP device with bdf 0000:99:00.0"))
    result = probe_via_exec(remote, NODE, gpu)
    assert not result["ok"] and "no HIP device" in result["msg"]
    backend.set_command(("croagent", "probe", "--bdf", "0000:99:00.0"),
                        (0, "garbage not json", ""))
    result = probe_via_exec(remote, NODE, gpu)
    assert not result["ok"] and "unparseable" in result["msg"]


def test_agent_token_auth():
    """CRO_AGENT_TOKEN gates every /agent route (the standalone analog of
    the RBAC around the reference's pods/exec path)."""
    backend = MockNodeExec()
    kfd_fixture(backend, 1, node=NODE)
    app = build_agent_app(backend, node_name=NODE, token="s3cret")

    anon = RemoteNodeExec({NODE: "http://agent"})
    anon._http = TestClient(app)
    with pytest.raises(Exception):  # 401 surfaces as HTTPStatusError
        anon.path_exists(NODE, "/sys/module/amdgpu")

    authed = RemoteNodeExec({NODE: "http://agent"})
    authed._http = TestClient(app, headers={"Authorization": "Bearer s3cret"})
    assert authed.path_exists(NODE, "/sys/module/amdgpu")

    wrong = RemoteNodeExec({NODE: "http://agent"})
    wrong._http = TestClient(app, headers={"Authorization": "Bearer nope"})
    with pytest.raises(Exception):
        wrong.read_file(NODE, "/sys/class/kfd/kfd/topology/nodes/1/properties")


def test_agent_app_fails_closed_without_token(monkeypatch):
    """No token and no explicit insecure opt-in → the /agent surface must
    refuse to build (it executes binaries and writes files)."""
    from cro_amd.server.agent_api import AgentAuthError

    monkeypatch.delenv("CRO_AGENT_TOKEN", raising=False)
    backend = MockNodeExec()
    with pytest.raises(AgentAuthError):
        build_agent_app(backend, node_name=NODE)
