"""CXL.mem node path: enumeration from a fixture sysfs tree, CDI emission,
drain, and a full cxlmemory lifecycle through the manager with the
per-type CompositeNodeOps dispatch."""

import json

import pytest

from cro_amd.api.v1alpha1.types import ComposabilityRequest, Node
from cro_amd.controllers import build_manager
from cro_amd.fabric.adapter import Adapter
from cro_amd.fabric.mock import MockFabric
from cro_amd.nodeops.amdgpu import MockNodeOps
from cro_amd.nodeops.composite import CompositeNodeOps
from cro_amd.nodeops.cxl import CXL_DEVICES, CxlNodeOps, enumerate_cxl_memdevs
from cro_amd.nodeops.execs import MockNodeExec
from tests.conftest import make_request

NODE = "node0"


def cxl_fixture(ex: MockNodeExec, n: int = 2):
    ids = []
    for i in range(n):
        base = f"{CXL_DEVICES}/mem{i}"
        serial = 0xC0FFEE00 + i
        ex.set_file(NODE, f"{base}/serial", f"0x{serial:x}\n")
        ex.set_file(NODE, f"{base}/ram/size", "0x4000000000\n")  # 256 GiB
        ex.set_file(NODE, f"{base}/numa_node", f"{2 + i}\n")
        ex.set_file(NODE, f"{base}/device/uevent",
                    f"DRIVER=cxl_pci\nPCI_SLOT_NAME=0000:{0x60 + i:02x}:00.0\n")
        ex.set_file(NODE, f"{base}/dax/dax{i}.0/uevent", "")
        ids.append(f"CXL-{serial:016x}")
    ex.set_file(NODE, "/sys/module/amdgpu/version", "6.x")
    return ids


def test_enumerate_cxl_memdevs():
    ex = MockNodeExec()
    ids = cxl_fixture(ex, 2)
    devs = enumerate_cxl_memdevs(ex, NODE)
    assert [d.device_id for d in devs] == ids
    d0 = devs[0]
    assert d0.size_bytes == 0x4000000000
    assert d0.numa_node == 2
    assert d0.pci_bdf == "0000:60:00.0"
    assert d0.dax_path == "/dev/dax0.0"


def test_enumerate_empty_without_bus():
    assert enumerate_cxl_memdevs(MockNodeExec(), NODE) == []


def test_cxl_cdi_spec():
    ex = MockNodeExec()
    ids = cxl_fixture(ex, 1)
    ops = CxlNodeOps(ex, cdi_dir="/etc/cdi")
    cdi_id = ops.write_cdi(NODE, ids[0])
    assert cdi_id == f"amd.com/cxlmem={ids[0]}"
    spec = json.loads(ex.files[(NODE, "/etc/cdi/amd.com-cxlmem-cro.json")])
    assert spec["kind"] == "amd.com/cxlmem"
    assert spec["containerEdits"]["deviceNodes"] == []  # no /dev/kfd here
    dev = spec["devices"][0]
    assert dev["containerEdits"]["deviceNodes"] == [{"path": "/dev/dax0.0"}]
    assert dev["annotations"]["cro.amd.com/size-bytes"] == str(0x4000000000)
    ops.remove_cdi(NODE, ids[0])
    assert ops.cdi.devices(NODE) == []


def test_cxl_drain_writes_pci_remove():
    ex = MockNodeExec()
    ids = cxl_fixture(ex, 1)
    ops = CxlNodeOps(ex, destructive=True)
    ops.drain(NODE, ids[0])
    assert ex.files[(NODE, "/sys/bus/pci/devices/0000:60:00.0/remove")] == "1"


def test_cxl_simulated_lifecycle():
    ex = MockNodeExec()
    ids = cxl_fixture(ex, 1)
    ops = CxlNodeOps(ex, destructive=False, initially_detached=ids)
    assert not ops.is_visible(NODE, ids[0])
    ops.simulate_compose(NODE, ids[0])
    assert ops.is_visible(NODE, ids[0])
    ops.drain(NODE, ids[0])
    assert not ops.is_visible(NODE, ids[0])


@pytest.fixture
def mixed_stack():
    """Manager with a per-type CompositeNodeOps: mock GPUs + real-shaped
    CXL memdevs behind a fixture sysfs tree."""
    ex = MockNodeExec()
    cxl_ids = cxl_fixture(ex, 2)
    cxl_ops = CxlNodeOps(ex, destructive=False, initially_detached=cxl_ids)

    fabric = MockFabric(models={"mi355x": 4})
    # seed the fabric pool with the CXL devices too
    from cro_amd.fabric.mock import _PoolDevice

    for cid in cxl_ids:
        fabric._pool[cid] = _PoolDevice(
            device_id=cid, cdi_device_id=f"amd.com/cxlmem={cid}", model="cxl-256g"
        )

    mgr = build_manager(Adapter("DRA", fabric), None)
    gpu_ops = MockNodeOps(client=mgr.client)
    composite = CompositeNodeOps({"gpu": gpu_ops, "cxlmemory": cxl_ops})
    mgr.resource_reconciler.node_ops = composite

    orig_add = fabric.add_resource

    def add_resource(resource):
        did, cdi = orig_add(resource)
        if resource.spec.type == "cxlmemory":
            cxl_ops.simulate_compose(resource.spec.target_node, did)
        else:
            gpu_ops.fabric_composed(resource.spec.target_node, did)
        return did, cdi

    fabric.add_resource = add_resource

    node = Node()
    node.metadata.name = NODE
    mgr.client.create(node)
    gpu_ops.set_driver(NODE, True)
    mgr.start()

    class Stack:
        pass

    s = Stack()
    s.mgr, s.fabric, s.cxl_ops, s.gpu_ops, s.cxl_ids = mgr, fabric, cxl_ops, gpu_ops, cxl_ids
    yield s
    mgr.stop()


def test_cxlmemory_lifecycle_through_manager(mixed_stack):
    mgr = mixed_stack.mgr
    req = make_request("mem1", rtype="cxlmemory", model="cxl-256g", size=1,
                       target_node=NODE)
    mgr.client.create(req)
    assert mgr.wait_for(
        lambda: (r := mgr.client.try_get(ComposabilityRequest, "mem1")) is not None
        and r.status.state == "Running",
        timeout=15,
    ), mgr.client.get(ComposabilityRequest, "mem1").status
    r = mgr.client.get(ComposabilityRequest, "mem1")
    entry = next(iter(r.status.resources.values()))
    assert entry.device_id.startswith("CXL-")
    assert mixed_stack.cxl_ops.cdi.devices(NODE) == [entry.device_id]

    # a GPU request coexists through the same manager (distinct type+model
    # per node keeps admission happy)
    mgr.client.create(make_request("gpu1", size=1, target_node=NODE))
    assert mgr.wait_for(
        lambda: (g := mgr.client.try_get(ComposabilityRequest, "gpu1")) is not None
        and g.status.state == "Running",
        timeout=15,
    )

    for name in ("mem1", "gpu1"):
        mgr.client.delete(ComposabilityRequest, name)
    assert mgr.wait_for(
        lambda: mgr.client.list(ComposabilityRequest) == [], timeout=15
    )
    assert mixed_stack.cxl_ops.cdi.devices(NODE) == []
    assert mixed_stack.fabric.attached_to(NODE) == []


def test_dax_load_attribution_blocks_detach():
    """check_no_loads scans /proc/<pid>/maps for the memdev's dax path —
    the CXL analog of per-GPU KFD vram attribution."""
    from cro_amd.nodeops.amdgpu import GPULoadsPresent

    ex = MockNodeExec()
    ids = cxl_fixture(ex, 2)
    ops = CxlNodeOps(ex)
    # pid 100 maps mem0's dax; pid 200 maps something else; pid 300 unreadable
    ex.set_file(NODE, "/proc/100/maps",
                "7f00-7f10 rw-s 0 00:0e 42 /dev/dax0.0\n")
    ex.set_file(NODE, "/proc/200/maps",
                "7f00-7f10 rw-s 0 00:0e 43 /dev/shm/x\n")
    ex.set_file(NODE, "/proc/self/status", "")  # non-numeric entry ignored

    assert ops.dax_holders(NODE, ids[0]) == [100]
    assert ops.dax_holders(NODE, ids[1]) == []
    with pytest.raises(GPULoadsPresent, match=r"\[100\]"):
        ops.check_no_loads(NODE, ids[0])
    ops.check_no_loads(NODE, ids[1])  # other memdev unaffected
    with pytest.raises(GPULoadsPresent):
        ops.check_no_loads(NODE)  # whole-node form sees mem0's holder

    # holder exits → detach unblocks
    ex.files.pop((NODE, "/proc/100/maps"))
    ops.check_no_loads(NODE)
