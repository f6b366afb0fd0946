"""Tests against a captured REAL 8×MI355X node layout.

tests/data/mi355x_live_node_ident.json was captured on live MI355X
hardware (tools/ident.py): 8× device 0x75a3 class 0x120000 ("Processing
accelerator"), each amdgpu-bound in its own IOMMU group behind an AMD
0x1501 PCIe bridge; KFD shows gfx_target_version 90500 and 7 xGMI links
per GPU. These tests pin the discovery/naming behavior to that reality.
"""
import json
import os

from kata_xpu_device_plugin_amd.discovery import scan_node
from kata_xpu_device_plugin_amd.discovery.naming import device_model_name
from kata_xpu_device_plugin_amd.discovery.sysfs import PCIFunction
from kata_xpu_device_plugin_amd.testing.mocknode import MockGPU, MockNode

FIXTURE = os.path.join(os.path.dirname(__file__), "data",
                       "mi355x_live_node_ident.json")


def _load():
    with open(FIXTURE) as f:
        return json.load(f)


def test_fixture_shape():
    doc = _load()
    gpus = [f for f in doc["functions"] if f["device"] == "75a3"]
    bridges = [f for f in doc["functions"] if f["device"] == "1501"]
    assert len(gpus) == 8 and len(bridges) == 8
    assert all(f["class"] == "120000" for f in gpus)
    assert all(f["class"] == "060400" for f in bridges)
    assert doc["amdsmi"][0]["device_id"] == "0x75a3"
    assert doc["amdsmi"][0]["target_graphics_version"] == "gfx950"
    assert doc["amdsmi"][0]["num_compute_units"] == "256"


def test_accelerator_class_is_gpu():
    fn = PCIFunction(bdf="0000:0a:00.0", vendor=0x1002, device=0x75A3,
                     class_code=0x120000, driver="vfio-pci", iommu_group="94")
    assert fn.is_gpu
    bridge = PCIFunction(bdf="0000:09:00.0", vendor=0x1002, device=0x1501,
                         class_code=0x060400, driver="pcieport", iommu_group="93")
    assert not bridge.is_gpu


def test_live_layout_replayed_through_discovery(tmp_path):
    """Rebuild the live node's layout (vfio-bound variant) in a mock tree:
    bridges must not become schedulable devices; all 8 GPUs must."""
    doc = _load()
    node = MockNode(root=str(tmp_path))
    for f in doc["functions"]:
        node.add_gpu(MockGPU(
            bdf=f["bdf"],
            device_id=int(f["device"], 16),
            iommu_group=f["iommu_group"],
            driver="vfio-pci",  # as after vfio binding
            numa_node=f["numa"],
            class_code=int(f["class"], 16),
            hive_id=1,
        ))
    inv = scan_node(node.config())
    assert len(inv.devices) == 8
    assert set(inv.by_model) == {0x75A3}
    # bridge groups must not be schedulable
    bridge_groups = {f["iommu_group"] for f in doc["functions"]
                     if f["device"] == "1501"}
    assert not bridge_groups & set(inv.devices)


def test_naming_matches_live_amdsmi():
    doc = _load()
    assert device_model_name(0x75A3) == "INSTINCT_MI355X"
    # amd-smi market name agrees with the curated table
    assert "MI355" in doc["amdsmi"][0]["market_name"]
