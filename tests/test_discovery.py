"""Discovery-layer tests (reference behavior: device_plugin.go:126-180)."""
import os

import pytest

from kata_xpu_device_plugin_amd.config import Config
from kata_xpu_device_plugin_amd.discovery import scan_node
from kata_xpu_device_plugin_amd.discovery.sysfs import scan_functions
from kata_xpu_device_plugin_amd.testing.mocknode import (
    MI355X_DEVICE_ID,
    MockGPU,
    make_mock_node,
)


def test_scan_finds_all_gpus(mock_node, mock_cfg):
    inv = scan_node(mock_cfg)
    assert len(inv.devices) == 8
    assert inv.device_ids() == [str(70 + i) for i in range(8)]
    assert set(inv.by_model) == {MI355X_DEVICE_ID}
    assert len(inv.by_model[MI355X_DEVICE_ID]) == 8


def test_device_attributes(mock_node, mock_cfg):
    inv = scan_node(mock_cfg)
    dev = inv.devices["70"]
    assert dev.primary.bdf == "0000:0a:00.0"
    assert dev.primary.vendor == 0x1002
    assert dev.primary.device == MI355X_DEVICE_ID
    assert dev.primary.driver == "vfio-pci"
    assert dev.primary.is_gpu
    assert not dev.is_vf
    assert dev.vfio_node == "vfio/70"
    assert dev.numa_node == 0
    assert inv.devices["74"].numa_node == 1


def test_non_vfio_devices_excluded(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=4, driver="amdgpu", kfd=False, hint=False)
    cfg = node.config()
    inv = scan_node(cfg)
    assert inv.devices == {}
    # ...but the functions are still visible for diagnostics/health.
    assert len(inv.all_functions) == 4


def test_wrong_vendor_excluded(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=2, kfd=False, hint=False)
    # add a non-AMD GPU
    d = os.path.join(node.sysfs, "bus", "pci", "devices", "0000:ff:00.0")
    os.makedirs(d)
    open(os.path.join(d, "vendor"), "w").write("0x10de\n")
    open(os.path.join(d, "device"), "w").write("0x2330\n")
    open(os.path.join(d, "class"), "w").write("0x030200\n")
    inv = scan_node(node.config())
    assert len(inv.devices) == 2
    assert all(f.vendor == 0x1002 for f in inv.all_functions)


def test_multifunction_group_single_device(tmp_path):
    """A GPU+audio pair in one IOMMU group must be ONE schedulable device
    (the reference emitted one CDI device per function, duplicating the
    /dev/vfio node — device_plugin.go:59-77; we do not)."""
    node = make_mock_node(str(tmp_path), n_gpus=2, with_audio_fn=True, kfd=False, hint=False)
    inv = scan_node(node.config())
    assert len(inv.devices) == 2
    dev = inv.devices["70"]
    assert len(dev.functions) == 2
    assert dev.primary.is_gpu  # GPU sorted first
    assert dev.functions[1].class_code >> 16 == 0x04


def test_group_without_gpu_not_schedulable(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=1, kfd=False, hint=False)
    # lone audio function in its own group
    node.add_gpu(
        MockGPU(bdf="0000:50:00.1", device_id=0xAB30, iommu_group="99",
                class_code=0x040300),
    )
    inv = scan_node(node.config())
    assert "99" not in inv.devices
    assert len(inv.devices) == 1


def test_device_allowlist(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=2, kfd=False, hint=False)
    node.add_gpu(MockGPU(bdf="0000:60:00.0", device_id=0x74A1, iommu_group="90"))
    cfg = node.config(device_allowlist=(MI355X_DEVICE_ID,))
    inv = scan_node(cfg)
    assert len(inv.devices) == 2
    cfg2 = node.config()
    inv2 = scan_node(cfg2)
    assert len(inv2.devices) == 3
    assert set(inv2.by_model) == {MI355X_DEVICE_ID, 0x74A1}


def test_sriov_vf_discovery(tmp_path):
    """SR-IOV VFs (MxGPU) as schedulable partitioned devices
    (BASELINE.json config #5 — absent from the reference)."""
    node = make_mock_node(str(tmp_path), n_gpus=1, driver="amdgpu", kfd=False, hint=False)
    pf = node.gpus[0]
    pf_dir = os.path.join(node.sysfs, "bus", "pci", "devices", pf.bdf)
    open(os.path.join(pf_dir, "sriov_totalvfs"), "w").write("8\n")
    open(os.path.join(pf_dir, "sriov_numvfs"), "w").write("2\n")
    for k in range(2):
        node.add_gpu(
            MockGPU(
                bdf=f"0000:0a:02.{k}", device_id=0x75B3, iommu_group=str(100 + k),
                physfn_bdf=pf.bdf,
            )
        )
    inv = scan_node(node.config())
    assert len(inv.devices) == 2
    for gid in ("100", "101"):
        assert inv.devices[gid].is_vf
        assert inv.devices[gid].model_device_id == 0x75B3


def test_malformed_sysfs_entries_skipped(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=1, kfd=False, hint=False)
    d = os.path.join(node.sysfs, "bus", "pci", "devices", "0000:bb:00.0")
    os.makedirs(d)
    open(os.path.join(d, "vendor"), "w").write("garbage\n")
    inv = scan_node(node.config())
    assert len(inv.devices) == 1


def test_scan_functions_python_parity(mock_node, mock_cfg):
    """Pure-python and configured path agree (native ext tested separately)."""
    mock_cfg.native = "off"
    py = {f.bdf: f for f in scan_functions(mock_cfg)}
    assert len(py) == 8
    assert all(f.iommu_group is not None for f in py.values())


def test_scan_wall_time_recorded(mock_cfg):
    inv = scan_node(mock_cfg)
    assert inv.scan_wall_s > 0
