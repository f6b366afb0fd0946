"""CLI tool tests against the mock node: topo snapshot, bind, sriov, ident."""
import json
import os

import pytest

from kata_xpu_device_plugin_amd.config import Config
from kata_xpu_device_plugin_amd.testing.mocknode import MockGPU, MockNode, make_mock_node
from kata_xpu_device_plugin_amd.tools.bind import Binder
from kata_xpu_device_plugin_amd.tools.ident import collect
from kata_xpu_device_plugin_amd.tools.sriov import set_numvfs, vf_bdfs
from kata_xpu_device_plugin_amd.tools.topo import build_snapshot


def test_topo_snapshot(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=8,
                          hives=[[0, 1, 2, 3], [4, 5, 6, 7]], driver="amdgpu")
    doc = build_snapshot(node.sysfs)
    assert len(doc["gpus"]) == 8
    assert len(doc["hives"]) == 2
    assert all(len(h) == 4 for h in doc["hives"])
    assert doc["xgmi_link_gbps"] == 153.0
    assert doc["gpus"][0]["gfx_target_version"] == 90500


def test_topo_snapshot_feeds_hint_loader(tmp_path):
    from kata_xpu_device_plugin_amd.topology.hive import topology_from_hint
    node = make_mock_node(str(tmp_path), n_gpus=4, driver="amdgpu")
    doc = build_snapshot(node.sysfs)
    hint = tmp_path / "hint.json"
    hint.write_text(json.dumps(doc))
    topo = topology_from_hint(str(hint))
    assert topo is not None
    assert len(topo.hive_of) == 4


def test_ident_collect(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=2)
    cfg = node.config()
    doc = collect(cfg)
    assert len(doc["functions"]) == 2
    assert doc["functions"][0]["model_name"] == "INSTINCT_MI355X"


def test_binder_status_and_rebind(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=2, driver="amdgpu",
                          kfd=False, hint=False)
    cfg = node.config()
    b = Binder(cfg)
    gpus = b.gpu_functions()
    assert len(gpus) == 2
    # mock sysfs has no writable unbind/probe files → dry-run only
    b_dry = Binder(cfg, dry_run=True)
    b_dry.rebind(gpus[0].bdf, "vfio-pci")  # must not raise
    with pytest.raises(RuntimeError, match="no such device"):
        b_dry.rebind("0000:ff:00.0", "vfio-pci")


def test_sriov_enable_validation(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=1, driver="amdgpu",
                          kfd=False, hint=False)
    pf = node.gpus[0]
    cfg = node.config()
    with pytest.raises(RuntimeError, match="does not support SR-IOV"):
        set_numvfs(cfg, pf.bdf, 4)
    d = os.path.join(node.sysfs, "bus", "pci", "devices", pf.bdf)
    open(os.path.join(d, "sriov_totalvfs"), "w").write("8\n")
    open(os.path.join(d, "sriov_numvfs"), "w").write("0\n")
    with pytest.raises(RuntimeError, match="at most 8"):
        set_numvfs(cfg, pf.bdf, 16)
    set_numvfs(cfg, pf.bdf, 4)
    assert open(os.path.join(d, "sriov_numvfs")).read() == "4"


def test_vf_bdfs_listing(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=1, driver="amdgpu",
                          kfd=False, hint=False)
    pf = node.gpus[0]
    cfg = node.config()
    d = os.path.join(node.sysfs, "bus", "pci", "devices", pf.bdf)
    for k in range(2):
        vf_bdf = f"0000:0a:02.{k}"
        node.add_gpu(MockGPU(bdf=vf_bdf, device_id=0x75B3,
                             iommu_group=str(100 + k), physfn_bdf=pf.bdf))
        os.symlink(os.path.join(node.sysfs, "bus", "pci", "devices", vf_bdf),
                   os.path.join(d, f"virtfn{k}"))
    assert vf_bdfs(cfg, pf.bdf) == ["0000:0a:02.0", "0000:0a:02.1"]


def test_resourceslice_tool(tmp_path, monkeypatch, capsys):
    import json as _json
    from kata_xpu_device_plugin_amd.tools.resourceslice import main as rs_main
    node = make_mock_node(str(tmp_path), n_gpus=4)
    cfg = node.config()
    monkeypatch.setenv("KXDP_SYSFS_ROOT", cfg.sysfs_root)
    monkeypatch.setenv("KXDP_DEV_ROOT", cfg.dev_root)
    monkeypatch.setenv("KXDP_TOPOLOGY_HINT", cfg.topology_hint_path)
    rc = rs_main(["--node", "n1", "--pool", "p1"])
    assert rc == 0
    obj = _json.loads(capsys.readouterr().out)
    assert obj["kind"] == "ResourceSlice"
    assert obj["spec"]["nodeName"] == "n1"
    assert len(obj["spec"]["devices"]) == 4


def test_all_cli_entrypoints_have_help():
    """Every tools CLI must at least render --help (import-time errors in
    any tool would break provisioning scripts)."""
    import subprocess
    import sys
    mods = ["topo", "burnin", "ident", "bind", "sriov", "validate", "partition",
            "assignments", "resourceslice", "doctor"]
    for m in mods:
        out = subprocess.run(
            [sys.executable, "-m", f"kata_xpu_device_plugin_amd.tools.{m}",
             "--help"], capture_output=True, text=True, timeout=60)
        assert out.returncode == 0, (m, out.stderr[-500:])


def _doctor_env(monkeypatch, cfg):
    monkeypatch.setenv("KXDP_SYSFS_ROOT", cfg.sysfs_root)
    monkeypatch.setenv("KXDP_DEV_ROOT", cfg.dev_root)
    monkeypatch.setenv("KXDP_CDI_DIR", cfg.cdi_dir)
    monkeypatch.setenv("KXDP_TOPOLOGY_HINT", cfg.topology_hint_path)


def test_doctor_ready_node(tmp_path, monkeypatch, capsys):
    from kata_xpu_device_plugin_amd.tools.doctor import main as doctor_main
    node = make_mock_node(str(tmp_path), n_gpus=4)
    _doctor_env(monkeypatch, node.config())
    rc = doctor_main([])
    out = capsys.readouterr().out
    assert rc == 0, out
    assert "verdict: OK" in out


def test_doctor_detects_missing_vfio_node(tmp_path, monkeypatch, capsys):
    from kata_xpu_device_plugin_amd.tools.doctor import main as doctor_main
    node = make_mock_node(str(tmp_path), n_gpus=2)
    node.remove_vfio_node("71")
    _doctor_env(monkeypatch, node.config())
    rc = doctor_main(["--json"])
    import json as _json
    doc = _json.loads(capsys.readouterr().out)
    assert rc == 1
    assert any("71" in p for p in doc["problems"])


def test_doctor_preprovisioning_note(tmp_path, monkeypatch, capsys):
    from kata_xpu_device_plugin_amd.tools.doctor import main as doctor_main
    node = make_mock_node(str(tmp_path), n_gpus=2, driver="amdgpu",
                          kfd=True, hint=False)
    _doctor_env(monkeypatch, node.config())
    rc = doctor_main([])
    out = capsys.readouterr().out
    assert rc == 0, out
    assert "pre-provisioning" in out


def test_doctor_no_gpus(tmp_path, monkeypatch, capsys):
    from kata_xpu_device_plugin_amd.tools.doctor import main as doctor_main
    node = MockNode(root=str(tmp_path))  # empty node
    os.makedirs(os.path.join(node.sysfs, "bus", "pci", "devices"), exist_ok=True)
    _doctor_env(monkeypatch, node.config())
    rc = doctor_main([])
    assert rc == 2


def test_daemonset_manifest_sane():
    """Guard the deploy manifest: parses, mounts every path the daemon
    needs, and the container command points at this package."""
    import yaml as _yaml
    path = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "deploy",
        "kata-xpu-device-plugin-amd.yaml")
    doc = _yaml.safe_load(open(path))
    assert doc["kind"] == "DaemonSet"
    spec = doc["spec"]["template"]["spec"]
    ctr = spec["containers"][0]
    assert "kata_xpu_device_plugin_amd" in " ".join(ctr["command"])
    mounts = {m["mountPath"] for m in ctr["volumeMounts"]}
    for required in ("/var/lib/kubelet/device-plugins", "/dev/vfio",
                     "/var/run/cdi", "/sys",
                     "/var/lib/kubelet/pod-resources",
                     "/var/lib/kubelet/plugins_registry"):
        assert required in mounts, required
    vols = {v["name"] for v in spec["volumes"]}
    assert {m["name"] for m in ctr["volumeMounts"]} <= vols
    assert ctr["livenessProbe"]["httpGet"]["path"] == "/metrics"


# --- partition tool (MI355X compute/memory partitioning) -------------------

def _part_node(tmp_path):
    from kata_xpu_device_plugin_amd.testing.mocknode import MockNode, MockGPU
    node = MockNode(root=str(tmp_path))
    node.add_gpu(MockGPU(bdf="0000:0a:00.0", driver="amdgpu", iommu_group="70",
                         compute_partition="SPX",
                         compute_available="SPX, DPX, QPX, CPX",
                         memory_partition="NPS1",
                         memory_available="NPS1, NPS4"))
    node.add_gpu(MockGPU(bdf="0000:12:00.0", iommu_group="71"))  # vfio, no files
    return node


def test_partition_show(tmp_path, capsys):
    import json as _json
    from kata_xpu_device_plugin_amd.tools.partition import list_states, main
    node = _part_node(tmp_path)
    cfg = node.config()
    states = {s.bdf: s for s in list_states(cfg)}
    assert states["0000:0a:00.0"].compute_current == "SPX"
    assert states["0000:0a:00.0"].compute_available == ["SPX", "DPX", "QPX", "CPX"]
    assert states["0000:0a:00.0"].memory_current == "NPS1"
    assert not states["0000:12:00.0"].supported   # vfio-bound: no files
    rc = main(["--sysfs-root", cfg.sysfs_root, "--dev-root", cfg.dev_root,
               "show"])
    assert rc == 0
    doc = _json.loads(capsys.readouterr().out)
    assert {d["bdf"] for d in doc} == {"0000:0a:00.0", "0000:12:00.0"}


def test_partition_set_paths(tmp_path):
    import pytest as _pytest
    from kata_xpu_device_plugin_amd.tools.partition import (
        read_partition_state, set_partition)
    node = _part_node(tmp_path)
    cfg = node.config()
    # happy path: amdgpu-bound, mode in available set
    set_partition(cfg, "0000:0a:00.0", compute="CPX")
    assert read_partition_state(cfg, "0000:0a:00.0").compute_current == "CPX"
    set_partition(cfg, "0000:0a:00.0", memory="NPS4")
    assert read_partition_state(cfg, "0000:0a:00.0").memory_current == "NPS4"
    # refusals: vfio-bound GPU, unsupported mode, unknown bdf
    with _pytest.raises(ValueError, match="amdgpu"):
        set_partition(cfg, "0000:12:00.0", compute="CPX")
    with _pytest.raises(ValueError, match="available"):
        set_partition(cfg, "0000:0a:00.0", compute="TPX")
    with _pytest.raises(ValueError, match="not an AMD"):
        set_partition(cfg, "0000:ff:00.0", compute="CPX")
    # dry run leaves state untouched
    set_partition(cfg, "0000:0a:00.0", compute="SPX", dry_run=True)
    assert read_partition_state(cfg, "0000:0a:00.0").compute_current == "CPX"
