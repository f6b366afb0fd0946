"""CDI layer tests, incl. the Kata contract golden file.

Reference behavior being matched: generateCDISpec
(pkg/device_plugin/device_plugin.go:55-80) + cdi/spec.go writer.
"""
import os

import pytest
import yaml

from kata_xpu_device_plugin_amd.cdi import (
    build_spec,
    parse_qualified_name,
    qualified_name,
    write_spec,
)
from kata_xpu_device_plugin_amd.cdi.spec import read_spec, spec_path
from kata_xpu_device_plugin_amd.discovery import scan_node
from kata_xpu_device_plugin_amd.testing.mocknode import make_mock_node


def test_qualified_name_roundtrip():
    qn = qualified_name("amd.com/gpu", "70")
    assert qn == "amd.com/gpu=70"
    assert parse_qualified_name(qn) == ("amd.com/gpu", "70")
    with pytest.raises(ValueError):
        parse_qualified_name("no-equals")
    with pytest.raises(ValueError):
        parse_qualified_name("badkind=70")


def test_build_spec_shape(mock_cfg):
    inv = scan_node(mock_cfg)
    spec = build_spec(inv, "amd.com/gpu", mock_cfg.dev_root)
    assert spec.cdi_version == "0.8.0"
    assert spec.kind == "amd.com/gpu"
    assert len(spec.devices) == 8
    d0 = spec.devices[0]
    assert d0.name == "70"
    # Kata contract (reference device_plugin.go:62-68)
    assert d0.annotations["attach-pci"] == "true"
    assert d0.annotations["bdf"] == "0000:0a:00.0"
    assert d0.annotations["cdi.k8s.io/vfio70"] == "amd.com/gpu=70"
    assert d0.device_nodes == [os.path.join(mock_cfg.dev_root, "vfio", "70")]


def test_multifunction_group_one_device_one_node(tmp_path):
    """Reference quirk NOT reproduced: per-function devices duplicated the
    /dev/vfio node (device_plugin.go:59-77). Here: one device per group,
    bdf annotation carries both functions."""
    node = make_mock_node(str(tmp_path), n_gpus=1, with_audio_fn=True, kfd=False, hint=False)
    cfg = node.config()
    spec = build_spec(scan_node(cfg), "amd.com/gpu", cfg.dev_root)
    assert len(spec.devices) == 1
    d = spec.devices[0]
    assert d.annotations["bdf"] == "0000:0a:00.0,0000:0a:00.1"
    assert len(d.device_nodes) == 1


def test_write_yaml_golden(mock_cfg, tmp_path):
    inv = scan_node(mock_cfg)
    spec = build_spec(inv, "amd.com/gpu", "/dev")
    path = write_spec(spec, str(tmp_path), "kxdp-vfio", "yaml")
    assert path.endswith("kxdp-vfio.yaml")
    obj = yaml.safe_load(open(path))
    assert obj["cdiVersion"] == "0.8.0"
    assert obj["kind"] == "amd.com/gpu"
    assert [d["name"] for d in obj["devices"]] == [str(70 + i) for i in range(8)]
    dev = obj["devices"][3]
    assert dev["annotations"]["attach-pci"] == "true"
    assert dev["annotations"]["bdf"] == "0000:22:00.0"
    assert dev["containerEdits"]["deviceNodes"] == [
        {"path": "/dev/vfio/73", "permissions": "rw"}
    ]
    # file mode is world-readable (runtime reads it)
    assert oct(os.stat(path).st_mode & 0o777) == "0o644"


def test_write_json_and_format_switch_removes_stale(mock_cfg, tmp_path):
    inv = scan_node(mock_cfg)
    spec = build_spec(inv, "amd.com/gpu", "/dev")
    ypath = write_spec(spec, str(tmp_path), "kxdp-vfio", "yaml")
    jpath = write_spec(spec, str(tmp_path), "kxdp-vfio", "json")
    assert os.path.exists(jpath)
    assert not os.path.exists(ypath), "stale yaml spec must be removed"
    back = read_spec(jpath)
    assert back.device_names() == spec.device_names()
    assert back.devices[0].annotations == spec.devices[0].annotations


def test_roundtrip_yaml(mock_cfg, tmp_path):
    inv = scan_node(mock_cfg)
    spec = build_spec(inv, "amd.com/gpu", mock_cfg.dev_root)
    path = write_spec(spec, str(tmp_path), "s", "yaml")
    back = read_spec(path)
    assert back.kind == spec.kind
    assert back.device_names() == spec.device_names()
    assert back.devices[5].device_nodes == spec.devices[5].device_nodes


def test_validation_rejects_duplicates_and_empty(mock_cfg):
    inv = scan_node(mock_cfg)
    spec = build_spec(inv, "amd.com/gpu", mock_cfg.dev_root)
    spec.devices.append(spec.devices[0])
    with pytest.raises(ValueError, match="duplicate"):
        spec.validate()
    spec.devices.pop()
    spec.devices[0].device_nodes = []
    with pytest.raises(ValueError, match="no device nodes"):
        spec.validate()


def test_spec_path_helper(tmp_path):
    assert spec_path(str(tmp_path), "n", "yaml").endswith("n.yaml")
    assert spec_path(str(tmp_path), "n", "json").endswith("n.json")


# --- CDI 0.8.0 schema validation -----------------------

def _schema_check(path):
    from kata_xpu_device_plugin_amd.cdi.schema import validate_spec_file
    problems = validate_spec_file(path)
    assert problems == [], problems


def test_schema_valid_pf_node(mock_cfg, tmp_path):
    """8-PF node: written spec (both formats) passes the vendored CDI
    0.8.0 schema."""
    inv = scan_node(mock_cfg)
    spec = build_spec(inv, "amd.com/gpu", mock_cfg.dev_root)
    for fmt in ("yaml", "json"):
        _schema_check(write_spec(spec, str(tmp_path), "s", fmt))


def test_schema_valid_multifunction_node(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=2, with_audio_fn=True,
                          kfd=False, hint=False)
    cfg = node.config()
    spec = build_spec(scan_node(cfg), "amd.com/gpu", cfg.dev_root)
    _schema_check(write_spec(spec, str(tmp_path), "s", "yaml"))


def test_schema_valid_vf_node(tmp_path):
    from kata_xpu_device_plugin_amd.testing.mocknode import MockGPU
    node = make_mock_node(str(tmp_path), n_gpus=0, kfd=False, hint=False)
    node.add_gpu(MockGPU(bdf="0000:60:00.0", driver="amdgpu",
                         iommu_group="50", sriov_totalvfs=4))
    for k in range(4):
        node.add_gpu(MockGPU(bdf=f"0000:60:02.{k}", device_id=0x75B3,
                             iommu_group=str(110 + k),
                             physfn_bdf="0000:60:00.0"))
    cfg = node.config()
    spec = build_spec(scan_node(cfg), "amd.com/gpu", cfg.dev_root)
    assert len(spec.devices) == 4   # VFs only; amdgpu-bound PF not schedulable
    _schema_check(write_spec(spec, str(tmp_path), "s", "yaml"))


def test_schema_rejects_contract_violations():
    """The validator actually bites: malformed specs are caught before
    they could reach /var/run/cdi."""
    from kata_xpu_device_plugin_amd.cdi.schema import validate_spec_obj
    good = {
        "cdiVersion": "0.8.0",
        "kind": "amd.com/gpu",
        "devices": [{
            "name": "70",
            "annotations": {"attach-pci": "true", "bdf": "0000:0a:00.0"},
            "containerEdits": {
                "deviceNodes": [{"path": "/dev/vfio/70", "permissions": "rw"}]
            },
        }],
    }
    assert validate_spec_obj(good) == []

    import copy
    bad_version = copy.deepcopy(good)
    bad_version["cdiVersion"] = "9.9.9"
    assert any("cdiVersion" in e for e in validate_spec_obj(bad_version))

    no_edits = copy.deepcopy(good)
    del no_edits["devices"][0]["containerEdits"]
    assert any("containerEdits" in e for e in validate_spec_obj(no_edits))

    empty_devices = copy.deepcopy(good)
    empty_devices["devices"] = []
    assert any("items" in e for e in validate_spec_obj(empty_devices))

    non_string_ann = copy.deepcopy(good)
    non_string_ann["devices"][0]["annotations"]["attach-pci"] = True
    assert any("string" in e for e in validate_spec_obj(non_string_ann))

    rel_node = copy.deepcopy(good)
    rel_node["devices"][0]["containerEdits"]["deviceNodes"][0]["path"] = "dev/vfio/70"
    assert validate_spec_obj(rel_node)

    bad_perm = copy.deepcopy(good)
    bad_perm["devices"][0]["containerEdits"]["deviceNodes"][0]["permissions"] = "rwx"
    assert validate_spec_obj(bad_perm)

    stray = copy.deepcopy(good)
    stray["devices"][0]["bdf"] = "0000:0a:00.0"   # annotation leaked to device
    assert any("unexpected" in e for e in validate_spec_obj(stray))


def test_kata_runtime_consumption_contract(tmp_path):
    """Consume the written spec exactly the way kata-runtime does
    (annotation semantics of reference device_plugin.go:62-68): look up the
    CDI device by qualified name, read attach-pci/bdf, inject the node."""
    node = make_mock_node(str(tmp_path), n_gpus=1, with_audio_fn=True,
                          kfd=False, hint=False)
    cfg = node.config()
    inv = scan_node(cfg)
    spec = build_spec(inv, "amd.com/gpu", cfg.dev_root)
    path = write_spec(spec, str(tmp_path), "kxdp-vfio", "yaml")
    doc = yaml.safe_load(open(path))
    # runtime resolves "amd.com/gpu=70": kind match + device name match
    kind, name = parse_qualified_name("amd.com/gpu=70")
    assert doc["kind"] == kind
    dev = next(d for d in doc["devices"] if d["name"] == name)
    # Kata cold-plug decision: attach-pci gate, bdf list to pass through
    assert dev["annotations"]["attach-pci"] == "true"
    bdfs = dev["annotations"]["bdf"].split(",")
    assert bdfs == ["0000:0a:00.0", "0000:0a:00.1"]  # whole IOMMU group
    # container edit the runtime applies before VM boot
    nodes = [n["path"] for n in dev["containerEdits"]["deviceNodes"]]
    assert nodes == [os.path.join(cfg.dev_root, "vfio", "70")]


def test_cdi_version_knob(tmp_path, monkeypatch):
    """KXDP_CDI_VERSION lets operators emit 0.6.0 specs for runtimes
    predating 0.8.0 (the reference's frozen version, cdi/spec.go:12);
    unknown versions are rejected at config validation."""
    import pytest as _pytest
    from kata_xpu_device_plugin_amd.config import Config
    node = make_mock_node(str(tmp_path), n_gpus=1, kfd=False, hint=False)
    cfg = node.config(cdi_version="0.6.0")
    spec = build_spec(scan_node(cfg), "amd.com/gpu", cfg.dev_root,
                      cfg.cdi_version)
    path = write_spec(spec, str(tmp_path), "s", "yaml")
    doc = yaml.safe_load(open(path))
    assert doc["cdiVersion"] == "0.6.0"
    _schema_check(path)
    with _pytest.raises(ValueError):
        node.config(cdi_version="9.9.9")
