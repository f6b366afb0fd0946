"""Naming tests (reference: getDeviceName device_plugin.go:208-259)."""
from kata_xpu_device_plugin_amd.discovery.naming import (
    device_model_name,
    resource_name,
    sanitize,
)


def test_curated_table():
    assert device_model_name(0x75A3) == "INSTINCT_MI355X"
    assert device_model_name(0x74A1) == "INSTINCT_MI300X"
    assert device_model_name(0x75B3) == "INSTINCT_MI355X_VF"


def test_vf_suffix():
    assert device_model_name(0x75A3, is_vf=True) == "INSTINCT_MI355X_VF"
    # no double suffix
    assert device_model_name(0x75B3, is_vf=True) == "INSTINCT_MI355X_VF"


def test_unknown_device_generic_name():
    assert device_model_name(0x1234) == "DEVICE_1234"


def test_pci_ids_fallback(tmp_path):
    ids = tmp_path / "pci.ids"
    ids.write_text(
        "# test pci.ids\n"
        "1002  Advanced Micro Devices, Inc. [AMD/ATI]\n"
        "\t9999  Fictional Accelerator [Test]\n"
        "\t\t1002 0001  Subsystem line must be ignored\n"
        "10de  NVIDIA Corporation\n"
        "\t9999  Wrong vendor block\n"
    )
    name = device_model_name(0x9999, pci_ids_paths=(str(ids),))
    assert name == "FICTIONAL_ACCELERATOR_TEST"


def test_sanitize():
    # Same semantics as the reference's sanitizer (device_plugin.go:236-252):
    # non-alphanumerics collapse to '_', uppercase, trimmed.
    assert sanitize("Instinct MI355X [OAM]") == "INSTINCT_MI355X_OAM"
    assert sanitize("a  b--c") == "A_B_C"


def test_resource_name():
    assert resource_name(0x75A3) == "amd.com/INSTINCT_MI355X"
    assert resource_name(0x75A3, namespace="example.org") == "example.org/INSTINCT_MI355X"
    assert resource_name(0x75A3, unified="gpu") == "amd.com/GPU"
