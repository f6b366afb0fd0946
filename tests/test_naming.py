"""Naming tests (reference: getDeviceName device_plugin.go:208-259)."""
import os

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


# --- bundled pci.ids + full fallback chain -------------

def test_bundled_pci_ids_exists_and_parses():
    from kata_xpu_device_plugin_amd.config import bundled_pci_ids_path
    from kata_xpu_device_plugin_amd.discovery.naming import _load_pci_ids
    path = bundled_pci_ids_path()
    assert os.path.exists(path), "bundled pci.ids must ship with the package"
    amd = _load_pci_ids(path, 0x1002)
    assert amd[0x74A1].startswith("Aqua Vanjaram")
    assert "MI355X" in amd[0x75A3]
    assert 0x73BF in amd   # consumer silicon beyond the curated table


def test_bundled_pci_ids_is_default_first_fallback(monkeypatch):
    from kata_xpu_device_plugin_amd.config import Config, bundled_pci_ids_path
    monkeypatch.delenv("KXDP_PCI_IDS", raising=False)
    cfg = Config()
    assert cfg.pci_ids_paths[0] == bundled_pci_ids_path()
    assert "/usr/pci.ids" in cfg.pci_ids_paths  # reference's side-load path


def test_naming_fallback_chain(tmp_path):
    """curated table → bundled pci.ids → host pci.ids → DEVICE_XXXX."""
    from kata_xpu_device_plugin_amd.config import bundled_pci_ids_path
    from kata_xpu_device_plugin_amd.discovery.naming import device_model_name

    host_ids = tmp_path / "host-pci.ids"
    host_ids.write_text(
        "1002  Advanced Micro Devices, Inc. [AMD/ATI]\n"
        "\tdead  Hypothetical Future GPU\n")
    chain = (bundled_pci_ids_path(), str(host_ids))

    # 1. curated wins without touching any file
    assert device_model_name(0x75A3, pci_ids_paths=chain) == "INSTINCT_MI355X"
    # 2. not curated → bundled snapshot names it
    assert device_model_name(0x73BF, pci_ids_paths=chain) == \
        "NAVI_21_RADEON_RX_6800_6800_XT_6900_XT"
    # 3. not curated, not bundled → host pci.ids
    assert device_model_name(0xDEAD, pci_ids_paths=chain) == \
        "HYPOTHETICAL_FUTURE_GPU"
    # 4. nowhere → deterministic generic name
    assert device_model_name(0xBEEF, pci_ids_paths=chain) == "DEVICE_BEEF"
    # 4b. empty chain (KXDP_PCI_IDS="") also degrades gracefully
    assert device_model_name(0xBEEF, pci_ids_paths=()) == "DEVICE_BEEF"
