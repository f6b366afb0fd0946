"""GPU model → Kubernetes resource name.

The reference resolves names by scanning a bundled pci.ids snapshot
(`pkg/device_plugin/device_plugin.go:208-259`, data `utils/pci.ids`). We keep
that mechanism as a *fallback* (host pci.ids, if present), but lead with a
curated AMD Instinct device-ID table: pci.ids snapshots lag new silicon (the
reference's 2024 snapshot tops out at MI300X 0x74a1) and a device plugin
must name the hardware it was built for without a data file.

Names are sanitized exactly like the reference (non-alphanumeric → '_',
uppercase: `device_plugin.go:236-252`) so operators migrating from it see
familiar resource strings, e.g. ``amd.com/INSTINCT_MI355X``.
"""
from __future__ import annotations

import os
import re
from functools import lru_cache
from typing import Dict, Optional

# Curated AMD Instinct table (vendor 0x1002). IDs confirmed against live
# sysfs where possible; unknown IDs fall through to pci.ids then to a
# generic DEVICE_<hex> name, so a wrong/missing row degrades gracefully.
AMD_INSTINCT_DEVICES: Dict[int, str] = {
    0x738C: "INSTINCT_MI100",
    0x740F: "INSTINCT_MI210",
    0x7408: "INSTINCT_MI250X",
    0x740C: "INSTINCT_MI250",
    0x74A0: "INSTINCT_MI300A",
    0x74A1: "INSTINCT_MI300X",
    0x74A2: "INSTINCT_MI308X",
    0x74A5: "INSTINCT_MI325X",
    0x74A9: "INSTINCT_MI300X_HF",
    0x74B5: "INSTINCT_MI300X_VF",
    0x74B9: "INSTINCT_MI325X_VF",
    # gfx950 family (CDNA4). The MI355X id is additionally re-checked
    # against the live node at runtime (tools/ident.py) — SURVEY.md §7
    # "MI355X device-ID discovery without guessing".
    0x75A0: "INSTINCT_MI350X",
    0x75A3: "INSTINCT_MI355X",
    0x75B0: "INSTINCT_MI350X_VF",
    0x75B3: "INSTINCT_MI355X_VF",
}

_SANITIZE_RE = re.compile(r"[^A-Za-z0-9_]+")


def sanitize(name: str) -> str:
    """'Instinct MI355X [OAM]' → 'INSTINCT_MI355X_OAM_' → trimmed."""
    return _SANITIZE_RE.sub("_", name).strip("_").upper()


@lru_cache(maxsize=8)
def _load_pci_ids(path: str, vendor: int) -> Dict[int, str]:
    """Parse one vendor's block of a pci.ids file → {device_id: raw name}.

    Format: vendor lines start at column 0 ('1002  Advanced Micro ...'),
    device lines are '\t<id>  <name>', subsystem lines are '\t\t...'.
    (Same data format the reference walks at `device_plugin.go:208-259`.)
    """
    out: Dict[int, str] = {}
    try:
        f = open(path, "r", encoding="utf-8", errors="replace")
    except OSError:
        return out
    with f:
        in_vendor = False
        for line in f:
            if not line or line.startswith("#"):
                continue
            if not line.startswith("\t"):
                # vendor line or section break
                head = line.split(None, 1)[0] if line.strip() else ""
                if len(head) == 4:
                    try:
                        in_vendor = int(head, 16) == vendor
                    except ValueError:
                        in_vendor = False
                else:
                    in_vendor = False
                continue
            if not in_vendor or line.startswith("\t\t"):
                continue
            parts = line.strip().split(None, 1)
            if len(parts) != 2:
                continue
            try:
                dev_id = int(parts[0], 16)
            except ValueError:
                continue
            out[dev_id] = parts[1].strip()
    return out


def device_model_name(
    device_id: int,
    vendor: int = 0x1002,
    pci_ids_paths: Optional[tuple] = None,
    is_vf: bool = False,
) -> str:
    """Resolve a PCI device id to a sanitized model name."""
    name = AMD_INSTINCT_DEVICES.get(device_id)
    if name is None and pci_ids_paths:
        for path in pci_ids_paths:
            if not os.path.exists(path):
                continue
            raw = _load_pci_ids(path, vendor).get(device_id)
            if raw:
                name = sanitize(raw)
                break
    if name is None:
        name = f"DEVICE_{device_id:04X}"
    if is_vf and not name.endswith("_VF"):
        name += "_VF"
    return name


def resource_name(
    device_id: int,
    namespace: str = "amd.com",
    vendor: int = 0x1002,
    pci_ids_paths: Optional[tuple] = None,
    is_vf: bool = False,
    unified: str = "",
) -> str:
    """Full extended-resource name, e.g. 'amd.com/INSTINCT_MI355X'."""
    if unified:
        return f"{namespace}/{sanitize(unified)}"
    return f"{namespace}/{device_model_name(device_id, vendor, pci_ids_paths, is_vf)}"
