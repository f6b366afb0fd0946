"""KFD topology reader: xGMI hives from /sys/class/kfd.

The AMD KFD driver exports the fabric topology at
``/sys/class/kfd/kfd/topology/nodes/<n>/``:

* ``properties`` — key/value lines; GPUs have ``simd_count > 0``; the xGMI
  hive is ``hive_id`` (same non-zero value = same fabric); the PCI location
  is ``domain`` + ``location_id`` (location_id = bus<<8 | devfn);
* ``io_links/<m>/properties`` — ``type 11`` is XGMI (type 2 = PCIe), with
  ``node_to`` and ``min/max_bandwidth`` in MB/s.

Note: KFD only enumerates **amdgpu-bound** GPUs. On a node whose GPUs are
already vfio-pci-bound (the device plugin's normal state) this reader sees
nothing — that is why topology/hive.py also supports a snapshot hint file
generated while the GPUs were amdgpu-bound (tools/topo.py). SURVEY.md §2.2
"xGMI-hive topology from sysfs (io_links or amd-smi topology)".
"""
from __future__ import annotations

import os
import re
from dataclasses import dataclass, field
from typing import Dict, List, Optional

IOLINK_TYPE_PCIE = 2
IOLINK_TYPE_XGMI = 11


@dataclass
class KFDLink:
    node_to: int
    type: int
    max_bandwidth_mbps: int = 0


@dataclass
class KFDNode:
    node_id: int
    is_gpu: bool
    hive_id: int = 0
    bdf: Optional[str] = None        # "0000:0a:00.0" from domain+location_id
    links: List[KFDLink] = field(default_factory=list)
    gfx_target_version: int = 0

    @property
    def xgmi_links(self) -> List[KFDLink]:
        return [l for l in self.links if l.type == IOLINK_TYPE_XGMI]


_PROP_RE = re.compile(r"^(\S+)\s+(-?\d+)\s*$")


def _read_props(path: str) -> Dict[str, int]:
    out: Dict[str, int] = {}
    try:
        with open(path) as f:
            for line in f:
                m = _PROP_RE.match(line)
                if m:
                    out[m.group(1)] = int(m.group(2))
    except OSError:
        pass
    return out


def _bdf_from_location(domain: int, location_id: int) -> str:
    bus = (location_id >> 8) & 0xFF
    devfn = location_id & 0xFF
    return f"{domain:04x}:{bus:02x}:{devfn >> 3:02x}.{devfn & 0x7}"


def read_kfd_topology(sysfs_root: str = "/sys") -> List[KFDNode]:
    """Parse all KFD topology nodes (GPUs and CPUs)."""
    base = os.path.join(sysfs_root, "class", "kfd", "kfd", "topology", "nodes")
    nodes: List[KFDNode] = []
    try:
        entries = sorted(os.listdir(base), key=lambda s: int(s) if s.isdigit() else 1 << 30)
    except OSError:
        return nodes
    for ent in entries:
        if not ent.isdigit():
            continue
        ndir = os.path.join(base, ent)
        props = _read_props(os.path.join(ndir, "properties"))
        if not props:
            continue
        is_gpu = props.get("simd_count", 0) > 0
        node = KFDNode(
            node_id=int(ent),
            is_gpu=is_gpu,
            hive_id=props.get("hive_id", 0),
            gfx_target_version=props.get("gfx_target_version", 0),
        )
        if is_gpu:
            node.bdf = _bdf_from_location(
                props.get("domain", 0), props.get("location_id", 0)
            )
        links_dir = os.path.join(ndir, "io_links")
        if os.path.isdir(links_dir):
            for lent in sorted(os.listdir(links_dir)):
                lprops = _read_props(os.path.join(links_dir, lent, "properties"))
                if not lprops:
                    continue
                node.links.append(
                    KFDLink(
                        node_to=lprops.get("node_to", -1),
                        type=lprops.get("type", 0),
                        max_bandwidth_mbps=lprops.get("max_bandwidth", 0),
                    )
                )
        nodes.append(node)
    return nodes
