"""Topology snapshot tool.

MI355X-native analog of SURVEY.md §2.2's "xGMI-hive topology from
sysfs" concept row (the reference has no topology awareness at all; its
GetPreferredAllocation is a stub, generic_device_plugin.go:378-386).

KFD only enumerates amdgpu-bound GPUs, so the xGMI hive layout must be
captured BEFORE binding GPUs to vfio-pci. Run this once per node (e.g. an
init container or provisioning step):

    python -m kata_xpu_device_plugin_amd.tools.topo snapshot \
        [--out /etc/kata-xpu-amd/topology.json]

The daemon's topology loader (topology/hive.py) falls back to this file
when KFD is empty.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
from typing import Dict, List

from ..topology.kfd import read_kfd_topology


def build_snapshot(sysfs_root: str = "/sys") -> Dict:
    nodes = [n for n in read_kfd_topology(sysfs_root) if n.is_gpu and n.bdf]
    hives: Dict[int, List[str]] = {}
    max_bw = 0
    for n in nodes:
        if n.hive_id:
            hives.setdefault(n.hive_id, []).append(n.bdf)
        for l in n.xgmi_links:
            max_bw = max(max_bw, l.max_bandwidth_mbps)
    return {
        "version": 1,
        "hives": [sorted(v) for _, v in sorted(hives.items())],
        "xgmi_link_gbps": max_bw / 1000.0,
        "gpus": [
            {"bdf": n.bdf, "hive_id": n.hive_id,
             "gfx_target_version": n.gfx_target_version}
            for n in nodes
        ],
    }


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="kxdp-topo")
    sub = p.add_subparsers(dest="cmd", required=True)
    snap = sub.add_parser("snapshot", help="write xGMI topology snapshot from KFD")
    snap.add_argument("--sysfs-root", default="/sys")
    snap.add_argument("--out", default="/etc/kata-xpu-amd/topology.json")
    show = sub.add_parser("show", help="print topology from KFD as JSON")
    show.add_argument("--sysfs-root", default="/sys")
    args = p.parse_args(argv)

    doc = build_snapshot(args.sysfs_root)
    if args.cmd == "show":
        json.dump(doc, sys.stdout, indent=2)
        print()
        return 0
    if not doc["gpus"]:
        print("no amdgpu-bound GPUs visible in KFD; nothing to snapshot",
              file=sys.stderr)
        return 1
    os.makedirs(os.path.dirname(args.out), exist_ok=True)
    with open(args.out, "w") as f:
        json.dump(doc, f, indent=2)
    print(f"wrote {args.out}: {len(doc['gpus'])} GPUs, "
          f"{len(doc['hives'])} hive(s)")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
