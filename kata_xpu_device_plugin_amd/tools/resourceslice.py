"""Dump this node's ResourceSlice (DRA control-plane inventory).

    python -m kata_xpu_device_plugin_amd.tools.resourceslice [--node NAME]

Produces the resource.k8s.io/v1beta1 ResourceSlice object a control-plane
publisher would apply for this node: one named device per IOMMU group
with PCI model, BDF, xGMI hive and NUMA attributes the DRA scheduler can
select on (see plugin/dra.py — experimental).
"""
from __future__ import annotations

import argparse
import json
import socket
import sys

from ..config import Config
from ..discovery.sysfs import scan_node
from ..plugin.dra import resource_slice_obj
from ..topology.hive import load_topology


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="kxdp-resourceslice")
    Config.add_args(p)
    p.add_argument("--node", default=socket.gethostname())
    p.add_argument("--pool", default="default")
    args = p.parse_args(argv)
    cfg = Config.from_args(args)
    inv = scan_node(cfg)
    topo = load_topology(cfg, inv)
    json.dump(resource_slice_obj(inv, topo, args.node, args.pool),
              sys.stdout, indent=2)
    print()
    return 0 if inv.devices else 1


if __name__ == "__main__":
    raise SystemExit(main())
