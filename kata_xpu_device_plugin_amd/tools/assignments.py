"""Show which pods currently hold which xPU devices.

    python -m kata_xpu_device_plugin_amd.tools.assignments \
        [--socket /var/lib/kubelet/pod-resources/kubelet.sock]

Built on the live pod-resources client (plugin/podresources.py) — the
capability the reference shipped as dead code (utils/pod_resources.go).
"""
from __future__ import annotations

import argparse
import json
import sys

from ..plugin.podresources import DEFAULT_SOCKET, PodResourcesClient


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="kxdp-assignments")
    p.add_argument("--socket", default=DEFAULT_SOCKET)
    p.add_argument("--prefix", default="amd.com/")
    p.add_argument("--allocatable", action="store_true",
                   help="show kubelet's allocatable capacity view instead "
                        "of current pod assignments")
    args = p.parse_args(argv)
    client = PodResourcesClient(args.socket)
    try:
        data = (client.allocatable(args.prefix) if args.allocatable
                else client.assignments(args.prefix))
    except Exception as e:
        print(f"cannot query pod-resources at {args.socket}: {e}", file=sys.stderr)
        return 1
    json.dump(data, sys.stdout, indent=2)
    print()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
