"""vfio-pci bind/unbind provisioning tool.

Takes a GPU from amdgpu (host driver) to vfio-pci (passthrough) and back —
the node-provisioning step between `tools/topo snapshot` and deploying the
DaemonSet. Works through the standard sysfs driver interfaces:

  unbind:        /sys/bus/pci/devices/<bdf>/driver/unbind
  driver select: /sys/bus/pci/devices/<bdf>/driver_override
  rebind:        /sys/bus/pci/drivers_probe (or <driver>/bind)

Usage:
  python -m kata_xpu_device_plugin_amd.tools.bind to-vfio   [BDF ...|--all-gpus]
  python -m kata_xpu_device_plugin_amd.tools.bind to-amdgpu [BDF ...|--all-gpus]
  python -m kata_xpu_device_plugin_amd.tools.bind status
"""
from __future__ import annotations

import argparse
import os
import sys
from typing import List, Optional

from ..config import Config
from ..discovery.sysfs import PCIFunction, scan_functions


def _write(path: str, value: str) -> None:
    with open(path, "w") as f:
        f.write(value)


class Binder:
    def __init__(self, cfg: Config, dry_run: bool = False):
        self.cfg = cfg
        self.dry_run = dry_run
        self.devices_dir = os.path.join(cfg.sysfs_root, "bus", "pci", "devices")
        self.drivers_probe = os.path.join(cfg.sysfs_root, "bus", "pci", "drivers_probe")

    def _dev_dir(self, bdf: str) -> str:
        return os.path.join(self.devices_dir, bdf)

    def _do(self, path: str, value: str, what: str) -> None:
        if self.dry_run:
            print(f"DRY: echo {value.strip()!r} > {path}")
            return
        try:
            _write(path, value)
        except OSError as e:
            raise RuntimeError(f"{what} failed for {path}: {e}") from e

    def rebind(self, bdf: str, driver: Optional[str]) -> None:
        """Move one function to `driver` (None = kernel default match)."""
        d = self._dev_dir(bdf)
        if not os.path.isdir(d):
            raise RuntimeError(f"no such device {bdf}")
        cur = None
        drv_link = os.path.join(d, "driver")
        if os.path.islink(drv_link):
            cur = os.path.basename(os.readlink(drv_link))
            if cur == driver:
                return
            self._do(os.path.join(drv_link, "unbind"), bdf, "unbind")
        self._do(os.path.join(d, "driver_override"), (driver or "") + "\n",
                 "driver_override")
        self._do(self.drivers_probe, bdf, "drivers_probe")

    def gpu_functions(self) -> List[PCIFunction]:
        prefixes = set(self.cfg.gpu_class_prefixes)
        return [f for f in scan_functions(self.cfg)
                if (f.class_code >> 16) in prefixes]


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="kxdp-bind")
    p.add_argument("cmd", choices=("to-vfio", "to-amdgpu", "status"))
    p.add_argument("bdfs", nargs="*", help="PCI addresses (default with --all-gpus)")
    p.add_argument("--all-gpus", action="store_true",
                   help="apply to every discovered AMD GPU function")
    p.add_argument("--dry-run", action="store_true")
    args = p.parse_args(argv)

    cfg = Config()
    binder = Binder(cfg, dry_run=args.dry_run)
    if args.cmd == "status":
        for f in binder.gpu_functions():
            print(f"{f.bdf}  {f.device:04x}  driver={f.driver or '-'}  "
                  f"iommu={f.iommu_group or '-'}  vf={f.is_vf}")
        return 0

    targets = list(args.bdfs)
    if args.all_gpus:
        targets = [f.bdf for f in binder.gpu_functions()]
    if not targets:
        print("no target BDFs (pass addresses or --all-gpus)", file=sys.stderr)
        return 2
    driver = "vfio-pci" if args.cmd == "to-vfio" else None
    rc = 0
    for bdf in targets:
        try:
            binder.rebind(bdf, driver)
            print(f"{bdf} → {driver or 'default driver'}")
        except RuntimeError as e:
            print(str(e), file=sys.stderr)
            rc = 1
    return rc


if __name__ == "__main__":
    raise SystemExit(main())
