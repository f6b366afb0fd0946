"""SR-IOV (MxGPU) VF management.

SURVEY.md §7 hard-parts item "SR-IOV/MxGPU VFs" (absent from the
reference, which handles whole GPUs only).

BASELINE.json config #5: MI355X SR-IOV VFs exposed as partitioned devices
to Kata pods. This tool drives the standard sysfs SR-IOV interface:

  enable:  echo N > /sys/bus/pci/devices/<pf>/sriov_numvfs
  VFs appear as their own functions (virtfn* links on the PF, physfn link
  on each VF) with their own IOMMU groups; bind them to vfio-pci
  (tools/bind.py) and the plugin advertises them as
  ``amd.com/INSTINCT_MI355X_VF`` automatically (discovery/naming VF
  handling).

Usage:
  python -m kata_xpu_device_plugin_amd.tools.sriov enable  <pf-bdf> <num>
  python -m kata_xpu_device_plugin_amd.tools.sriov disable <pf-bdf>
  python -m kata_xpu_device_plugin_amd.tools.sriov status
"""
from __future__ import annotations

import argparse
import os
import sys
from typing import List

from ..config import Config
from ..discovery.sysfs import scan_functions


def _pf_dir(cfg: Config, bdf: str) -> str:
    return os.path.join(cfg.sysfs_root, "bus", "pci", "devices", bdf)


def set_numvfs(cfg: Config, pf_bdf: str, num: int, dry_run: bool = False) -> None:
    d = _pf_dir(cfg, pf_bdf)
    total_path = os.path.join(d, "sriov_totalvfs")
    numvfs_path = os.path.join(d, "sriov_numvfs")
    if not os.path.exists(total_path):
        raise RuntimeError(f"{pf_bdf} does not support SR-IOV (no sriov_totalvfs)")
    with open(total_path) as f:
        total = int(f.read().strip())
    if num > total:
        raise RuntimeError(f"{pf_bdf} supports at most {total} VFs, asked {num}")
    if dry_run:
        print(f"DRY: echo {num} > {numvfs_path}")
        return
    with open(numvfs_path) as f:
        cur = int(f.read().strip())
    if cur != 0 and num != 0:
        # kernel requires 0 before a different non-zero value
        with open(numvfs_path, "w") as f:
            f.write("0")
    with open(numvfs_path, "w") as f:
        f.write(str(num))


def vf_bdfs(cfg: Config, pf_bdf: str) -> List[str]:
    d = _pf_dir(cfg, pf_bdf)
    out = []
    try:
        for ent in sorted(os.listdir(d)):
            if ent.startswith("virtfn"):
                out.append(os.path.basename(os.readlink(os.path.join(d, ent))))
    except OSError:
        pass
    return out


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="kxdp-sriov")
    sub = p.add_subparsers(dest="cmd", required=True)
    en = sub.add_parser("enable")
    en.add_argument("pf_bdf")
    en.add_argument("num", type=int)
    en.add_argument("--dry-run", action="store_true")
    dis = sub.add_parser("disable")
    dis.add_argument("pf_bdf")
    dis.add_argument("--dry-run", action="store_true")
    sub.add_parser("status")
    args = p.parse_args(argv)

    cfg = Config()
    if args.cmd == "status":
        for f in scan_functions(cfg):
            if f.sriov_totalvfs or f.is_vf:
                kind = f"VF of {f.physfn_bdf}" if f.is_vf else \
                    f"PF {f.sriov_numvfs}/{f.sriov_totalvfs} VFs"
                print(f"{f.bdf}  {f.device:04x}  {kind}  driver={f.driver or '-'}")
        return 0
    try:
        set_numvfs(cfg, args.pf_bdf, 0 if args.cmd == "disable" else args.num,
                   dry_run=args.dry_run)
        if args.cmd == "enable" and not args.dry_run:
            vfs = vf_bdfs(cfg, args.pf_bdf)
            print(f"{args.pf_bdf}: {len(vfs)} VF(s): {', '.join(vfs)}")
    except RuntimeError as e:
        print(str(e), file=sys.stderr)
        return 1
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
