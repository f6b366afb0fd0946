"""kxdp-partition: MI355X compute/memory partition-mode provisioning.

    python -m kata_xpu_device_plugin_amd.tools.partition show
    python -m kata_xpu_device_plugin_amd.tools.partition set <bdf> \
        --compute CPX [--memory NPS4] [--dry-run]

CDNA3/CDNA4 Instinct GPUs partition at two levels (amdgpu sysfs,
`current_compute_partition` / `current_memory_partition` on the PCI
device dir — the AMD analog of NVIDIA MIG for sizing what a passthrough
node exposes):

* compute: SPX (1 partition) / DPX / TPX / QPX / CPX (one per XCD — 8 on
  MI355X); with SR-IOV each partition can back a VF,
* memory: NPS1 (unified HBM) / NPS4 (quadrant) — changing it requires
  the amdgpu driver to reinitialize the GPU.

Like `tools/sriov` and `tools/bind`, this is a PRE-PROVISIONING tool: it
only operates on amdgpu-bound GPUs (a vfio-bound GPU's partition state
belongs to the guest). After repartitioning, re-run `tools/topo snapshot`
and let the daemon rescan (SIGHUP) so placement and resource counts
reflect the new layout.

The reference has no partitioning support at all (SURVEY.md §2.2 —
whole-GPU VFIO only).
"""
from __future__ import annotations

import argparse
import json
import os
import sys
from dataclasses import dataclass, field
from typing import List, Optional

from ..config import Config
from ..discovery.sysfs import scan_functions

COMPUTE_MODES = ("SPX", "DPX", "TPX", "QPX", "CPX")
MEMORY_MODES = ("NPS1", "NPS2", "NPS4", "NPS8")


@dataclass
class PartitionState:
    bdf: str
    driver: str = ""
    compute_current: str = ""
    compute_available: List[str] = field(default_factory=list)
    memory_current: str = ""
    memory_available: List[str] = field(default_factory=list)

    @property
    def supported(self) -> bool:
        return bool(self.compute_current or self.memory_current)


def _read(path: str) -> str:
    try:
        with open(path) as f:
            return f.read().strip()
    except OSError:
        return ""


def _read_list(path: str) -> List[str]:
    raw = _read(path)
    return [t for t in raw.replace(",", " ").split() if t]


def read_partition_state(cfg: Config, bdf: str, driver: str = "") -> PartitionState:
    d = os.path.join(cfg.sysfs_root, "bus", "pci", "devices", bdf)
    return PartitionState(
        bdf=bdf,
        driver=driver,
        compute_current=_read(os.path.join(d, "current_compute_partition")),
        compute_available=_read_list(
            os.path.join(d, "available_compute_partition")),
        memory_current=_read(os.path.join(d, "current_memory_partition")),
        memory_available=_read_list(
            os.path.join(d, "available_memory_partition")),
    )


def list_states(cfg: Config) -> List[PartitionState]:
    out = []
    for fn in scan_functions(cfg):
        if not fn.is_gpu or fn.is_vf:
            continue
        out.append(read_partition_state(cfg, fn.bdf, fn.driver))
    return out


def set_partition(cfg: Config, bdf: str, compute: Optional[str] = None,
                  memory: Optional[str] = None, dry_run: bool = False) -> None:
    """Write the requested partition mode(s). Raises on any refusal so
    operators see exactly why (wrong driver, unsupported mode, EBUSY)."""
    fn = next((f for f in scan_functions(cfg) if f.bdf == bdf), None)
    if fn is None:
        raise ValueError(f"{bdf}: not an AMD PCI function on this node")
    if fn.driver != "amdgpu":
        raise ValueError(
            f"{bdf}: bound to {fn.driver or '(none)'} — partitioning needs "
            "amdgpu (unbind from vfio-pci first; a vfio GPU's partition "
            "state belongs to the guest)")
    st = read_partition_state(cfg, bdf, fn.driver)
    d = os.path.join(cfg.sysfs_root, "bus", "pci", "devices", bdf)
    for want, kind, cur, avail, fname in (
        (compute, "compute", st.compute_current, st.compute_available,
         "current_compute_partition"),
        (memory, "memory", st.memory_current, st.memory_available,
         "current_memory_partition"),
    ):
        if want is None:
            continue
        want = want.upper()
        if not cur:
            raise ValueError(f"{bdf}: {kind} partitioning not supported "
                             f"(no {fname} in sysfs)")
        if avail and want not in avail:
            raise ValueError(
                f"{bdf}: {kind} mode {want} not in available set {avail}")
        if want == cur:
            print(f"{bdf}: {kind} already {want}")
            continue
        path = os.path.join(d, fname)
        if dry_run:
            print(f"DRY: echo {want} > {path}")
            continue
        try:
            with open(path, "w") as f:
                f.write(want)
        except OSError as e:
            raise RuntimeError(
                f"{bdf}: writing {want} to {fname} failed: {e} "
                "(memory-mode changes may need a driver reload; "
                "compute-mode changes need an idle GPU)") from e
        print(f"{bdf}: {kind} partition → {want}")


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="kxdp-partition")
    Config.add_args(p)
    sub = p.add_subparsers(dest="cmd", required=True)
    sub.add_parser("show", help="partition state of every AMD GPU PF")
    ps = sub.add_parser("set", help="set compute/memory partition mode")
    ps.add_argument("bdf")
    ps.add_argument("--compute", choices=COMPUTE_MODES, type=str.upper)
    ps.add_argument("--memory", choices=MEMORY_MODES, type=str.upper)
    ps.add_argument("--dry-run", action="store_true")
    args = p.parse_args(argv)
    cfg = Config.from_args(args)

    if args.cmd == "show":
        states = list_states(cfg)
        out = []
        for st in states:
            out.append({
                "bdf": st.bdf, "driver": st.driver,
                "supported": st.supported,
                "compute": {"current": st.compute_current,
                            "available": st.compute_available},
                "memory": {"current": st.memory_current,
                           "available": st.memory_available},
            })
        json.dump(out, sys.stdout, indent=2)
        print()
        return 0

    if args.compute is None and args.memory is None:
        print("set: nothing to do (pass --compute and/or --memory)",
              file=sys.stderr)
        return 2
    try:
        set_partition(cfg, args.bdf, args.compute, args.memory,
                      dry_run=args.dry_run)
    except (ValueError, RuntimeError) as e:
        print(f"error: {e}", file=sys.stderr)
        return 1
    print("re-run `kxdp-topo snapshot` and SIGHUP the daemon so placement "
          "and resource counts reflect the new layout")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
