"""GPU burn-in CLI: validate every visible GPU before vfio binding.

    python -m kata_xpu_device_plugin_amd.tools.burnin [--json] [--quick]

Exit code 0 iff every GPU passes. See health/gpuprobe.py for the probes.
"""
from __future__ import annotations

import argparse
import json
import sys

from ..health.gpuprobe import probe_all
from ..utils.log import configure


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="kxdp-burnin")
    p.add_argument("--json", action="store_true", help="machine-readable output")
    p.add_argument("--quick", action="store_true",
                   help="small buffers / short burn (smoke, not soak)")
    p.add_argument("--bandwidth-gib", type=float, default=2.0)
    p.add_argument("--memtest-gib", type=float, default=2.0)
    p.add_argument("--parallel", action="store_true",
                   help="probe all GPUs concurrently (one process per GPU)")
    p.add_argument("--soak-minutes", type=float, default=0.0,
                   help="repeat the full suite for this long; any failing "
                        "pass fails the soak (thermal/sustained validation)")
    args = p.parse_args(argv)
    configure("INFO")

    kw = {}
    if args.quick:
        kw = dict(bandwidth_bytes=64 << 20, memtest_bytes=64 << 20, burn_iters=2000)
    else:
        kw = dict(
            bandwidth_bytes=int(args.bandwidth_gib * (1 << 30)),
            memtest_bytes=int(args.memtest_gib * (1 << 30)),
        )
    if args.soak_minutes > 0:
        import time
        deadline = time.monotonic() + args.soak_minutes * 60
        passes, failures = 0, 0
        worst = []
        while time.monotonic() < deadline:
            reports = probe_all(parallel=args.parallel, **kw)
            if not reports:
                print("no GPUs visible", file=sys.stderr)
                return 2
            passes += 1
            if not all(r.passed for r in reports):
                failures += 1
                worst = [r.as_dict() for r in reports if not r.passed]
        summary = {"soak_minutes": args.soak_minutes, "passes": passes,
                   "failing_passes": failures, "failures": worst}
        json.dump(summary, sys.stdout, indent=2)
        print()
        return 0 if failures == 0 else 1

    reports = probe_all(parallel=args.parallel, **kw)
    if args.json:
        json.dump([r.as_dict() for r in reports], sys.stdout, indent=2)
        print()
    if not reports:
        print("no GPUs visible", file=sys.stderr)
        return 2
    return 0 if all(r.passed for r in reports) else 1


if __name__ == "__main__":
    raise SystemExit(main())
