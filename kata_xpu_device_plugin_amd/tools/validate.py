"""Node validation: does every advertised device resolve end-to-end?

    python -m kata_xpu_device_plugin_amd.tools.validate

Runs discovery, generates the CDI spec (to a temp dir unless --cdi-dir),
then resolves every schedulable device the way the container runtime
would, checking the Kata cold-plug contract (attach-pci/bdf annotations,
/dev/vfio node existence). Exit 0 iff the node is consistent.
"""
from __future__ import annotations

import argparse
import os
import sys
import tempfile

from ..cdi.resolve import CDIResolver
from ..cdi.spec import build_spec, qualified_name, write_spec
from ..config import Config
from ..discovery.sysfs import scan_node


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="kxdp-validate")
    Config.add_args(p)
    args = p.parse_args(argv)
    cfg = Config.from_args(args)

    inv = scan_node(cfg)
    print(f"discovered {len(inv.devices)} schedulable device(s) "
          f"in {inv.scan_wall_s * 1e3:.1f} ms")
    if not inv.devices:
        print("nothing to validate (no vfio-bound GPU-class functions)",
              file=sys.stderr)
        return 1
    cdi_dir = cfg.cdi_dir if os.access(cfg.cdi_dir, os.W_OK) else \
        tempfile.mkdtemp(prefix="kxdp-validate-")
    spec = build_spec(inv, cfg.cdi_kind, cfg.dev_root, cfg.cdi_version)
    spec_file = write_spec(spec, cdi_dir, cfg.cdi_spec_name, cfg.cdi_format)
    from ..cdi.schema import validate_spec_file
    schema_problems = validate_spec_file(spec_file)
    if schema_problems:
        print(f"FAIL CDI schema: {schema_problems}")
        return 1
    print(f"CDI spec {spec_file}: schema-valid (CDI {spec.cdi_version})")
    resolver = CDIResolver(cdi_dir)

    rc = 0
    for gid in inv.device_ids():
        dev = inv.devices[gid]
        qn = qualified_name(cfg.cdi_kind, gid)
        try:
            resolved = resolver.resolve(qn)
        except Exception as e:
            print(f"FAIL {qn}: {e}")
            rc = 1
            continue
        problems = []
        if not resolved.kata_cold_plug:
            problems.append("missing attach-pci annotation")
        if resolved.bdfs != dev.bdfs:
            problems.append(f"bdf mismatch {resolved.bdfs} != {dev.bdfs}")
        for node in resolved.device_nodes:
            if not os.path.exists(node):
                problems.append(f"device node {node} absent")
        status = "OK " if not problems else "FAIL"
        if problems:
            rc = 1
        print(f"{status} {qn} → {','.join(resolved.bdfs)} "
              f"{resolved.device_nodes}{' — ' + '; '.join(problems) if problems else ''}")
    return rc


if __name__ == "__main__":
    raise SystemExit(main())
