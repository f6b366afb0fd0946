"""kxdp-doctor: one-shot node diagnosis / support bundle.

    python -m kata_xpu_device_plugin_amd.tools.doctor [--json] [--probe]

Aggregates everything an operator (or a support ticket) needs about a
node in one pass:

* config as resolved from env,
* PCI identification (every AMD function, drivers, IOMMU groups),
* schedulability verdict per device + CDI round-trip validation,
* topology source and hive layout,
* amd-smi health snapshot (amdgpu-bound GPUs),
* optional quick GPU probe (--probe; needs HIP-visible GPUs).

Exit 0 = node looks ready for the mode it is in (vfio-bound devices
validate end-to-end, or no vfio devices but amdgpu GPUs present pre-
provisioning); 1 = problems found; 2 = no AMD GPUs at all.
"""
from __future__ import annotations

import argparse
import dataclasses
import json
import sys
import tempfile

from ..cdi.resolve import CDIResolver
from ..cdi.spec import build_spec, qualified_name, write_spec
from ..config import Config
from ..discovery.naming import device_model_name
from ..discovery.sysfs import scan_node
from ..health.amdsmi_health import snapshot as amdsmi_snapshot
from ..topology.hive import load_topology


def diagnose(cfg: Config, probe: bool = False) -> dict:
    doc = {"config": {k: v for k, v in dataclasses.asdict(cfg).items()},
           "problems": []}
    inv = scan_node(cfg)
    doc["discovery"] = {
        "scan_ms": round(inv.scan_wall_s * 1e3, 2),
        "functions": len(inv.all_functions),
        "schedulable_devices": len(inv.devices),
        "models": {f"{did:04x}": device_model_name(did)
                   for did in inv.by_model},
    }
    gpus = [f for f in inv.all_functions
            if (f.class_code >> 16) in set(cfg.gpu_class_prefixes)]
    if not gpus:
        doc["problems"].append("no AMD GPU-class PCI functions on this node")
        return doc
    from .partition import read_partition_state
    doc["gpus"] = []
    for f in gpus:
        entry = {"bdf": f.bdf, "device": f"{f.device:04x}", "driver": f.driver,
                 "iommu_group": f.iommu_group, "numa": f.numa_node,
                 "vf": f.is_vf}
        if not f.is_vf:
            pst = read_partition_state(cfg, f.bdf, f.driver)
            if pst.supported:
                entry["partition"] = {"compute": pst.compute_current,
                                      "memory": pst.memory_current}
        doc["gpus"].append(entry)

    topo = load_topology(cfg, inv)
    doc["topology"] = {
        "source": topo.source,
        "hives": sorted({topo.hive(f.bdf) for f in gpus if topo.hive(f.bdf)}),
        "xgmi_gbps": topo.xgmi_gbps,
    }
    if inv.devices and topo.source == "none":
        doc["problems"].append(
            "vfio devices present but no xGMI topology source — run "
            "tools/topo snapshot before binding, or placement degrades "
            "to NUMA-only")

    if inv.devices:
        cdi_dir = tempfile.mkdtemp(prefix="kxdp-doctor-")
        spec = build_spec(inv, cfg.cdi_kind, cfg.dev_root, cfg.cdi_version)
        write_spec(spec, cdi_dir, cfg.cdi_spec_name, cfg.cdi_format)
        resolver = CDIResolver(cdi_dir)
        cdi_report = []
        import os
        for gid in inv.device_ids():
            qn = qualified_name(cfg.cdi_kind, gid)
            entry = {"device": qn, "ok": True}
            try:
                resolved = resolver.resolve(qn)
                missing = [n for n in resolved.device_nodes
                           if not os.path.exists(n)]
                if missing:
                    entry["ok"] = False
                    entry["error"] = f"device node(s) absent: {missing}"
                if not resolved.kata_cold_plug:
                    entry["ok"] = False
                    entry["error"] = "attach-pci annotation missing"
            except Exception as e:
                entry["ok"] = False
                entry["error"] = str(e)
            if not entry["ok"]:
                doc["problems"].append(f"{qn}: {entry.get('error')}")
            cdi_report.append(entry)
        doc["cdi"] = cdi_report
    else:
        amdgpu = [f for f in gpus if f.driver == "amdgpu"]
        if amdgpu:
            doc["note"] = (f"{len(amdgpu)} GPU(s) amdgpu-bound — node is in "
                           "pre-provisioning state (bind to vfio-pci to serve)")
        else:
            doc["problems"].append(
                f"no devices bound to {cfg.required_driver} and none on "
                "amdgpu — check driver binding")

    smi = amdsmi_snapshot()
    doc["amdsmi"] = {
        b: {"healthy": h.healthy, "temp_c": h.temperature_c,
            "uncorrectable_ecc": h.uncorrectable_errors, "reasons": h.reasons,
            "ecc_by_block": dict(h.ecc_by_block),
            "xgmi_links": [
                {"index": l.index, "status": l.status, "errors": l.errors}
                for l in h.xgmi_links
            ]}
        for b, h in smi.items()
    }
    for b, h in smi.items():
        if not h.healthy:
            doc["problems"].append(f"amd-smi: {b} unhealthy: {h.reasons}")
        if h.xgmi_sick:
            doc["problems"].append(
                f"amd-smi: {b} xGMI link degraded "
                f"(placement will avoid this GPU's hive membership)")

    if probe:
        try:
            from ..health.gpuprobe import probe_all
            reports = probe_all(bandwidth_bytes=256 << 20,
                                memtest_bytes=256 << 20, burn_iters=2000)
            doc["gpu_probe"] = [r.as_dict() for r in reports]
            for r in reports:
                if not r.passed:
                    doc["problems"].append(
                        f"gpu probe dev{r.device}: {r.failures}")
        except Exception as e:
            doc["gpu_probe_error"] = str(e)
    return doc


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="kxdp-doctor")
    p.add_argument("--json", action="store_true")
    p.add_argument("--probe", action="store_true",
                   help="also run quick GPU probes (HIP-visible GPUs)")
    args = p.parse_args(argv)
    cfg = Config()
    doc = diagnose(cfg, probe=args.probe)
    if args.json:
        json.dump(doc, sys.stdout, indent=2, default=str)
        print()
    else:
        d = doc["discovery"]
        print(f"discovery: {d['schedulable_devices']} schedulable / "
              f"{d['functions']} AMD function(s) in {d['scan_ms']} ms; "
              f"models {d['models']}")
        if "topology" in doc:
            t = doc["topology"]
            print(f"topology: source={t['source']} hives={t['hives']}")
        if "note" in doc:
            print(f"note: {doc['note']}")
        for pr in doc["problems"]:
            print(f"PROBLEM: {pr}")
        print("verdict:", "OK" if not doc["problems"] else "PROBLEMS FOUND")
    if "no AMD GPU-class PCI functions" in " ".join(doc["problems"]):
        return 2
    return 0 if not doc["problems"] else 1


if __name__ == "__main__":
    raise SystemExit(main())
