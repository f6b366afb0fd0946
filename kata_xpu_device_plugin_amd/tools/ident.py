"""Node identification: what AMD silicon is here, under which driver?

    python -m kata_xpu_device_plugin_amd.tools.ident

Prints every vendor-0x1002 PCI function with device id, class, driver,
IOMMU group, plus KFD gfx target and amd-smi ASIC info when available.
Used to confirm real MI355X device IDs against the curated naming table
("MI355X device-ID discovery without guessing").
"""
from __future__ import annotations

import json
import sys

from ..config import Config
from ..discovery.naming import device_model_name
from ..discovery.sysfs import scan_functions
from ..topology.kfd import read_kfd_topology


def collect(cfg: Config) -> dict:
    fns = scan_functions(cfg)
    doc = {
        "functions": [
            {
                "bdf": f.bdf,
                "device": f"{f.device:04x}",
                "class": f"{f.class_code:06x}",
                "driver": f.driver,
                "iommu_group": f.iommu_group,
                "numa": f.numa_node,
                "is_vf": f.is_vf,
                "model_name": device_model_name(
                    f.device, pci_ids_paths=tuple(cfg.pci_ids_paths)),
            }
            for f in fns
        ],
        "kfd": [
            {"bdf": n.bdf, "hive_id": n.hive_id,
             "gfx_target_version": n.gfx_target_version,
             "xgmi_links": len(n.xgmi_links)}
            for n in read_kfd_topology(cfg.sysfs_root) if n.is_gpu
        ],
    }
    try:
        import amdsmi
        amdsmi.amdsmi_init()
        try:
            devs = amdsmi.amdsmi_get_processor_handles()
            doc["amdsmi"] = []
            for h in devs:
                info = {}
                try:
                    asic = amdsmi.amdsmi_get_gpu_asic_info(h)
                    info.update({k: str(v) for k, v in asic.items()})
                except Exception as e:
                    info["asic_error"] = str(e)
                doc["amdsmi"].append(info)
        finally:
            amdsmi.amdsmi_shut_down()
    except Exception as e:
        doc["amdsmi_error"] = str(e)
    return doc


def main(argv=None) -> int:
    cfg = Config()
    json.dump(collect(cfg), sys.stdout, indent=2)
    print()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
