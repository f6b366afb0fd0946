"""CDI (Container Device Interface) spec generation for VFIO passthrough.

Replaces the reference's hand-rolled CDI structs + writer (`cdi/spec.go:17-127`)
and its driver `generateCDISpec` (`pkg/device_plugin/device_plugin.go:55-80`).

The **Kata contract** is kept intact — it is what the Kata v3.8+ runtime
consumes to cold-plug the VFIO device into the guest VM (reference
README.md:3, annotations built at `device_plugin.go:62-68`):

* per-device annotation ``attach-pci: "true"``
* per-device annotation ``bdf: <pci-address>`` (comma-joined when the IOMMU
  group holds multiple functions)
* per-device annotation ``cdi.k8s.io/vfio<group>: <kind>=<name>``
* container edit: device node ``/dev/vfio/<group>``

Deliberate divergences:

* One CDI device **per IOMMU group**, named by the group id (the unit
  kubelet schedules — `device_plugin.go:93-98`), not one per PCI function.
  The reference's per-function layout emits duplicate /dev/vfio/<group>
  edits for multi-function groups.
* CDI version 0.8.0 (current schema level of the `cdi.k8s.io` ecosystem)
  instead of the frozen 0.6.0 (`cdi/spec.go:12`).
* Atomic write (tmp + rename) with 0644, JSON or YAML; a stale-spec file
  for the same spec name in the other format is removed.
"""
from __future__ import annotations

import json
import os
import re
import tempfile
from dataclasses import dataclass, field
from typing import Dict, List, Tuple

import yaml

from ..discovery.sysfs import NodeInventory, XPUDevice
from ..utils.log import get_logger

log = get_logger(__name__)

CDI_VERSION = "0.8.0"
ANNOTATION_ATTACH_PCI = "attach-pci"
ANNOTATION_BDF = "bdf"
ANNOTATION_PREFIX = "cdi.k8s.io/"  # reference: cdi/constant.go:11

_NAME_RE = re.compile(r"^[A-Za-z0-9][A-Za-z0-9_.:-]*$")
_KIND_RE = re.compile(r"^[A-Za-z0-9][A-Za-z0-9.-]*/[A-Za-z0-9][A-Za-z0-9_.-]*$")


def qualified_name(kind: str, name: str) -> str:
    """'amd.com/gpu' + '70' → 'amd.com/gpu=70' (reference cdi/cdi-utils.go:9-11)."""
    return f"{kind}={name}"


def parse_qualified_name(qn: str) -> Tuple[str, str]:
    kind, sep, name = qn.partition("=")
    if not sep or not _KIND_RE.match(kind) or not _NAME_RE.match(name):
        raise ValueError(f"invalid CDI qualified name: {qn!r}")
    return kind, name


@dataclass
class CDIDeviceEntry:
    name: str
    annotations: Dict[str, str]
    device_nodes: List[str]  # host paths

    def to_obj(self) -> dict:
        return {
            "name": self.name,
            "annotations": dict(self.annotations),
            "containerEdits": {
                "deviceNodes": [{"path": p, "permissions": "rw"} for p in self.device_nodes]
            },
        }


@dataclass
class CDISpec:
    kind: str
    devices: List[CDIDeviceEntry] = field(default_factory=list)
    cdi_version: str = CDI_VERSION

    def to_obj(self) -> dict:
        return {
            "cdiVersion": self.cdi_version,
            "kind": self.kind,
            "devices": [d.to_obj() for d in self.devices],
        }

    def device_names(self) -> List[str]:
        return [d.name for d in self.devices]

    def validate(self) -> None:
        if not _KIND_RE.match(self.kind):
            raise ValueError(f"invalid CDI kind {self.kind!r}")
        seen = set()
        for d in self.devices:
            if not _NAME_RE.match(d.name):
                raise ValueError(f"invalid CDI device name {d.name!r}")
            if d.name in seen:
                raise ValueError(f"duplicate CDI device name {d.name!r}")
            seen.add(d.name)
            if not d.device_nodes:
                raise ValueError(f"CDI device {d.name!r} has no device nodes")


def build_spec(inv: NodeInventory, kind: str, dev_root: str = "/dev",
               cdi_version: str = CDI_VERSION) -> CDISpec:
    """NodeInventory → CDISpec. Reference analog: generateCDISpec
    (`device_plugin.go:55-80`), with per-group devices instead of
    per-function."""
    spec = CDISpec(kind=kind, cdi_version=cdi_version)
    for gid in inv.device_ids():
        dev = inv.devices[gid]
        spec.devices.append(device_entry(dev, kind, dev_root))
    spec.validate()
    return spec


def device_entry(dev: XPUDevice, kind: str, dev_root: str = "/dev") -> CDIDeviceEntry:
    ann = {
        ANNOTATION_ATTACH_PCI: "true",
        ANNOTATION_BDF: ",".join(dev.bdfs),
        f"{ANNOTATION_PREFIX}vfio{dev.id}": qualified_name(kind, dev.id),
    }
    return CDIDeviceEntry(
        name=dev.id,
        annotations=ann,
        device_nodes=[os.path.join(dev_root, dev.vfio_node)],
    )


def spec_path(cdi_dir: str, spec_name: str, fmt: str) -> str:
    ext = "yaml" if fmt == "yaml" else "json"
    return os.path.join(cdi_dir, f"{spec_name}.{ext}")


def write_spec(spec: CDISpec, cdi_dir: str, spec_name: str, fmt: str = "yaml") -> str:
    """Atomically write the spec file; returns its path.

    Reference analog: (*CdiSpec).Save (`cdi/spec.go:85-127`) — which wrote
    non-atomically with a fixed file name `cdi-vfio-xxxx.yaml`
    (`device_plugin.go:79`).
    """
    spec.validate()
    os.makedirs(cdi_dir, exist_ok=True)
    path = spec_path(cdi_dir, spec_name, fmt)
    obj = spec.to_obj()
    # Hard contract check: nothing schema-invalid ever reaches /var/run/cdi
    # (the runtime would reject or — worse — misparse it).
    from .schema import validate_spec_obj
    problems = validate_spec_obj(obj)
    if problems:
        raise ValueError(f"CDI spec fails schema validation: {problems}")
    if fmt == "yaml":
        payload = yaml.safe_dump(obj, sort_keys=False)
    else:
        payload = json.dumps(obj, indent=2) + "\n"
    fd, tmp = tempfile.mkstemp(dir=cdi_dir, prefix=f".{spec_name}.")
    try:
        with os.fdopen(fd, "w") as f:
            f.write(payload)
        os.chmod(tmp, 0o644)
        os.replace(tmp, path)
    except BaseException:
        try:
            os.unlink(tmp)
        except OSError:
            pass
        raise
    # Drop a stale spec of the other format so the runtime never resolves
    # against an outdated file.
    other = spec_path(cdi_dir, spec_name, "json" if fmt == "yaml" else "yaml")
    if os.path.exists(other):
        try:
            os.unlink(other)
        except OSError as e:
            log.warning("could not remove stale CDI spec %s: %s", other, e)
    log.info("wrote CDI spec %s (%d devices)", path, len(spec.devices))
    return path


def read_spec(path: str) -> CDISpec:
    """Load a spec file back (used by tests and the allocate validator)."""
    with open(path) as f:
        obj = yaml.safe_load(f) if path.endswith((".yaml", ".yml")) else json.load(f)
    spec = CDISpec(kind=obj["kind"], cdi_version=obj.get("cdiVersion", CDI_VERSION))
    for d in obj.get("devices", []):
        spec.devices.append(
            CDIDeviceEntry(
                name=d["name"],
                annotations=dict(d.get("annotations", {})),
                device_nodes=[n["path"] for n in d.get("containerEdits", {}).get("deviceNodes", [])],
            )
        )
    return spec
