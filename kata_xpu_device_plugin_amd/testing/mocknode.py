"""Mock MI355X node: fake sysfs + /dev tree for tests and the benchmark.

SURVEY.md §4 test plan item 1 and BASELINE.json config #1 call for a mock
sysfs tree with fake 0x1002 BDFs plus xGMI topology files. This builder
creates, under a temp root:

* ``sys/bus/pci/devices/<bdf>/{vendor,device,class,numa_node,...}`` with
  ``driver`` / ``iommu_group`` / ``physfn`` symlinks,
* ``sys/kernel/iommu_groups/<g>/devices/<bdf>`` back-links,
* ``dev/vfio/<g>`` group nodes (plain files — the plugin only stats them),
* ``sys/class/kfd/kfd/topology/nodes/<n>/{properties,io_links/*/properties}``
  describing the xGMI hive (type 11 links, per-KFD-convention), used when
  GPUs are amdgpu-bound,
* optionally a topology hint JSON (for the vfio-bound case where KFD can't
  see the GPUs).

Default shape: one 8×MI355X OAM node, all 8 GPUs in one xGMI hive with
7 p2p links each (the MI355X fabric), one GPU per IOMMU group.
"""
from __future__ import annotations

import json
import tempfile
import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

from ..config import Config

MI355X_DEVICE_ID = 0x75A3
GPU_CLASS = 0x120000       # Processing accelerator (real MI355X class, live-node confirmed)
AUDIO_CLASS = 0x040300     # Audio device (companion function example)


@dataclass
class MockGPU:
    bdf: str
    device_id: int = MI355X_DEVICE_ID
    iommu_group: str = ""
    driver: str = "vfio-pci"
    numa_node: int = 0
    hive_id: int = 1
    physfn_bdf: Optional[str] = None   # set ⇒ VF
    sriov_totalvfs: int = 0
    class_code: int = GPU_CLASS
    # amdgpu partitioning sysfs (MI300/MI355 PFs; "" = files absent)
    compute_partition: str = ""        # e.g. "SPX"
    compute_available: str = ""        # e.g. "SPX, DPX, QPX, CPX"
    memory_partition: str = ""         # e.g. "NPS1"
    memory_available: str = ""         # e.g. "NPS1, NPS4"


def default_bdfs(n: int) -> List[str]:
    # Two GPUs per NUMA-ish root port, separate buses like a real OAM board.
    return [f"0000:{0x0a + 8 * i:02x}:00.0" for i in range(n)]


@dataclass
class MockNode:
    root: str
    gpus: List[MockGPU] = field(default_factory=list)
    cfg: Config = None  # type: ignore[assignment]

    # ------------------------------------------------------------------
    @property
    def sysfs(self) -> str:
        return os.path.join(self.root, "sys")

    @property
    def dev(self) -> str:
        return os.path.join(self.root, "dev")

    def config(self, **overrides) -> Config:
        # Unix socket paths are capped at ~107 chars; pytest tmp roots are
        # deep, so sockets live in a short mkdtemp instead of under root.
        sock_dir = tempfile.mkdtemp(prefix="kxdp-")
        cfg = Config(
            sysfs_root=self.sysfs,
            dev_root=self.dev,
            cdi_dir=os.path.join(self.root, "var", "run", "cdi"),
            kubelet_socket_dir=sock_dir,
            topology_hint_path=os.path.join(self.root, "etc", "topology.json"),
            pci_ids_paths=(),
            metrics_port=0,
        )
        for k, v in overrides.items():
            setattr(cfg, k, v)
        cfg.validate()
        os.makedirs(cfg.cdi_dir, exist_ok=True)
        os.makedirs(cfg.kubelet_socket_dir, exist_ok=True)
        return cfg

    # ------------------------------------------------------------------
    def _pci_dir(self, bdf: str) -> str:
        return os.path.join(self.sysfs, "bus", "pci", "devices", bdf)

    def _write(self, path: str, content: str) -> None:
        os.makedirs(os.path.dirname(path), exist_ok=True)
        with open(path, "w") as f:
            f.write(content)

    def _symlink(self, target: str, link: str) -> None:
        os.makedirs(os.path.dirname(link), exist_ok=True)
        if os.path.islink(link):
            os.unlink(link)
        os.makedirs(target, exist_ok=True)
        os.symlink(target, link)

    def add_gpu(self, gpu: MockGPU, with_audio_fn: bool = False) -> None:
        d = self._pci_dir(gpu.bdf)
        self._write(os.path.join(d, "vendor"), "0x1002\n")
        self._write(os.path.join(d, "device"), f"0x{gpu.device_id:04x}\n")
        self._write(os.path.join(d, "class"), f"0x{gpu.class_code:06x}\n")
        self._write(os.path.join(d, "numa_node"), f"{gpu.numa_node}\n")
        if gpu.sriov_totalvfs:
            self._write(os.path.join(d, "sriov_totalvfs"), f"{gpu.sriov_totalvfs}\n")
            self._write(os.path.join(d, "sriov_numvfs"), "0\n")
        if gpu.compute_partition:
            self._write(os.path.join(d, "current_compute_partition"),
                        f"{gpu.compute_partition}\n")
            self._write(os.path.join(d, "available_compute_partition"),
                        f"{gpu.compute_available or gpu.compute_partition}\n")
        if gpu.memory_partition:
            self._write(os.path.join(d, "current_memory_partition"),
                        f"{gpu.memory_partition}\n")
            self._write(os.path.join(d, "available_memory_partition"),
                        f"{gpu.memory_available or gpu.memory_partition}\n")
        if gpu.driver:
            self._symlink(
                os.path.join(self.sysfs, "bus", "pci", "drivers", gpu.driver),
                os.path.join(d, "driver"),
            )
        if gpu.iommu_group:
            gdir = os.path.join(self.sysfs, "kernel", "iommu_groups", gpu.iommu_group)
            self._symlink(gdir, os.path.join(d, "iommu_group"))
            os.makedirs(os.path.join(gdir, "devices"), exist_ok=True)
            self._write(os.path.join(gdir, "devices", gpu.bdf), "")
            self.add_vfio_node(gpu.iommu_group)
        if gpu.physfn_bdf:
            self._symlink(self._pci_dir(gpu.physfn_bdf), os.path.join(d, "physfn"))
        if with_audio_fn:
            abdf = gpu.bdf[:-1] + "1"
            ad = self._pci_dir(abdf)
            self._write(os.path.join(ad, "vendor"), "0x1002\n")
            self._write(os.path.join(ad, "device"), "0xab30\n")
            self._write(os.path.join(ad, "class"), f"0x{AUDIO_CLASS:06x}\n")
            self._write(os.path.join(ad, "numa_node"), f"{gpu.numa_node}\n")
            if gpu.driver:
                self._symlink(
                    os.path.join(self.sysfs, "bus", "pci", "drivers", gpu.driver),
                    os.path.join(ad, "driver"),
                )
            if gpu.iommu_group:
                self._symlink(
                    os.path.join(self.sysfs, "kernel", "iommu_groups", gpu.iommu_group),
                    os.path.join(ad, "iommu_group"),
                )
        self.gpus.append(gpu)

    def remove_gpu(self, bdf: str) -> None:
        """Inverse of add_gpu: delete the PCI function (and its IOMMU-group
        membership + vfio node when it was the last member) — models VF
        disable / device unplug for rescan tests."""
        import shutil
        gpu = next((g for g in self.gpus if g.bdf == bdf), None)
        d = self._pci_dir(bdf)
        if os.path.isdir(d):
            shutil.rmtree(d)
        if gpu is not None:
            self.gpus.remove(gpu)
            if gpu.iommu_group:
                gdir = os.path.join(self.sysfs, "kernel", "iommu_groups",
                                    gpu.iommu_group)
                member = os.path.join(gdir, "devices", bdf)
                if os.path.exists(member):
                    os.unlink(member)
                remaining = os.listdir(os.path.join(gdir, "devices")) \
                    if os.path.isdir(os.path.join(gdir, "devices")) else []
                if not remaining:
                    self.remove_vfio_node(gpu.iommu_group)

    # --- /dev/vfio ----------------------------------------------------
    def add_vfio_node(self, group: str) -> str:
        path = os.path.join(self.dev, "vfio", group)
        self._write(path, "")
        return path

    def remove_vfio_node(self, group: str) -> None:
        path = os.path.join(self.dev, "vfio", group)
        if os.path.exists(path):
            os.unlink(path)

    # --- KFD topology (amdgpu-bound view) -----------------------------
    def write_kfd_topology(self, xgmi_gbps: float = 153.0) -> None:
        """Emit KFD topology nodes for the GPUs: node 0 = CPU, then GPUs.

        Mirrors the live layout of /sys/class/kfd/kfd/topology/nodes/*:
        `properties` has simd_count/location_id/domain/hive_id; io_links/*/
        properties has type (11 = xGMI), node_to, min/max_bandwidth (MB/s).
        """
        base = os.path.join(self.sysfs, "class", "kfd", "kfd", "topology", "nodes")
        # CPU node (simd_count 0)
        self._write(
            os.path.join(base, "0", "properties"),
            "cpu_cores_count 128\nsimd_count 0\nlocation_id 0\ndomain 0\nhive_id 0\n",
        )
        bw_mbps = int(xgmi_gbps * 1000)
        for i, gpu in enumerate(self.gpus):
            if gpu.physfn_bdf:
                continue
            node = i + 1
            dom, bus, devfn = _parse_bdf(gpu.bdf)
            loc = (bus << 8) | devfn
            props = (
                f"cpu_cores_count 0\nsimd_count 1024\n"
                f"location_id {loc}\ndomain {dom}\nhive_id {gpu.hive_id}\n"
                f"gfx_target_version 90500\n"
            )
            self._write(os.path.join(base, str(node), "properties"), props)
            li = 0
            for j, peer in enumerate(self.gpus):
                if peer.physfn_bdf or j == i:
                    continue
                peer_node = j + 1
                if peer.hive_id == gpu.hive_id and gpu.hive_id != 0:
                    self._write(
                        os.path.join(base, str(node), "io_links", str(li), "properties"),
                        f"type 11\nversion_major 0\nnode_from {node}\n"
                        f"node_to {peer_node}\nweight 15\n"
                        f"min_bandwidth {bw_mbps}\nmax_bandwidth {bw_mbps}\n",
                    )
                    li += 1
            # PCIe link to CPU
            self._write(
                os.path.join(base, str(node), "io_links", str(li), "properties"),
                f"type 2\nversion_major 0\nnode_from {node}\nnode_to 0\nweight 20\n"
                f"min_bandwidth 63000\nmax_bandwidth 63000\n",
            )

    # --- topology hint (vfio-bound view) ------------------------------
    def write_topology_hint(self, path: Optional[str] = None) -> str:
        path = path or os.path.join(self.root, "etc", "topology.json")
        hives: Dict[int, List[str]] = {}
        for g in self.gpus:
            if g.physfn_bdf:
                continue
            hives.setdefault(g.hive_id, []).append(g.bdf)
        doc = {
            "version": 1,
            "hives": [sorted(v) for k, v in sorted(hives.items()) if k != 0],
            "xgmi_link_gbps": 153.0,
        }
        self._write(path, json.dumps(doc, indent=2))
        return path


def make_mock_node(
    root: str,
    n_gpus: int = 8,
    hives: Optional[Sequence[Sequence[int]]] = None,
    driver: str = "vfio-pci",
    with_audio_fn: bool = False,
    kfd: bool = True,
    hint: bool = True,
) -> MockNode:
    """Build the default 8×MI355X mock node.

    hives: list of GPU-index groups sharing an xGMI hive; default all in one.
    """
    node = MockNode(root=root)
    bdfs = default_bdfs(n_gpus)
    hive_of = {}
    if hives is None:
        hives = [list(range(n_gpus))]
    for hid, members in enumerate(hives, start=1):
        for m in members:
            hive_of[m] = hid
    for i, bdf in enumerate(bdfs):
        node.add_gpu(
            MockGPU(
                bdf=bdf,
                iommu_group=str(70 + i),
                driver=driver,
                numa_node=i // max(1, n_gpus // 2),
                hive_id=hive_of.get(i, 0),
            ),
            with_audio_fn=with_audio_fn,
        )
    if kfd:
        node.write_kfd_topology()
    if hint:
        node.write_topology_hint()
    return node


def _parse_bdf(bdf: str):
    dom, bus, devfn_s = bdf.split(":")
    dev, fn = devfn_s.split(".")
    return int(dom, 16), int(bus, 16), (int(dev, 16) << 3) | int(fn, 16)
