"""xGMI hive model + placement scoring → real GetPreferredAllocation.

The reference stubs GetPreferredAllocation (`generic_device_plugin.go:378-386`
returns nil, nil, and does not even advertise it in
GetDevicePluginOptions). On MI355X this is where topology knowledge pays:
the OAM fabric is point-to-point xGMI (7 links × ≈153 GB/s per GPU inside a
hive); ring collectives inside a pod are per-link bound, so a pod whose GPUs
span two hives collapses to PCIe bandwidth. Placement policy:

1. maximize pairwise xGMI connectivity inside the allocation,
2. among equally-connected choices, prefer NUMA co-location,
3. **best-fit packing**: take devices from the most-depleted hive first so
   future large pods still find an intact hive (the 4+4 split of
   BASELINE.json config #4 must leave each pod on one fabric).

Topology sources, in priority order (`load_topology`):
* live KFD (amdgpu-bound GPUs — pre-passthrough or health-probe nodes),
* a snapshot hint JSON written by ``python -m
  kata_xpu_device_plugin_amd.tools.topo snapshot`` while GPUs were
  amdgpu-bound (vfio-bound GPUs are invisible to KFD),
* none — fall back to NUMA/PCI-locality-only scoring from sysfs attributes
  that remain readable under vfio-pci (numa_node, bus topology).

Within one hive MI355X GPUs are all-to-all connected and interchangeable,
so optimal selection reduces to choosing **how many devices to take from
each (hive, numa) bucket** — enumerated exactly (compositions of k over
buckets) instead of over C(n, k) subsets. A native C++ twin of this
selection lives in native/selector.cpp for the hot path.
"""
from __future__ import annotations

import itertools
import json
import os
from dataclasses import dataclass, field
from typing import Dict, Iterable, List, Optional, Sequence, Tuple

from ..config import Config
from ..discovery.sysfs import NodeInventory
from ..utils.log import get_logger
from .kfd import read_kfd_topology

log = get_logger(__name__)

# Pairwise locality weights. Orders of magnitude apart so tiers never trade
# against each other: one xGMI pair beats any number of NUMA pairs.
W_XGMI = 1_000_000
W_NUMA = 1_000
W_SAME_BUS_PFX = 1  # same PCIe segment/root-ish: bus number high nibble


@dataclass
class GPUTopology:
    hive_of: Dict[str, str] = field(default_factory=dict)   # bdf → hive key
    numa_of: Dict[str, int] = field(default_factory=dict)   # bdf → numa node
    xgmi_gbps: float = 0.0
    source: str = "none"   # kfd | hint | none

    def hive(self, bdf: str) -> str:
        return self.hive_of.get(bdf, "")


def topology_from_kfd(sysfs_root: str) -> Optional[GPUTopology]:
    nodes = [n for n in read_kfd_topology(sysfs_root) if n.is_gpu and n.bdf]
    if not nodes:
        return None
    topo = GPUTopology(source="kfd")
    max_bw = 0
    for n in nodes:
        if n.hive_id:
            topo.hive_of[n.bdf] = f"hive-{n.hive_id}"
        for l in n.xgmi_links:
            max_bw = max(max_bw, l.max_bandwidth_mbps)
    topo.xgmi_gbps = max_bw / 1000.0
    return topo


def topology_from_hint(path: str) -> Optional[GPUTopology]:
    try:
        with open(path) as f:
            doc = json.load(f)
    except (OSError, ValueError) as e:
        if os.path.exists(path):
            log.warning("unreadable topology hint %s: %s", path, e)
        return None
    topo = GPUTopology(source="hint", xgmi_gbps=float(doc.get("xgmi_link_gbps", 0.0)))
    for i, members in enumerate(doc.get("hives", []), start=1):
        for bdf in members:
            topo.hive_of[bdf] = f"hive-{i}"
    return topo


def load_topology(cfg: Config, inv: NodeInventory) -> GPUTopology:
    topo = topology_from_kfd(cfg.sysfs_root)
    if topo is None:
        topo = topology_from_hint(cfg.topology_hint_path)
    if topo is None:
        topo = GPUTopology(source="none")
        log.info("no xGMI topology source (KFD empty, no hint file); "
                 "placement falls back to NUMA/PCI locality")
    # NUMA comes from PCI sysfs — readable regardless of bound driver.
    for dev in inv.devices.values():
        topo.numa_of[dev.primary.bdf] = dev.numa_node
    return topo


# ---------------------------------------------------------------------------
# Scoring
# ---------------------------------------------------------------------------

def _pair_weight(topo: GPUTopology, a: str, b: str) -> int:
    ha, hb = topo.hive(a), topo.hive(b)
    if ha and ha == hb:
        return W_XGMI
    w = 0
    if topo.numa_of.get(a, -1) == topo.numa_of.get(b, -2):
        w += W_NUMA
    if a.split(":")[0] == b.split(":")[0] and a[5] == b[5]:
        w += W_SAME_BUS_PFX
    return w


def score_set(topo: GPUTopology, bdfs: Sequence[str]) -> int:
    """Pairwise locality score of an allocation (higher = better)."""
    return sum(
        _pair_weight(topo, a, b) for a, b in itertools.combinations(sorted(set(bdfs)), 2)
    )


# ---------------------------------------------------------------------------
# Preferred-set selection
# ---------------------------------------------------------------------------

def _buckets(
    topo: GPUTopology, bdf_of: Dict[str, str], ids: Iterable[str]
) -> Dict[Tuple[str, int], List[str]]:
    """Group device IDs by (hive key, numa). '' hive = no fabric info."""
    out: Dict[Tuple[str, int], List[str]] = {}
    for did in ids:
        bdf = bdf_of[did]
        key = (topo.hive(bdf), topo.numa_of.get(bdf, -1))
        out.setdefault(key, []).append(did)
    for v in out.values():
        v.sort(key=lambda d: (len(d), d))
    return out


def preferred_sets(
    topo: GPUTopology,
    bdf_of: Dict[str, str],
    available: Sequence[str],
    must_include: Sequence[str],
    size: int,
    use_native: bool = True,
) -> List[str]:
    """Choose `size` device IDs from `available` (⊇ must_include).

    Exact over (hive, numa) bucket compositions; within a bucket devices are
    interchangeable, so this is optimal for the pairwise score above.
    Returns [] if the request is unsatisfiable (kubelet then falls back to
    its own pick — Allocate still validates whatever arrives).
    """
    avail = list(dict.fromkeys(available))
    must = list(dict.fromkeys(must_include))
    if size <= 0 or size > len(avail):
        return []
    for m in must:
        if m not in avail:
            return []
    if len(must) >= size:
        return must[:size]

    try:
        from .. import _native
    except ImportError:
        _native = None
    if use_native and _native is not None:
        locality = {
            d: (topo.hive(bdf_of[d]), topo.numa_of.get(bdf_of[d], -1))
            for d in avail
        }
        return _native.select_preferred(locality, avail, must, size)

    remaining_ids = [d for d in avail if d not in set(must)]
    need = size - len(must)
    buckets = _buckets(topo, bdf_of, remaining_ids)
    keys = sorted(buckets, key=lambda k: (len(buckets[k]), k))  # best-fit: small first
    caps = [len(buckets[k]) for k in keys]

    # Forced members contribute fixed pair terms with each candidate bucket;
    # precompute per-bucket affinity to the must-set.
    must_bdfs = [bdf_of[d] for d in must]

    def bucket_affinity(key: Tuple[str, int]) -> int:
        hive, numa = key
        aff = 0
        for mb in must_bdfs:
            if hive and topo.hive(mb) == hive:
                aff += W_XGMI
            elif topo.numa_of.get(mb, -1) == numa and numa != -1:
                aff += W_NUMA
        return aff

    affinities = [bucket_affinity(k) for k in keys]

    # Packing tie-break: among equal-locality choices prefer the set that
    # leaves the least fragmentation — deplete small/partial hives first so
    # a later large pod still finds an intact hive (config #4's 4+4 split).
    def packing_key(key: Tuple[str, int]) -> Tuple[str, int]:
        hive, numa = key
        return (hive, -1) if hive else ("numa", numa)

    hive_free: Dict[Tuple[str, int], int] = {}
    for k, cap in zip(keys, caps):
        hive_free[packing_key(k)] = hive_free.get(packing_key(k), 0) + cap

    def packing(take: Sequence[int]) -> int:
        taken: Dict[Tuple[str, int], int] = {}
        for k, c in zip(keys, take):
            if c:
                taken[packing_key(k)] = taken.get(packing_key(k), 0) + c
        # negative leftover across touched hives; 0 is best (hive depleted)
        return -sum(hive_free[h] - c for h, c in taken.items())

    best_score = (-1, 0)
    best_take: Optional[Tuple[int, ...]] = None

    def bucket_pair_score(key: Tuple[str, int], c: int) -> int:
        hive, numa = key
        w = W_XGMI if hive else (W_NUMA if numa != -1 else 0)
        return w * c * (c - 1) // 2

    def cross_score(take: Sequence[int]) -> int:
        # cross-bucket terms: same hive different numa → W_XGMI; different
        # hive same numa → W_NUMA.
        s = 0
        for i in range(len(take)):
            if not take[i]:
                continue
            hi, ni = keys[i]
            for j in range(i + 1, len(take)):
                if not take[j]:
                    continue
                hj, nj = keys[j]
                if hi and hi == hj:
                    s += W_XGMI * take[i] * take[j]
                elif ni == nj and ni != -1:
                    s += W_NUMA * take[i] * take[j]
        return s

    # Depth-first enumeration of compositions with pruning by remaining cap.
    n_b = len(keys)
    suffix_cap = [0] * (n_b + 1)
    for i in range(n_b - 1, -1, -1):
        suffix_cap[i] = suffix_cap[i + 1] + caps[i]

    take = [0] * n_b

    def dfs(i: int, left: int, acc: int):
        nonlocal best_score, best_take
        if left == 0:
            total = (acc + cross_score(take), packing(take))
            if total > best_score:
                best_score = total
                best_take = tuple(take)
            return
        if i >= n_b or suffix_cap[i] < left:
            return
        for c in range(min(caps[i], left), -1, -1):
            take[i] = c
            dfs(i + 1, left - c, acc + bucket_pair_score(keys[i], c) + affinities[i] * c)
        take[i] = 0

    dfs(0, need, 0)
    if best_take is None:
        return []

    chosen = list(must)
    for k, c in zip(keys, best_take):
        chosen.extend(buckets[k][:c])
    return chosen


def preferred_allocation(
    topo: GPUTopology,
    inv: NodeInventory,
    available: Sequence[str],
    must_include: Sequence[str],
    size: int,
) -> List[str]:
    """Device-ID (IOMMU-group) level entry point used by the gRPC server."""
    bdf_of = {gid: dev.primary.bdf for gid, dev in inv.devices.items()}
    usable = [d for d in available if d in bdf_of]
    if len(usable) < len(list(available)):
        missing = set(available) - set(usable)
        log.warning("preferred allocation: unknown device ids %s", sorted(missing))
    return preferred_sets(topo, bdf_of, usable, must_include, size)
