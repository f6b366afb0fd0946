"""xGMI topology + placement tests (reference gap: GetPreferredAllocation
is a stub at generic_device_plugin.go:378-386; BASELINE.json configs #3/#4
demand hive-aware 8 and 4+4 placement)."""
import json

from kata_xpu_device_plugin_amd.discovery import scan_node
from kata_xpu_device_plugin_amd.testing.mocknode import make_mock_node
from kata_xpu_device_plugin_amd.topology import (
    load_topology,
    preferred_sets,
    read_kfd_topology,
    score_set,
)
from kata_xpu_device_plugin_amd.topology.hive import (
    W_NUMA,
    W_XGMI,
    preferred_allocation,
    topology_from_hint,
    topology_from_kfd,
)


def _topo_and_inv(tmp_path, **kw):
    node = make_mock_node(str(tmp_path), **kw)
    cfg = node.config()
    inv = scan_node(cfg)
    return node, cfg, inv


def test_kfd_parse(tmp_path):
    node, cfg, inv = _topo_and_inv(tmp_path, n_gpus=4)
    nodes = read_kfd_topology(cfg.sysfs_root)
    gpus = [n for n in nodes if n.is_gpu]
    assert len(gpus) == 4
    assert gpus[0].bdf == "0000:0a:00.0"
    assert all(n.hive_id == 1 for n in gpus)
    # 3 xGMI peers + 1 PCIe link each
    assert all(len(n.xgmi_links) == 3 for n in gpus)
    assert gpus[0].xgmi_links[0].max_bandwidth_mbps == 153000


def test_topology_from_kfd(tmp_path):
    node, cfg, inv = _topo_and_inv(tmp_path, n_gpus=8)
    topo = topology_from_kfd(cfg.sysfs_root)
    assert topo is not None and topo.source == "kfd"
    assert len(set(topo.hive_of.values())) == 1
    assert abs(topo.xgmi_gbps - 153.0) < 1e-6


def test_topology_hint_fallback(tmp_path):
    """vfio-bound GPUs are invisible to KFD → hint file must be used."""
    node, cfg, inv = _topo_and_inv(tmp_path, n_gpus=8, kfd=False)
    topo = load_topology(cfg, inv)
    assert topo.source == "hint"
    assert len(topo.hive_of) == 8
    # NUMA mapping filled from PCI sysfs in all cases
    assert topo.numa_of["0000:0a:00.0"] == 0
    assert topo.numa_of["0000:2a:00.0"] == 1


def test_topology_none_fallback(tmp_path):
    node, cfg, inv = _topo_and_inv(tmp_path, n_gpus=4, kfd=False, hint=False)
    topo = load_topology(cfg, inv)
    assert topo.source == "none"
    assert topo.hive_of == {}
    assert len(topo.numa_of) == 4


def test_score_set_tiers(tmp_path):
    node, cfg, inv = _topo_and_inv(tmp_path, n_gpus=8, hives=[[0, 1, 2, 3], [4, 5, 6, 7]])
    topo = load_topology(cfg, inv)
    bdfs = [inv.devices[str(70 + i)].primary.bdf for i in range(8)]
    # same hive pair ≫ cross-hive same-numa pair
    assert score_set(topo, [bdfs[0], bdfs[1]]) >= W_XGMI
    cross = score_set(topo, [bdfs[0], bdfs[4]])
    assert cross < W_XGMI


def test_preferred_whole_hive(tmp_path):
    """4+4 hives: a 4-GPU pod must land entirely on one hive."""
    node, cfg, inv = _topo_and_inv(tmp_path, n_gpus=8, hives=[[0, 1, 2, 3], [4, 5, 6, 7]])
    topo = load_topology(cfg, inv)
    ids = inv.device_ids()
    pick = preferred_allocation(topo, inv, ids, [], 4)
    assert len(pick) == 4
    hives = {topo.hive(inv.devices[d].primary.bdf) for d in pick}
    assert len(hives) == 1


def test_preferred_best_fit_leaves_hive_intact(tmp_path):
    """With hives of size 2 and 4 free, a 2-GPU pod takes the 2-hive,
    leaving the 4-hive intact for a later 4-GPU pod."""
    node, cfg, inv = _topo_and_inv(
        tmp_path, n_gpus=6, hives=[[0, 1], [2, 3, 4, 5]]
    )
    topo = load_topology(cfg, inv)
    ids = inv.device_ids()
    pick = preferred_sets(
        topo, {g: inv.devices[g].primary.bdf for g in ids}, ids, [], 2
    )
    bdfs = {inv.devices[d].primary.bdf for d in pick}
    assert {topo.hive(b) for b in bdfs} == {"hive-1"}


def test_preferred_must_include(tmp_path):
    node, cfg, inv = _topo_and_inv(tmp_path, n_gpus=8, hives=[[0, 1, 2, 3], [4, 5, 6, 7]])
    topo = load_topology(cfg, inv)
    ids = inv.device_ids()
    # force one device from hive 2 → the rest must come from hive 2
    pick = preferred_allocation(topo, inv, ids, ["74"], 4)
    assert "74" in pick and len(pick) == 4
    hives = {topo.hive(inv.devices[d].primary.bdf) for d in pick}
    assert hives == {"hive-2"}


def test_preferred_unsatisfiable(tmp_path):
    node, cfg, inv = _topo_and_inv(tmp_path, n_gpus=2)
    topo = load_topology(cfg, inv)
    ids = inv.device_ids()
    assert preferred_allocation(topo, inv, ids, [], 3) == []
    assert preferred_allocation(topo, inv, ids, ["not-a-device"], 1) == []
    assert preferred_allocation(topo, inv, ids, [], 0) == []


def test_preferred_numa_fallback(tmp_path):
    """No hive info at all → NUMA co-location decides."""
    node, cfg, inv = _topo_and_inv(tmp_path, n_gpus=8, kfd=False, hint=False)
    topo = load_topology(cfg, inv)
    ids = inv.device_ids()
    pick = preferred_allocation(topo, inv, ids, [], 4)
    assert len(pick) == 4
    numas = {inv.devices[d].numa_node for d in pick}
    assert len(numas) == 1


def test_preferred_full_node(tmp_path):
    node, cfg, inv = _topo_and_inv(tmp_path, n_gpus=8)
    topo = load_topology(cfg, inv)
    ids = inv.device_ids()
    pick = preferred_allocation(topo, inv, ids, [], 8)
    assert sorted(pick) == sorted(ids)


def test_hint_file_parsing(tmp_path):
    p = tmp_path / "topo.json"
    p.write_text(json.dumps({
        "version": 1,
        "hives": [["0000:0a:00.0", "0000:12:00.0"], ["0000:1a:00.0"]],
        "xgmi_link_gbps": 153.0,
    }))
    topo = topology_from_hint(str(p))
    assert topo.source == "hint"
    assert topo.hive("0000:0a:00.0") == topo.hive("0000:12:00.0") == "hive-1"
    assert topo.hive("0000:1a:00.0") == "hive-2"
    assert topology_from_hint(str(tmp_path / "missing.json")) is None


def test_vf_inherits_pf_hive(tmp_path):
    """SR-IOV VFs share their PF's fabric: VFs of a hive-mapped PF join
    that hive; VFs of an unmapped PF form a per-PF pseudo-hive, so a
    multi-VF pod still prefers one physical GPU."""
    from kata_xpu_device_plugin_amd.testing.mocknode import MockGPU

    node = make_mock_node(str(tmp_path), n_gpus=2, driver="vfio-pci")
    # PF0 (in hive via hint) with 2 VFs; PF1 not hive-mapped with 2 VFs
    for k in range(2):
        node.add_gpu(MockGPU(bdf=f"0000:0a:02.{k}", device_id=0x75B3,
                             iommu_group=str(120 + k), physfn_bdf="0000:0a:00.0"))
        node.add_gpu(MockGPU(bdf=f"0000:66:02.{k}", device_id=0x75B3,
                             iommu_group=str(130 + k), physfn_bdf="0000:66:00.0"))
    cfg = node.config()
    inv = scan_node(cfg)
    topo = load_topology(cfg, inv)
    # PF 0000:0a:00.0 is in the hint's hive-1 → its VFs inherit it
    pf_hive = topo.hive("0000:0a:00.0")
    assert pf_hive
    assert topo.hive("0000:0a:02.0") == topo.hive("0000:0a:02.1") == pf_hive
    # unmapped PF → pseudo-hive shared by its VFs only
    assert topo.hive("0000:66:02.0") == topo.hive("0000:66:02.1") == \
        "pf-0000:66:00.0"
    # placement: a 2-VF pod sticks to one PF
    ids = ["120", "121", "130", "131"]
    pick = preferred_allocation(topo, inv, ids, [], 2)
    hives = {topo.hive(inv.devices[d].primary.bdf) for d in pick}
    assert len(hives) == 1


def test_degraded_must_include_still_honored():
    """kubelet may force (must_include) a degraded-fabric GPU; the
    selector honors it and packs the remainder for locality."""
    from kata_xpu_device_plugin_amd.topology.hive import (
        GPUTopology, preferred_sets, score_set)
    topo = GPUTopology(source="hint")
    bdf_of = {}
    for i in range(6):
        bdf = f"0000:{10+i:02x}:00.0"
        bdf_of[str(i)] = bdf
        topo.hive_of[bdf] = "hive-1" if i < 3 else "hive-2"
    topo.set_degraded([bdf_of["0"]])
    pick = preferred_sets(topo, bdf_of, list(bdf_of), ["0"], 3)
    assert "0" in pick and len(pick) == 3
    others = [d for d in pick if d != "0"]
    hives = {topo.hive(bdf_of[d]) for d in others}
    assert len(hives) == 1 and hives != {""}, \
        "remaining picks must share one intact hive"
    # degradation is reversible
    topo.set_degraded([])
    assert topo.hive(bdf_of["0"]) == "hive-1"
