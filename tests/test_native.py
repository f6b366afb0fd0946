"""Native C++ extension tests: scan parity, revalidation, selector parity.

The native layer replaces the reference's (Go-native) hot paths; the
Python implementations are the executable spec, so every native function
is tested for exact agreement with its Python twin.
"""
import random

import pytest

from kata_xpu_device_plugin_amd.discovery import scan_node
from kata_xpu_device_plugin_amd.discovery.sysfs import _scan_functions_py, scan_functions
from kata_xpu_device_plugin_amd.testing.mocknode import make_mock_node
from kata_xpu_device_plugin_amd.topology.hive import (
    GPUTopology,
    load_topology,
    preferred_sets,
    score_set,
)

_native = pytest.importorskip("kata_xpu_device_plugin_amd._native")


def test_scan_parity(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=8, with_audio_fn=True,
                          kfd=False, hint=False)
    cfg = node.config()
    cfg.native = "off"
    py = _scan_functions_py(cfg)
    cfg.native = "require"
    nat = scan_functions(cfg)
    assert [f.__dict__ for f in nat] == [f.__dict__ for f in py]
    assert len(nat) == 16  # 8 GPUs + 8 audio functions


def test_scan_node_with_native(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=4, kfd=False, hint=False)
    cfg = node.config(native="require")
    inv = scan_node(cfg)
    assert len(inv.devices) == 4


def test_revalidate_group(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=2, kfd=False, hint=False)
    cfg = node.config()
    devices_dir = f"{cfg.sysfs_root}/bus/pci/devices"
    ok = _native.revalidate_group(devices_dir, "70", ["0000:0a:00.0"],
                                  [0x1002], "vfio-pci")
    assert ok == ""
    bad = _native.revalidate_group(devices_dir, "71", ["0000:0a:00.0"],
                                   [0x1002], "vfio-pci")
    assert "no longer in IOMMU group" in bad
    bad = _native.revalidate_group(devices_dir, "70", ["0000:0a:00.0"],
                                   [0x1002], "amdgpu")
    assert "not bound to" in bad
    bad = _native.revalidate_group(devices_dir, "70", ["0000:ff:00.0"],
                                   [0x1002], "vfio-pci")
    assert "vanished" in bad


def _random_topo(rng, n):
    """Random hive/numa layout over n devices."""
    topo = GPUTopology(source="hint")
    bdf_of = {}
    n_hives = rng.randint(1, 3)
    for i in range(n):
        bdf = f"0000:{10 + i:02x}:00.0"
        did = str(70 + i)
        bdf_of[did] = bdf
        h = rng.randint(0, n_hives)
        if h:
            topo.hive_of[bdf] = f"hive-{h}"
        topo.numa_of[bdf] = rng.randint(0, 1)
    return topo, bdf_of


def test_selector_parity_randomized():
    """Native and Python selectors must pick sets with identical scores
    (ties may break differently only between equally-scored sets)."""
    rng = random.Random(1234)
    for trial in range(200):
        n = rng.randint(2, 10)
        topo, bdf_of = _random_topo(rng, n)
        ids = sorted(bdf_of)
        size = rng.randint(1, n)
        must = rng.sample(ids, rng.randint(0, min(2, size)))
        py = preferred_sets(topo, bdf_of, ids, must, size, use_native=False)
        nat = preferred_sets(topo, bdf_of, ids, must, size, use_native=True)
        assert len(py) == len(nat)
        if py:
            s_py = score_set(topo, [bdf_of[d] for d in py])
            s_nat = score_set(topo, [bdf_of[d] for d in nat])
            assert s_nat == s_py, (
                f"trial {trial}: native score {s_nat} != python {s_py} "
                f"(py={py} nat={nat} hives={topo.hive_of} numa={topo.numa_of})"
            )
            assert set(must) <= set(nat)


def test_selector_parity_mock_node(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=8, hives=[[0, 1, 2, 3], [4, 5, 6, 7]])
    cfg = node.config()
    inv = scan_node(cfg)
    topo = load_topology(cfg, inv)
    bdf_of = {g: inv.devices[g].primary.bdf for g in inv.device_ids()}
    ids = inv.device_ids()
    for k in (1, 2, 4, 8):
        py = preferred_sets(topo, bdf_of, ids, [], k, use_native=False)
        nat = preferred_sets(topo, bdf_of, ids, [], k, use_native=True)
        assert score_set(topo, [bdf_of[d] for d in py]) == \
            score_set(topo, [bdf_of[d] for d in nat])


def test_revalidate_groups_batched(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=4, kfd=False, hint=False)
    cfg = node.config()
    devices_dir = f"{cfg.sysfs_root}/bus/pci/devices"
    groups = [(str(70 + i), [f"0000:{0x0a + 8 * i:02x}:00.0"]) for i in range(4)]
    assert _native.revalidate_groups(devices_dir, groups, [0x1002], "vfio-pci") == ""
    bad = groups + [("99", ["0000:0a:00.0"])]
    assert "no longer in IOMMU group" in _native.revalidate_groups(
        devices_dir, bad, [0x1002], "vfio-pci")
    assert "vanished" in _native.revalidate_groups(
        devices_dir, [("70", ["0000:ff:00.0"])], [0x1002], "vfio-pci")


def test_selector_vf_scale_bounded():
    """64-VF symmetric node: selection must stay fast (bounded search) and
    concentration-optimal at every size."""
    import time
    from kata_xpu_device_plugin_amd.topology.hive import GPUTopology, score_set

    topo = GPUTopology(source="hint")
    bdf_of = {}
    ids = []
    for h in range(8):
        for v in range(8):
            did = f"{100 + h * 8 + v}"
            bdf = f"0000:{0x10 + h:02x}:02.{v}"
            bdf_of[did] = bdf
            topo.hive_of[bdf] = f"hive-{h + 1}"
            topo.numa_of[bdf] = h // 4
            ids.append(did)
    from kata_xpu_device_plugin_amd.topology.hive import W_NUMA, W_XGMI
    for k in (8, 16, 32, 64):
        t0 = time.perf_counter()
        pick = preferred_sets(topo, bdf_of, ids, [], k, use_native=True)
        elapsed = time.perf_counter() - t0
        assert len(pick) == k
        # round 2: symmetry reduction + exact concentration bound put the
        # worst case (~73 us on MI355X hosts) far under this ceiling even
        # on slow shared CI (was 23 ms before — 0.2 s ceiling kept 10x
        # slack for CI noise, now 100x)
        assert elapsed < 0.05, f"selection at k={k} took {elapsed:.3f}s"
        # xGMI tier must be concentration-optimal: ceil(k/8) hives
        hives = {topo.hive(bdf_of[d]) for d in pick}
        assert len(hives) == (k + 7) // 8
        # score sanity: full hives contribute C(8,2) xGMI pairs each
        s = score_set(topo, [bdf_of[d] for d in pick])
        full, rem = divmod(k, 8)
        expect_xgmi = full * 28 + rem * (rem - 1) // 2
        assert s // W_XGMI == expect_xgmi
