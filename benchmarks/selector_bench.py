"""Microbenchmark: preferred-set selection latency across node shapes.

    python benchmarks/selector_bench.py

Covers whole-GPU nodes (8×MI355X, 1 or 2 hives) and VF-partitioned nodes
(64 VFs over 8 pseudo-hives) at every pod size — the GetPreferredAllocation
server-side cost. The branch-and-bound selector (topology/hive.py /
native/xpu_native.cpp) is exact within a 50k-node search budget and falls
back to the concentration-greedy first descent beyond it.
"""
import sys
import os
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kata_xpu_device_plugin_amd.topology.hive import (  # noqa: E402
    GPUTopology,
    preferred_sets,
)


def build(n_hives, per_hive, numa_split=2):
    topo = GPUTopology(source="hint")
    bdf_of = {}
    ids = []
    for h in range(n_hives):
        for v in range(per_hive):
            did = f"{100 + h * per_hive + v}"
            bdf = f"0000:{0x10 + h:02x}:02.{v:x}" if per_hive > 1 else \
                f"0000:{0x10 + h:02x}:00.0"
            bdf_of[did] = bdf
            topo.hive_of[bdf] = f"hive-{h + 1}"
            topo.numa_of[bdf] = h * numa_split // max(1, n_hives)
            ids.append(did)
    return topo, bdf_of, ids


def bench(label, topo, bdf_of, ids, sizes, native):
    for k in sizes:
        best = float("inf")
        for _ in range(5):
            t0 = time.perf_counter()
            pick = preferred_sets(topo, bdf_of, ids, [], k, use_native=native)
            best = min(best, time.perf_counter() - t0)
        assert len(pick) == k
        impl = "native" if native else "python"
        print(f"{label:28s} k={k:3d} [{impl}] {best * 1e6:9.1f} us")


def main():
    shapes = [
        ("8xGPU single hive", build(1, 8)),
        ("8xGPU 4+4 hives", build(2, 4)),
        ("64 VF / 8 hives", build(8, 8)),
    ]
    for label, (topo, bdf_of, ids) in shapes:
        sizes = [1, 2, 4, len(ids) // 2, len(ids)]
        for native in (True, False):
            bench(label, topo, bdf_of, ids, sizes, native)
        print()


if __name__ == "__main__":
    main()
