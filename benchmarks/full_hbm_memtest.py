"""Near-full-HBM memtest: single allocation covering all free HBM minus
headroom; address-pattern write+verify on-GPU (probes/xpu_probe.hip)."""
import sys
import time

from kata_xpu_device_plugin_amd import _gpuprobe as g


def main() -> int:
    info = g.device_info(0)
    free = info["free_mem_bytes"]
    test_bytes = free - (8 << 30)
    print(f"free {free / (1 << 30):.1f} GiB; testing {test_bytes / (1 << 30):.1f} GiB")
    t0 = time.time()
    r = g.memtest(0, test_bytes)
    gib = r["bytes"] / (1 << 30)
    print(f"full-HBM memtest: {gib:.1f} GiB, {r['mismatches']} mismatches, "
          f"{time.time() - t0:.1f}s")
    return 0 if r["mismatches"] == 0 else 1


if __name__ == "__main__":
    sys.exit(main())
