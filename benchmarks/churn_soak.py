#!/usr/bin/env python3
"""Sustained-churn soak of the real daemon process.

    python benchmarks/churn_soak.py --minutes 10 --clients 3

Exercises, concurrently and continuously, everything round 2 changed:

* N churn clients: GetPreferredAllocation → Allocate admissions,
* a health flapper removing/re-creating a /dev/vfio group node
  (event-driven ListAndWatch pushes),
* a long-lived ListAndWatch consumer counting pushes,
* a repartitioner enabling/disabling SR-IOV VFs + SIGHUP every cycle
  (dynamic plugin start/registration/retirement),

while watching the daemon's RSS for leaks. Exits 0 iff no unexpected
errors, the daemon survived, shut down cleanly on SIGTERM, and RSS growth
stayed bounded. One JSON result line on stdout.
"""
from __future__ import annotations

import argparse
import json
import os
import signal
import subprocess
import sys
import tempfile
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import grpc  # noqa: E402

from kata_xpu_device_plugin_amd.plugin import api  # noqa: E402
from kata_xpu_device_plugin_amd.testing.kubelet_stub import KubeletStub  # noqa: E402
from kata_xpu_device_plugin_amd.testing.mocknode import MockGPU, make_mock_node  # noqa: E402


def rss_kb(pid: int) -> int:
    try:
        with open(f"/proc/{pid}/status") as f:
            for line in f:
                if line.startswith("VmRSS:"):
                    return int(line.split()[1])
    except OSError:
        pass
    return -1


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--minutes", type=float, default=10.0)
    ap.add_argument("--clients", type=int, default=3)
    ap.add_argument("--flap-interval", type=float, default=0.1)
    ap.add_argument("--repartition-interval", type=float, default=20.0)
    ap.add_argument("--rss-slack-mb", type=float, default=32.0,
                    help="allowed RSS growth before flagging a leak")
    args = ap.parse_args()

    root = tempfile.mkdtemp(prefix="kxdp-soak-")
    node = make_mock_node(root, n_gpus=8)
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    env = os.environ.copy()
    env.update({
        "KXDP_SYSFS_ROOT": cfg.sysfs_root,
        "KXDP_DEV_ROOT": cfg.dev_root,
        "KXDP_CDI_DIR": cfg.cdi_dir,
        "KXDP_KUBELET_DIR": cfg.kubelet_socket_dir,
        "KXDP_TOPOLOGY_HINT": cfg.topology_hint_path,
        "KXDP_PCI_IDS": "",
        "KXDP_AMDSMI_HEALTH": "0",
        "KXDP_LOG_LEVEL": "WARNING",
    })
    daemon = subprocess.Popen(
        [sys.executable, "-m", "kata_xpu_device_plugin_amd"], env=env)
    reg = stub.wait_for_registration(1, timeout=30)[0]
    sock = os.path.join(cfg.kubelet_socket_dir, reg.endpoint)

    stop = threading.Event()
    errors: list = []
    counters = {"allocs": 0, "pushes": 0, "repartitions": 0,
                "vf_registrations": 0}
    lock = threading.Lock()
    stable_ids = [str(70 + i) for i in range(7)]   # 77 is the flap victim

    def churn():
        ch = grpc.insecure_channel(f"unix://{sock}")
        plugin = api.DevicePluginStub(ch)
        try:
            while not stop.is_set():
                try:
                    pref = plugin.GetPreferredAllocation(
                        api.PreferredAllocationRequest(container_requests=[
                            api.ContainerPreferredAllocationRequest(
                                available_device_ids=stable_ids,
                                allocation_size=2)]), timeout=5)
                    pick = list(pref.container_responses[0].device_ids)
                    resp = plugin.Allocate(api.AllocateRequest(
                        container_requests=[api.ContainerAllocateRequest(
                            devices_ids=pick)]), timeout=5)
                    assert len(resp.container_responses[0].cdi_devices) == 2
                    with lock:
                        counters["allocs"] += 1
                except Exception as e:  # noqa: BLE001
                    errors.append(f"churn: {type(e).__name__}: {e}")
                    if len(errors) > 20:
                        return
        finally:
            ch.close()

    def flapper():
        while not stop.is_set():
            node.remove_vfio_node("77")
            time.sleep(args.flap_interval)
            node.add_vfio_node("77")
            time.sleep(args.flap_interval)

    def lw_consumer():
        ch = grpc.insecure_channel(f"unix://{sock}")
        plugin = api.DevicePluginStub(ch)
        stream = plugin.ListAndWatch(api.Empty())
        try:
            for upd in stream:
                with lock:
                    counters["pushes"] += 1
                if stop.is_set():
                    break
        except grpc.RpcError:
            if not stop.is_set():
                errors.append("lw stream died mid-soak")
        finally:
            stream.cancel()
            ch.close()

    def repartitioner():
        cycle = 0
        while not stop.wait(args.repartition_interval):
            cycle += 1
            try:
                for k in range(2):
                    node.add_gpu(MockGPU(
                        bdf=f"0000:60:02.{k}", device_id=0x75B3,
                        iommu_group=str(300 + k),
                        physfn_bdf="0000:0a:00.0"))
                daemon.send_signal(signal.SIGHUP)
                deadline = time.monotonic() + 10
                seen = False
                while time.monotonic() < deadline:
                    if any(r.resource_name.endswith("_VF")
                           for r in stub.registrations):
                        seen = True
                        break
                    time.sleep(0.1)
                if not seen:
                    errors.append(f"cycle {cycle}: VF resource never registered")
                else:
                    with lock:
                        counters["vf_registrations"] += 1
                stub.registrations.clear()
                if stop.wait(3.0):
                    return
                for k in range(2):
                    node.remove_gpu(f"0000:60:02.{k}")
                daemon.send_signal(signal.SIGHUP)
                with lock:
                    counters["repartitions"] += 1
            except Exception as e:  # noqa: BLE001
                errors.append(f"repartition {cycle}: {type(e).__name__}: {e}")

    rss_start = rss_kb(daemon.pid)
    threads = [threading.Thread(target=churn, daemon=True)
               for _ in range(args.clients)]
    threads += [threading.Thread(target=flapper, daemon=True),
                threading.Thread(target=lw_consumer, daemon=True),
                threading.Thread(target=repartitioner, daemon=True)]
    t0 = time.monotonic()
    for t in threads:
        t.start()
    time.sleep(args.minutes * 60)
    stop.set()
    elapsed = time.monotonic() - t0
    node.add_vfio_node("77")   # settle state so joiners finish cleanly
    for t in threads:
        t.join(timeout=10)
    rss_end = rss_kb(daemon.pid)

    daemon_alive = daemon.poll() is None
    if daemon_alive:
        daemon.terminate()
        try:
            clean_exit = daemon.wait(timeout=15) == 0
        except subprocess.TimeoutExpired:
            daemon.kill()
            clean_exit = False
    else:
        clean_exit = False
        errors.append(f"daemon died mid-soak rc={daemon.returncode}")
    stub.stop()

    rss_growth_mb = (rss_end - rss_start) / 1024 if rss_start > 0 else 0.0
    leak = rss_growth_mb > args.rss_slack_mb
    if leak:
        errors.append(f"RSS grew {rss_growth_mb:.1f} MiB "
                      f"(> {args.rss_slack_mb} MiB slack)")
    ok = not errors and clean_exit
    print(json.dumps({
        "soak_ok": ok,
        "elapsed_s": round(elapsed, 1),
        "allocations": counters["allocs"],
        "allocs_per_s": round(counters["allocs"] / elapsed, 1),
        "lw_pushes": counters["pushes"],
        "vf_repartitions": counters["repartitions"],
        "vf_registrations": counters["vf_registrations"],
        "rss_start_kb": rss_start,
        "rss_end_kb": rss_end,
        "rss_growth_mb": round(rss_growth_mb, 1),
        "clean_shutdown": clean_exit,
        "errors": errors[:10],
    }))
    return 0 if ok else 1


if __name__ == "__main__":
    raise SystemExit(main())
