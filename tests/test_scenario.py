"""Day-in-the-life scenario: the full operator flow on one mock node.

provision (snapshot topology while amdgpu-bound) → bind to vfio →
daemon serves → pods admitted with hive-aware placement → SR-IOV
repartition + live rescan → device failure and recovery → drain.
Stitches discovery, topology, CDI, server, health and tools together —
the closest CPU-only approximation of BASELINE configs #2–#5.
"""
import json
import os

import grpc
import pytest

from kata_xpu_device_plugin_amd.cdi.resolve import CDIResolver
from kata_xpu_device_plugin_amd.cdi.spec import read_spec
from kata_xpu_device_plugin_amd.plugin import api
from kata_xpu_device_plugin_amd.plugin.manager import PluginManager
from kata_xpu_device_plugin_amd.testing.kubelet_stub import KubeletStub
from kata_xpu_device_plugin_amd.testing.mocknode import MockGPU, MockNode, default_bdfs
from kata_xpu_device_plugin_amd.tools.topo import build_snapshot


def test_full_lifecycle(tmp_path):
    # ---- day 0: node arrives with GPUs on amdgpu; capture topology ----
    node = MockNode(root=str(tmp_path))
    for i, bdf in enumerate(default_bdfs(8)):
        node.add_gpu(MockGPU(bdf=bdf, iommu_group=str(70 + i), driver="amdgpu",
                             numa_node=i // 4,
                             hive_id=1 if i < 4 else 2))
    node.write_kfd_topology()
    snapshot = build_snapshot(node.sysfs)
    assert len(snapshot["hives"]) == 2
    hint_path = os.path.join(str(tmp_path), "etc", "topology.json")
    os.makedirs(os.path.dirname(hint_path), exist_ok=True)
    with open(hint_path, "w") as f:
        json.dump(snapshot, f)

    # ---- bind to vfio-pci (mock: flip driver symlinks, KFD goes dark) ----
    import shutil
    for g in node.gpus:
        g.driver = "vfio-pci"
        d = node._pci_dir(g.bdf)
        os.unlink(os.path.join(d, "driver"))
        node._symlink(os.path.join(node.sysfs, "bus", "pci", "drivers", "vfio-pci"),
                      os.path.join(d, "driver"))
    shutil.rmtree(os.path.join(node.sysfs, "class", "kfd"))

    cfg = node.config()
    cfg.topology_hint_path = hint_path

    # ---- daemon up, kubelet registers ----
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start()
    try:
        assert mgr.topology.source == "hint"
        reg = stub.wait_for_registration(1)[0]
        assert reg.resource_name == "amd.com/INSTINCT_MI355X"
        ps = stub.plugin_stub(reg.endpoint)

        # ---- two 4-GPU pods: each must land wholly on one hive (config #4)
        ids = [str(70 + i) for i in range(8)]
        taken = []
        for _ in range(2):
            avail = [d for d in ids if d not in taken]
            pref = ps.GetPreferredAllocation(api.PreferredAllocationRequest(
                container_requests=[api.ContainerPreferredAllocationRequest(
                    available_device_ids=avail, allocation_size=4)]))
            pick = list(pref.container_responses[0].device_ids)
            hives = {mgr.topology.hive(mgr.inventory.devices[d].primary.bdf)
                     for d in pick}
            assert len(hives) == 1, (pick, hives)
            resp = ps.Allocate(api.AllocateRequest(container_requests=[
                api.ContainerAllocateRequest(devices_ids=pick)]))
            # runtime side resolves every CDI id to the Kata contract
            resolver = CDIResolver(cfg.cdi_dir)
            for rd in resolver.resolve_allocate_response(
                    resp.container_responses[0]):
                assert rd.kata_cold_plug and rd.bdfs
            taken += pick
        assert sorted(taken) == ids  # both hives fully assigned

        # ---- SR-IOV repartition: admin enables 2 VFs on a returned GPU ----
        pf_bdf = "0000:0a:00.0"
        for k in range(2):
            node.add_gpu(MockGPU(bdf=f"0000:0a:02.{k}", device_id=0x75B3,
                                 iommu_group=str(200 + k), physfn_bdf=pf_bdf))
        mgr.rescan()
        spec = read_spec(mgr.cdi_spec_path)
        assert {"200", "201"} <= set(spec.device_names())
        # VFs are a NEW resource name → its plugin starts and registers IN
        # PLACE (round 2: no daemon restart), and the existing MI355X
        # plugin keeps serving
        vf_reg = next(r for r in stub.wait_for_registration(2)
                      if r.resource_name.endswith("_VF"))
        ps_vf = stub.plugin_stub(vf_reg.endpoint)
        resp = ps_vf.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["200"])]))
        assert resp.container_responses[0].cdi_devices[0].name == "amd.com/gpu=200"
        resp = ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["70"])]))
        assert resp.container_responses[0].cdi_devices[0].name == "amd.com/gpu=70"

        # ---- xGMI link sickness: placement avoids the GPU, health intact --
        from kata_xpu_device_plugin_amd.health.amdsmi_health import (
            DeviceHealth, XgmiLinkHealth)
        sick_bdf = mgr.inventory.devices["71"].primary.bdf   # hive 1
        mgr._on_xgmi_telemetry({sick_bdf: DeviceHealth(
            bdf=sick_bdf,
            xgmi_links=[XgmiLinkHealth(index=3, status="down", errors=1)])})
        pref = ps.GetPreferredAllocation(api.PreferredAllocationRequest(
            container_requests=[api.ContainerPreferredAllocationRequest(
                available_device_ids=ids, allocation_size=2)]))
        pick = list(pref.container_responses[0].device_ids)
        assert "71" not in pick, "degraded-fabric GPU must be avoided"
        # ... but it still allocates when kubelet insists (health intact)
        resp = ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["71"])]))
        assert resp.container_responses[0].cdi_devices[0].name == "amd.com/gpu=71"
        mgr._on_xgmi_telemetry({sick_bdf: DeviceHealth(bdf=sick_bdf)})

        # ---- device failure: vfio node vanishes → Unhealthy + reject ----
        stream = ps.ListAndWatch(api.Empty())
        next(stream)
        node.remove_vfio_node("73")
        upd = next(stream)
        assert {d.id: d.health for d in upd.devices}["73"] == api.UNHEALTHY
        with pytest.raises(grpc.RpcError):
            ps.Allocate(api.AllocateRequest(container_requests=[
                api.ContainerAllocateRequest(devices_ids=["73"])]))
        node.add_vfio_node("73")
        upd = next(stream)
        assert {d.id: d.health for d in upd.devices}["73"] == api.HEALTHY
        stream.cancel()

        # ---- recovered device allocatable again ----
        resp = ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["73"])]))
        assert resp.container_responses[0].cdi_devices[0].name == "amd.com/gpu=73"
    finally:
        mgr.stop()
        stub.stop()
