"""Integration tier: in-process kubelet stub over real
unix sockets exercising register → ListAndWatch → Allocate round-trips,
health flapping, and kubelet-restart recovery — BASELINE.json config #1,
no GPU required.

Reference surfaces matched: Register (generic_device_plugin.go:200-219),
ListAndWatch (:222-250), Allocate (:320-355), healthCheck (:389-457).
"""
import os
import threading
import time

import grpc
import pytest

from kata_xpu_device_plugin_amd.plugin import api
from kata_xpu_device_plugin_amd.plugin.manager import PluginManager
from kata_xpu_device_plugin_amd.testing.kubelet_stub import KubeletStub
from kata_xpu_device_plugin_amd.testing.mocknode import make_mock_node


@pytest.fixture
def running(tmp_path):
    """Mock node + kubelet stub + manager, fully started."""
    node = make_mock_node(str(tmp_path))
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start(register=True)
    yield node, cfg, stub, mgr
    mgr.stop()
    stub.stop()


def _first_plugin(mgr):
    return next(iter(mgr.plugins.values()))


def test_registration(running):
    node, cfg, stub, mgr = running
    regs = stub.wait_for_registration(1)
    assert len(regs) == 1
    r = regs[0]
    assert r.version == "v1beta1"
    assert r.resource_name == "amd.com/INSTINCT_MI355X"
    assert r.preferred_allocation, "GetPreferredAllocation must be advertised"
    assert os.path.exists(os.path.join(cfg.kubelet_socket_dir, r.endpoint))


def test_options(running):
    node, cfg, stub, mgr = running
    r = stub.wait_for_registration(1)[0]
    ps = stub.plugin_stub(r.endpoint)
    opts = ps.GetDevicePluginOptions(api.Empty())
    assert not opts.pre_start_required
    assert opts.get_preferred_allocation_available


def test_list_and_watch_initial(running):
    node, cfg, stub, mgr = running
    r = stub.wait_for_registration(1)[0]
    ps = stub.plugin_stub(r.endpoint)
    stream = ps.ListAndWatch(api.Empty())
    first = next(stream)
    assert len(first.devices) == 8
    ids = [d.id for d in first.devices]
    assert ids == [str(70 + i) for i in range(8)]
    assert all(d.health == api.HEALTHY for d in first.devices)
    # NUMA topology present (reference sends none)
    assert first.devices[0].topology.nodes[0].id == 0
    assert first.devices[7].topology.nodes[0].id == 1
    stream.cancel()


def test_health_flap_pushes_updates(running):
    node, cfg, stub, mgr = running
    r = stub.wait_for_registration(1)[0]
    ps = stub.plugin_stub(r.endpoint)
    stream = ps.ListAndWatch(api.Empty())
    assert len(next(stream).devices) == 8

    node.remove_vfio_node("73")  # device disappears
    upd = next(stream)
    health = {d.id: d.health for d in upd.devices}
    assert health["73"] == api.UNHEALTHY
    assert health["72"] == api.HEALTHY

    node.add_vfio_node("73")     # device comes back
    upd = next(stream)
    health = {d.id: d.health for d in upd.devices}
    assert health["73"] == api.HEALTHY
    stream.cancel()


def test_many_concurrent_listandwatch_streams(running):
    """8 concurrent ListAndWatch streams ALL receive every health push.
    Round 1 served stream waits from a 4-thread executor — a 5th stream
    starved silently; the asyncio bridge has no cap."""
    node, cfg, stub, mgr = running
    r = stub.wait_for_registration(1)[0]
    ps = stub.plugin_stub(r.endpoint)
    streams = [ps.ListAndWatch(api.Empty()) for _ in range(8)]
    for s in streams:
        assert len(next(s).devices) == 8

    node.remove_vfio_node("75")
    for i, s in enumerate(streams):
        upd = next(s)  # every stream gets the push, none starves
        health = {d.id: d.health for d in upd.devices}
        assert health["75"] == api.UNHEALTHY, f"stream {i} missed the update"

    node.add_vfio_node("75")
    for i, s in enumerate(streams):
        upd = next(s)
        health = {d.id: d.health for d in upd.devices}
        assert health["75"] == api.HEALTHY, f"stream {i} missed the recovery"
    for s in streams:
        s.cancel()


def test_allocate_cdi_cri(running):
    node, cfg, stub, mgr = running
    r = stub.wait_for_registration(1)[0]
    ps = stub.plugin_stub(r.endpoint)
    resp = ps.Allocate(api.AllocateRequest(container_requests=[
        api.ContainerAllocateRequest(devices_ids=["70", "71"]),
    ]))
    assert len(resp.container_responses) == 1
    cr = resp.container_responses[0]
    assert [c.name for c in cr.cdi_devices] == ["amd.com/gpu=70", "amd.com/gpu=71"]
    assert cr.envs["KUBERNETES_CDI_VENDOR_CLASS"] == "amd.com/gpu"
    assert cr.envs["PCI_RESOURCE_AMD_COM_INSTINCT_MI355X"] == (
        "0000:0a:00.0,0000:12:00.0"
    )
    assert len(cr.devices) == 0  # no raw device nodes under cdi-cri


def test_allocate_unknown_device_rejected(running):
    node, cfg, stub, mgr = running
    r = stub.wait_for_registration(1)[0]
    ps = stub.plugin_stub(r.endpoint)
    with pytest.raises(grpc.RpcError) as ei:
        ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["999"]),
        ]))
    assert ei.value.code() == grpc.StatusCode.INVALID_ARGUMENT


def test_allocate_revalidates_sysfs(running):
    """Device vanished from sysfs between ListAndWatch and Allocate
    (reference: generic_device_plugin.go:329-338 rejects)."""
    node, cfg, stub, mgr = running
    r = stub.wait_for_registration(1)[0]
    ps = stub.plugin_stub(r.endpoint)
    # unbind the GPU: driver symlink no longer vfio-pci
    d = os.path.join(cfg.sysfs_root, "bus", "pci", "devices", "0000:0a:00.0", "driver")
    os.unlink(d)
    with pytest.raises(grpc.RpcError) as ei:
        ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["70"]),
        ]))
    assert ei.value.code() == grpc.StatusCode.INVALID_ARGUMENT
    # other devices still fine
    resp = ps.Allocate(api.AllocateRequest(container_requests=[
        api.ContainerAllocateRequest(devices_ids=["71"]),
    ]))
    assert resp.container_responses[0].cdi_devices[0].name == "amd.com/gpu=71"


def test_preferred_allocation_rpc(running):
    node, cfg, stub, mgr = running
    r = stub.wait_for_registration(1)[0]
    ps = stub.plugin_stub(r.endpoint)
    resp = ps.GetPreferredAllocation(api.PreferredAllocationRequest(
        container_requests=[api.ContainerPreferredAllocationRequest(
            available_device_ids=[str(70 + i) for i in range(8)],
            allocation_size=4,
        )]
    ))
    pick = list(resp.container_responses[0].device_ids)
    assert len(pick) == 4


def test_pre_start_container(running):
    node, cfg, stub, mgr = running
    r = stub.wait_for_registration(1)[0]
    ps = stub.plugin_stub(r.endpoint)
    resp = ps.PreStartContainer(api.PreStartContainerRequest(devices_ids=["70"]))
    assert resp is not None


def test_socket_removal_triggers_restart(running):
    """kubelet wiping the plugin socket → plugin restarts and re-registers
    (reference: healthCheck :444-454)."""
    node, cfg, stub, mgr = running
    r = stub.wait_for_registration(1)[0]
    plugin = _first_plugin(mgr)
    os.unlink(plugin.socket_path)
    deadline = time.monotonic() + 5
    while len(stub.registrations) < 2 and time.monotonic() < deadline:
        time.sleep(0.05)
    assert len(stub.registrations) >= 2, "plugin must re-register after socket removal"
    assert plugin.serving
    assert os.path.exists(plugin.socket_path)


def test_kubelet_restart_reregisters(running):
    node, cfg, stub, mgr = running
    stub.wait_for_registration(1)
    # simulate kubelet restart: remove + re-create kubelet.sock server
    stub.stop()
    stub.start()
    deadline = time.monotonic() + 5
    while len(stub.registrations) < 1 and time.monotonic() < deadline:
        time.sleep(0.05)
    assert len(stub.registrations) >= 1, "plugin must re-register with new kubelet"


def test_cdi_spec_written(running):
    node, cfg, stub, mgr = running
    assert mgr.cdi_spec_path is not None
    assert os.path.exists(mgr.cdi_spec_path)


def test_concurrent_allocate_churn(running):
    """Pod-churn concurrency: parallel Allocate calls against one plugin
    must all succeed (the benchmark hammers this path)."""
    node, cfg, stub, mgr = running
    r = stub.wait_for_registration(1)[0]
    ps = stub.plugin_stub(r.endpoint)
    errs = []

    def churn(gid):
        try:
            for _ in range(50):
                resp = ps.Allocate(api.AllocateRequest(container_requests=[
                    api.ContainerAllocateRequest(devices_ids=[gid]),
                ]))
                assert resp.container_responses[0].cdi_devices[0].name.endswith(gid)
        except Exception as e:  # pragma: no cover
            errs.append(e)

    threads = [threading.Thread(target=churn, args=(str(70 + i),)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errs
    assert _first_plugin(mgr).allocations == 400


def test_listandwatch_unsubscribes_on_cancel(running):
    """Cancelled/closed streams must release their DeviceState callback
    subscriptions (the soak's RSS bound depends on this)."""
    node, cfg, stub, mgr = running
    r = stub.wait_for_registration(1)[0]
    ps = stub.plugin_stub(r.endpoint)
    state = next(iter(mgr.states.values()))
    base = len(state._callbacks)
    streams = [ps.ListAndWatch(api.Empty()) for _ in range(5)]
    for s in streams:
        next(s)
    assert len(state._callbacks) == base + 5
    for s in streams:
        s.cancel()
    deadline = time.time() + 5
    while len(state._callbacks) > base and time.time() < deadline:
        time.sleep(0.05)
    assert len(state._callbacks) == base, "stream subscriptions leaked"
