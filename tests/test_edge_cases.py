"""API edge cases: oversized requests, empty preferred sets, validate tool
on a not-ready node, resolver with multiple kinds."""
import grpc
import pytest

from kata_xpu_device_plugin_amd.plugin import api
from kata_xpu_device_plugin_amd.plugin.manager import PluginManager
from kata_xpu_device_plugin_amd.testing.kubelet_stub import KubeletStub
from kata_xpu_device_plugin_amd.testing.mocknode import make_mock_node
from kata_xpu_device_plugin_amd.tools.validate import main as validate_main


@pytest.fixture
def served(tmp_path):
    node = make_mock_node(str(tmp_path))
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start()
    r = stub.wait_for_registration(1)[0]
    ps = stub.plugin_stub(r.endpoint)
    yield node, cfg, mgr, ps
    mgr.stop()
    stub.stop()


def test_preferred_size_zero_and_oversize(served):
    node, cfg, mgr, ps = served
    ids = [str(70 + i) for i in range(8)]
    for size in (0, 9, -1):
        resp = ps.GetPreferredAllocation(api.PreferredAllocationRequest(
            container_requests=[api.ContainerPreferredAllocationRequest(
                available_device_ids=ids, allocation_size=size)]))
        assert list(resp.container_responses[0].device_ids) == []


def test_preferred_with_unknown_available(served):
    node, cfg, mgr, ps = served
    resp = ps.GetPreferredAllocation(api.PreferredAllocationRequest(
        container_requests=[api.ContainerPreferredAllocationRequest(
            available_device_ids=["70", "ghost", "71"], allocation_size=2)]))
    pick = list(resp.container_responses[0].device_ids)
    assert len(pick) == 2 and "ghost" not in pick


def test_allocate_duplicate_ids(served):
    node, cfg, mgr, ps = served
    resp = ps.Allocate(api.AllocateRequest(container_requests=[
        api.ContainerAllocateRequest(devices_ids=["70", "70"])]))
    # duplicates are echoed (kubelet never sends them, but the response
    # stays structurally consistent — one CDI name per requested entry)
    assert len(resp.container_responses[0].cdi_devices) == 2


def test_allocate_all_eight(served):
    node, cfg, mgr, ps = served
    ids = [str(70 + i) for i in range(8)]
    resp = ps.Allocate(api.AllocateRequest(container_requests=[
        api.ContainerAllocateRequest(devices_ids=ids)]))
    cr = resp.container_responses[0]
    assert len(cr.cdi_devices) == 8
    env = cr.envs["PCI_RESOURCE_AMD_COM_INSTINCT_MI355X"]
    assert len(env.split(",")) == 8


def test_validate_not_ready_node(tmp_path, monkeypatch, capsys):
    """amdgpu-bound node (no vfio devices) → validate exits 1 with a clear
    message (the live-node behavior verified in GPU run 3)."""
    node = make_mock_node(str(tmp_path), n_gpus=2, driver="amdgpu",
                          kfd=False, hint=False)
    cfg = node.config()
    monkeypatch.setenv("KXDP_SYSFS_ROOT", cfg.sysfs_root)
    monkeypatch.setenv("KXDP_DEV_ROOT", cfg.dev_root)
    monkeypatch.setenv("KXDP_CDI_DIR", cfg.cdi_dir)
    rc = validate_main([])
    assert rc == 1
    err = capsys.readouterr().err
    assert "nothing to validate" in err


def test_listandwatch_two_streams(served):
    """Two concurrent ListAndWatch streams (kubelet reconnects without
    closing the old one) both receive pushes."""
    node, cfg, mgr, ps = served
    s1 = ps.ListAndWatch(api.Empty())
    s2 = ps.ListAndWatch(api.Empty())
    assert len(next(s1).devices) == 8
    assert len(next(s2).devices) == 8
    node.remove_vfio_node("75")
    h1 = {d.id: d.health for d in next(s1).devices}
    h2 = {d.id: d.health for d in next(s2).devices}
    assert h1["75"] == h2["75"] == api.UNHEALTHY
    s1.cancel()
    s2.cancel()
