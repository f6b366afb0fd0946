"""End-to-end CDI contract: Allocate response → runtime-side resolution →
Kata cold-plug inputs (the chain a real containerd+Kata node executes)."""
import os

import pytest

from kata_xpu_device_plugin_amd.cdi.resolve import CDIResolutionError, CDIResolver
from kata_xpu_device_plugin_amd.plugin import api
from kata_xpu_device_plugin_amd.plugin.manager import PluginManager
from kata_xpu_device_plugin_amd.testing.kubelet_stub import KubeletStub
from kata_xpu_device_plugin_amd.testing.mocknode import make_mock_node
from kata_xpu_device_plugin_amd.tools.validate import main as validate_main


def test_allocate_response_resolves_to_kata_contract(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=4, with_audio_fn=True)
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start()
    try:
        r = stub.wait_for_registration(1)[0]
        ps = stub.plugin_stub(r.endpoint)
        resp = ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["70", "72"])]))
        cr = resp.container_responses[0]

        resolver = CDIResolver(cfg.cdi_dir)
        resolved = resolver.resolve_allocate_response(cr)
        assert len(resolved) == 2
        d0 = resolved[0]
        assert d0.kata_cold_plug, "attach-pci must be set for Kata"
        # multi-function group: GPU + audio both in the bdf annotation
        assert d0.bdfs == ["0000:0a:00.0", "0000:0a:00.1"]
        assert d0.device_nodes == [os.path.join(cfg.dev_root, "vfio", "70")]
        # the injected node actually exists on the (mock) host
        assert os.path.exists(d0.device_nodes[0])
    finally:
        mgr.stop()
        stub.stop()


def test_resolver_errors(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=1)
    cfg = node.config()
    mgr = PluginManager(cfg)
    mgr.setup()  # writes the spec
    resolver = CDIResolver(cfg.cdi_dir)
    with pytest.raises(CDIResolutionError, match="no CDI spec for kind"):
        resolver.resolve("other.com/gpu=70")
    with pytest.raises(CDIResolutionError, match="not in spec"):
        resolver.resolve("amd.com/gpu=999")
    with pytest.raises(CDIResolutionError, match="cannot read"):
        CDIResolver(str(tmp_path / "missing"))


def test_validate_tool_ok(tmp_path, monkeypatch, capsys):
    node = make_mock_node(str(tmp_path), n_gpus=2)
    cfg = node.config()
    monkeypatch.setenv("KXDP_SYSFS_ROOT", cfg.sysfs_root)
    monkeypatch.setenv("KXDP_DEV_ROOT", cfg.dev_root)
    monkeypatch.setenv("KXDP_CDI_DIR", cfg.cdi_dir)
    rc = validate_main([])
    out = capsys.readouterr().out
    assert rc == 0, out
    assert "OK  amd.com/gpu=70" in out


def test_validate_tool_detects_missing_node(tmp_path, monkeypatch, capsys):
    node = make_mock_node(str(tmp_path), n_gpus=2)
    cfg = node.config()
    node.remove_vfio_node("71")
    monkeypatch.setenv("KXDP_SYSFS_ROOT", cfg.sysfs_root)
    monkeypatch.setenv("KXDP_DEV_ROOT", cfg.dev_root)
    monkeypatch.setenv("KXDP_CDI_DIR", cfg.cdi_dir)
    rc = validate_main([])
    out = capsys.readouterr().out
    assert rc == 1
    assert "FAIL amd.com/gpu=71" in out
    assert "absent" in out


def test_unhealthy_device_allocation_rejected(tmp_path):
    import grpc
    node = make_mock_node(str(tmp_path), n_gpus=2)
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start()
    try:
        r = stub.wait_for_registration(1)[0]
        ps = stub.plugin_stub(r.endpoint)
        mgr.states[r.resource_name].set_health("70", False)
        with pytest.raises(grpc.RpcError) as ei:
            ps.Allocate(api.AllocateRequest(container_requests=[
                api.ContainerAllocateRequest(devices_ids=["70"])]))
        assert "Unhealthy" in ei.value.details()
        # healthy sibling still allocatable
        resp = ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["71"])]))
        assert resp.container_responses[0].cdi_devices[0].name.endswith("71")
    finally:
        mgr.stop()
        stub.stop()
