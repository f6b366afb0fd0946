"""Unit tests: config, DeviceState, inotify, amd-smi poller, metrics,
allocate strategies, manager rescan."""
import os
import threading
import time

import pytest

from kata_xpu_device_plugin_amd.config import Config
from kata_xpu_device_plugin_amd.discovery import scan_node
from kata_xpu_device_plugin_amd.health.amdsmi_health import AmdSmiPoller, DeviceHealth
from kata_xpu_device_plugin_amd.plugin import api
from kata_xpu_device_plugin_amd.plugin.manager import PluginManager
from kata_xpu_device_plugin_amd.plugin.state import DeviceState
from kata_xpu_device_plugin_amd.testing.kubelet_stub import KubeletStub
from kata_xpu_device_plugin_amd.testing.mocknode import MockGPU, make_mock_node
from kata_xpu_device_plugin_amd.utils import inotify


# --- config ----------------------------------------------------------------

def test_config_env_parsing(monkeypatch):
    monkeypatch.setenv("KXDP_VENDORS", "1002,10ee")
    monkeypatch.setenv("KXDP_DEVICES", "75a3")
    monkeypatch.setenv("KXDP_STRATEGY", "cdi-annotations")
    monkeypatch.setenv("KXDP_METRICS_PORT", "9110")
    monkeypatch.setenv("KXDP_AMDSMI_HEALTH", "false")
    cfg = Config()
    cfg.validate()
    assert cfg.vendor_allowlist == (0x1002, 0x10EE)
    assert cfg.device_allowlist == (0x75A3,)
    assert cfg.device_list_strategy == "cdi-annotations"
    assert cfg.metrics_port == 9110
    assert cfg.amdsmi_health is False


def test_config_validation_errors():
    cfg = Config(device_list_strategy="bogus")
    with pytest.raises(ValueError):
        cfg.validate()
    cfg = Config(cdi_kind="nokind")
    with pytest.raises(ValueError):
        cfg.validate()
    cfg = Config(cdi_format="xml")
    with pytest.raises(ValueError):
        cfg.validate()


def test_config_from_args():
    import argparse
    p = argparse.ArgumentParser()
    Config.add_args(p)
    args = p.parse_args(["--strategy", "device-nodes", "--namespace", "x.org"])
    cfg = Config.from_args(args)
    assert cfg.device_list_strategy == "device-nodes"
    assert cfg.resource_namespace == "x.org"


# --- DeviceState -----------------------------------------------------------

def _state_of(tmp_path, n=2):
    node = make_mock_node(str(tmp_path), n_gpus=n, kfd=False, hint=False)
    inv = scan_node(node.config())
    return DeviceState(inv.devices), inv


def test_state_snapshot_and_health(tmp_path):
    st, inv = _state_of(tmp_path)
    assert st.device_ids() == ["70", "71"]
    assert st.healthy_ids() == ["70", "71"]
    assert st.set_health("70", False)
    assert not st.set_health("70", False)  # no change → no notify
    assert st.healthy_ids() == ["71"]
    assert not st.set_health("nope", False)


def test_state_watch_notification(tmp_path):
    st, inv = _state_of(tmp_path)
    q = st.watch()
    st.set_health("70", False)
    assert q.get(timeout=1) is not None
    st.unwatch(q)
    st.set_health("70", True)
    assert q.empty()


def test_state_replace_devices_preserves_health(tmp_path):
    st, inv = _state_of(tmp_path, n=3)
    st.set_health("71", False)
    st.replace_devices({g: d for g, d in inv.devices.items() if g != "72"})
    assert st.device_ids() == ["70", "71"]
    assert st.healthy_ids() == ["70"]  # 71 stays unhealthy


def test_state_concurrent_mutation(tmp_path):
    st, inv = _state_of(tmp_path, n=8)
    stop = threading.Event()
    errs = []

    def flipper(gid):
        try:
            while not stop.is_set():
                st.set_health(gid, False)
                st.set_health(gid, True)
        except Exception as e:  # pragma: no cover
            errs.append(e)

    def reader():
        try:
            while not stop.is_set():
                snap = st.snapshot()
                assert len(snap) == 8
        except Exception as e:  # pragma: no cover
            errs.append(e)

    ts = [threading.Thread(target=flipper, args=(str(70 + i),)) for i in range(4)]
    ts += [threading.Thread(target=reader) for _ in range(2)]
    for t in ts:
        t.start()
    time.sleep(0.3)
    stop.set()
    for t in ts:
        t.join()
    assert not errs


# --- inotify ---------------------------------------------------------------

def test_inotify_create_delete(tmp_path):
    with inotify.Inotify() as ino:
        ino.add_watch(str(tmp_path), inotify.IN_CREATE | inotify.IN_DELETE)
        f = tmp_path / "x"
        f.write_text("")
        evs = ino.read_events(timeout=2)
        assert any(e.name == "x" and e.created for e in evs)
        f.unlink()
        evs = ino.read_events(timeout=2)
        assert any(e.name == "x" and e.removed for e in evs)


def test_inotify_bad_path():
    with inotify.Inotify() as ino:
        with pytest.raises(OSError):
            ino.add_watch("/does/not/exist", inotify.IN_CREATE)


# --- amd-smi poller (faked snapshots) --------------------------------------

def test_amdsmi_poller_transitions():
    calls = []
    healthy = {"0000:0a:00.0": True}

    def snap():
        return {b: DeviceHealth(bdf=b, healthy=h, reasons=[] if h else ["ecc"])
                for b, h in healthy.items()}

    p = AmdSmiPoller(999, calls_append := (lambda b, h, r: calls.append((b, h))),
                     snapshot_fn=snap)
    p.poll_once()
    assert calls == []  # initially healthy → no event
    healthy["0000:0a:00.0"] = False
    p.poll_once()
    assert calls == [("0000:0a:00.0", False)]
    p.poll_once()
    assert calls == [("0000:0a:00.0", False)]  # no repeat
    healthy["0000:0a:00.0"] = True
    p.poll_once()
    assert calls[-1] == ("0000:0a:00.0", True)


def test_amdsmi_initially_unhealthy_reported():
    calls = []
    p = AmdSmiPoller(
        999, lambda b, h, r: calls.append((b, h)),
        snapshot_fn=lambda: {"b": DeviceHealth(bdf="b", healthy=False,
                                               reasons=["ecc"])},
    )
    p.poll_once()
    assert calls == [("b", False)]


def test_amdsmi_snapshot_no_library_is_empty_or_dict():
    # On CPU boxes amdsmi may import but see no devices; must not raise.
    from kata_xpu_device_plugin_amd.health.amdsmi_health import snapshot
    assert isinstance(snapshot(), dict)


# --- allocate strategies ---------------------------------------------------

@pytest.mark.parametrize("strategy,check", [
    ("cdi-cri", "cdi"),
    ("cdi-annotations", "ann"),
    ("device-nodes", "nodes"),
])
def test_allocate_strategies(tmp_path, strategy, check):
    node = make_mock_node(str(tmp_path), n_gpus=2)
    cfg = node.config(device_list_strategy=strategy)
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start()
    try:
        reg = stub.wait_for_registration(1)[0]
        ps = stub.plugin_stub(reg.endpoint)
        resp = ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["70"]),
        ]))
        cr = resp.container_responses[0]
        if check == "cdi":
            assert [c.name for c in cr.cdi_devices] == ["amd.com/gpu=70"]
            assert len(cr.devices) == 0 and len(cr.annotations) == 0
        elif check == "ann":
            assert cr.annotations["cdi.k8s.io/vfio70"] == "amd.com/gpu=70"
            assert len(cr.cdi_devices) == 0
        else:
            # First node is the VFIO container device /dev/vfio/vfio (a runc
            # container cannot open a group fd without it), then the group.
            assert cr.devices[0].host_path.endswith("/vfio/vfio")
            assert cr.devices[1].host_path.endswith("/vfio/70")
            assert all(d.permissions == "rw" for d in cr.devices)
            assert len(cr.devices) == 2
            assert len(cr.cdi_devices) == 0
            assert "KUBERNETES_CDI_VENDOR_CLASS" not in cr.envs
    finally:
        mgr.stop()
        stub.stop()


# --- manager rescan --------------------------------------------------------

def test_manager_rescan_picks_up_new_device(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=2, kfd=False, hint=False)
    cfg = node.config()
    mgr = PluginManager(cfg)
    mgr.setup()
    assert len(mgr.inventory.devices) == 2
    node.add_gpu(MockGPU(bdf="0000:66:00.0", iommu_group="99"))
    mgr.rescan()
    assert len(mgr.inventory.devices) == 3
    state = mgr.states["amd.com/INSTINCT_MI355X"]
    assert "99" in state.device_ids()


def test_manager_unified_resource(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=2, kfd=False, hint=False)
    node.add_gpu(MockGPU(bdf="0000:66:00.0", device_id=0x74A1, iommu_group="99"))
    cfg = node.config(unified_resource_name="gpu")
    mgr = PluginManager(cfg)
    mgr.setup()
    assert list(mgr.plugins) == ["amd.com/GPU"]
    assert len(mgr.states["amd.com/GPU"].device_ids()) == 3


# --- metrics ---------------------------------------------------------------

def test_metrics_exporter_refresh(tmp_path):
    from prometheus_client import generate_latest
    from kata_xpu_device_plugin_amd.metrics import MetricsExporter

    node = make_mock_node(str(tmp_path), n_gpus=2, kfd=False, hint=False)
    mgr = PluginManager(node.config())
    mgr.setup()
    exp = MetricsExporter(mgr)
    text = generate_latest(exp.registry).decode()
    assert 'kxdp_devices{resource="amd.com/INSTINCT_MI355X"} 2.0' in text
    assert "kxdp_discovery_seconds" in text
    # *_total series must be Counter-typed (advisor r1): rate()/increase()
    # depend on the counter contract, and the exposition name gets the
    # _total suffix appended by the client library.
    assert "# TYPE kxdp_allocations_total counter" in text
    assert "# TYPE kxdp_allocation_failures_total counter" in text
    assert "# TYPE kxdp_allocate_seconds_total counter" in text
    assert 'kxdp_allocations_total{resource="amd.com/INSTINCT_MI355X"} 0.0' in text
    assert "# TYPE kxdp_devices gauge" in text


def test_state_multi_source_health(tmp_path):
    """A device is Healthy iff NO source holds it unhealthy; one source's
    recovery cannot mask another's outstanding failure."""
    st, inv = _state_of(tmp_path)
    assert st.set_health("70", False, source="vfio")
    assert not st.is_healthy("70")
    # amd-smi reports healthy — vfio verdict still stands
    assert not st.set_health("70", True, source="amdsmi")
    assert not st.is_healthy("70")
    # amd-smi also fails, then vfio recovers: still unhealthy via amdsmi
    st.set_health("70", False, source="amdsmi")
    assert not st.set_health("70", True, source="vfio")
    assert not st.is_healthy("70")
    # last source clears → healthy
    assert st.set_health("70", True, source="amdsmi")
    assert st.is_healthy("70")


# --- plugin-watcher registration ------------------------------------------

def test_watcher_registration_mode(tmp_path):
    """pluginregistration.v1: kubelet-side GetInfo/NotifyRegistrationStatus
    against the plugin's registry socket (the reference supports only
    legacy self-registration)."""
    import grpc as _grpc
    import tempfile as _tf
    from kata_xpu_device_plugin_amd.plugin.watcher_registration import (
        InfoRequest, RegistrationStatus, WatcherRegistrationStub)

    node = make_mock_node(str(tmp_path), n_gpus=1)
    registry = _tf.mkdtemp(prefix="kxdp-reg-")
    cfg = node.config(registration_mode="watcher", plugins_registry_dir=registry)
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start(register=True)  # no kubelet.sock needed in watcher mode
    try:
        plugin = next(iter(mgr.plugins.values()))
        assert plugin.watcher_socket_path and os.path.exists(plugin.watcher_socket_path)
        ch = _grpc.insecure_channel(f"unix://{plugin.watcher_socket_path}")
        _grpc.channel_ready_future(ch).result(timeout=5)
        stub = WatcherRegistrationStub(ch)
        info = stub.GetInfo(InfoRequest())
        assert info.type == "DevicePlugin"
        assert info.name == "amd.com/INSTINCT_MI355X"
        assert info.endpoint == plugin.socket_path
        assert list(info.supported_versions) == ["v1beta1"]
        stub.NotifyRegistrationStatus(RegistrationStatus(plugin_registered=True))
        assert plugin.watcher_servicer.last_status == (True, "")
        # the DevicePlugin service itself is reachable on BOTH sockets
        ps = api.DevicePluginStub(ch)
        opts = ps.GetDevicePluginOptions(api.Empty())
        assert opts.get_preferred_allocation_available
        ch.close()
    finally:
        mgr.stop()


def test_watcher_wire_format():
    from kata_xpu_device_plugin_amd.plugin.watcher_registration import PluginInfo
    msg = PluginInfo(type="DevicePlugin", name="amd.com/X", endpoint="/e.sock",
                     supported_versions=["v1beta1"])
    b = msg.SerializeToString()
    def _ld(n, p):
        return bytes([(n << 3) | 2, len(p)]) + p
    assert b == (_ld(1, b"DevicePlugin") + _ld(2, b"amd.com/X")
                 + _ld(3, b"/e.sock") + _ld(4, b"v1beta1"))


# --- in-daemon GPU probe poller (faked snapshots) --------------------------

def test_gpu_probe_poller_transitions(tmp_path):
    from kata_xpu_device_plugin_amd.health.probe_poller import GpuProbePoller
    state = {"0000:0a:00.0": True}
    calls = []
    p = GpuProbePoller(999, lambda b, h: calls.append((b, h)),
                       snapshot_fn=lambda: dict(state))
    p.poll_once()
    assert calls == []
    state["0000:0a:00.0"] = False
    p.poll_once()
    p.poll_once()
    assert calls == [("0000:0a:00.0", False)]
    state["0000:0a:00.0"] = True
    p.poll_once()
    assert calls[-1] == ("0000:0a:00.0", True)


def test_gpu_probe_marks_device_unhealthy(tmp_path):
    """Probe verdict reaches DeviceState under the 'probe' source and
    composes with the vfio source."""
    node = make_mock_node(str(tmp_path), n_gpus=2, kfd=False, hint=False)
    mgr = PluginManager(node.config())
    mgr.setup()
    st = mgr.states["amd.com/INSTINCT_MI355X"]
    mgr._on_probe_health("0000:0A:00.0", False)  # case-insensitive match
    assert not st.is_healthy("70")
    st.set_health("70", False, source="vfio")
    mgr._on_probe_health("0000:0a:00.0", True)
    assert not st.is_healthy("70")  # vfio verdict still outstanding
    st.set_health("70", True, source="vfio")
    assert st.is_healthy("70")


def test_probe_snapshot_no_gpu_is_empty():
    from kata_xpu_device_plugin_amd.health.probe_poller import probe_snapshot
    assert probe_snapshot() == {}  # CPU container: 0 HIP devices


def test_registration_mode_both(tmp_path):
    """'both' serves watcher registry socket AND self-registers legacy."""
    import tempfile as _tf
    node = make_mock_node(str(tmp_path), n_gpus=1)
    registry = _tf.mkdtemp(prefix="kxdp-reg2-")
    cfg = node.config(registration_mode="both", plugins_registry_dir=registry)
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start(register=True)
    try:
        regs = stub.wait_for_registration(1)  # legacy self-registration
        assert regs[0].resource_name == "amd.com/INSTINCT_MI355X"
        plugin = next(iter(mgr.plugins.values()))
        assert os.path.exists(plugin.watcher_socket_path)  # watcher socket
    finally:
        mgr.stop()
        stub.stop()
        assert not os.path.exists(plugin.watcher_socket_path), \
            "watcher socket must be cleaned up on stop"


def test_json_log_format(capsys):
    import json as _json
    import logging
    from kata_xpu_device_plugin_amd.utils import log as kxlog
    # force a fresh handler with json format
    root = logging.getLogger("kxdp")
    for h in list(root.handlers):
        root.removeHandler(h)
    kxlog._configured = False
    kxlog.configure("INFO", fmt="json")
    kxlog.get_logger("kxdp.test").info("hello %s", "world")
    err = capsys.readouterr().err.strip().splitlines()[-1]
    doc = _json.loads(err)
    assert doc["msg"] == "hello world"
    assert doc["level"] == "INFO"
    assert doc["logger"] == "kxdp.test"
    # restore text format for other tests
    for h in list(root.handlers):
        root.removeHandler(h)
    kxlog._configured = False
    kxlog.configure("INFO")


def test_inotify_rm_watch_and_path_of(tmp_path):
    from kata_xpu_device_plugin_amd.utils import inotify
    with inotify.Inotify() as ino:
        wd = ino.add_watch(str(tmp_path), inotify.IN_CREATE)
        assert ino.path_of(wd) == str(tmp_path)
        ino.rm_watch(wd)
        assert ino.path_of(wd) is None
        (tmp_path / "x").write_text("")
        # removed watch → no events for the path (IGNORED event may appear
        # for the removed wd; just ensure no CREATE for "x")
        evs = ino.read_events(timeout=0.3)
        assert not any(e.name == "x" and e.created for e in evs)


def test_rescan_no_change_is_noop(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=2, kfd=False, hint=False)
    mgr = PluginManager(node.config())
    mgr.setup()
    path = mgr.cdi_spec_path
    mtime = os.path.getmtime(path)
    assert mgr.rescan() is False          # nothing changed
    assert os.path.getmtime(path) == mtime  # CDI not rewritten
    node.add_gpu(MockGPU(bdf="0000:77:00.0", iommu_group="88"))
    assert mgr.rescan() is True
    assert "88" in mgr.states["amd.com/INSTINCT_MI355X"].device_ids()


def test_periodic_rescan_thread(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=1, kfd=False, hint=False)
    cfg = node.config(rescan_interval_s=0.1)
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start(register=False)
    try:
        node.add_gpu(MockGPU(bdf="0000:77:00.0", iommu_group="88"))
        deadline = time.monotonic() + 5
        st = mgr.states["amd.com/INSTINCT_MI355X"]
        while "88" not in st.device_ids() and time.monotonic() < deadline:
            time.sleep(0.05)
        assert "88" in st.device_ids(), "periodic rescan must pick up new VF"
    finally:
        mgr.stop()


def test_state_watcher_queue_overflow_is_safe(tmp_path):
    """A slow ListAndWatch consumer must never block or crash health
    updates: past the queue bound, notifications are dropped (the stream
    resyncs from a full snapshot on its next wakeup)."""
    st, inv = _state_of(tmp_path, n=2)
    q = st.watch()
    for i in range(200):  # far beyond the 64-entry bound
        st.set_health("70", i % 2 == 0)
    assert q.qsize() <= 64
    # consumer drains and sees a consistent final snapshot
    while not q.empty():
        q.get_nowait()
    snap = dict((d.id, h) for d, h in st.snapshot())
    assert snap["70"] in (True, False)
    st.unwatch(q)


# --- xGMI link health → hive degradation ---------------

def test_xgmi_degradation_steers_placement(tmp_path):
    """A GPU with sick xGMI links loses hive membership in placement: a
    2-GPU pod avoids it while intact-fabric peers exist — without the GPU
    going Unhealthy."""
    from kata_xpu_device_plugin_amd.health.amdsmi_health import (
        DeviceHealth, XgmiLinkHealth)
    from kata_xpu_device_plugin_amd.topology.hive import preferred_allocation

    # two 2-GPU hives
    node = make_mock_node(str(tmp_path), n_gpus=4, hives=[[0, 1], [2, 3]],
                          kfd=False, hint=True)
    mgr = PluginManager(node.config())
    mgr.setup()
    inv, topo = mgr.inventory, mgr.topology
    bdfs = sorted(d.primary.bdf for d in inv.devices.values())

    def pick2():
        return sorted(preferred_allocation(
            topo, inv, sorted(inv.devices), [], 2))

    healthy_pick = pick2()
    assert len(healthy_pick) == 2

    # GPU 0 (hive 1) reports link errors → degrade
    sick_bdf = bdfs[0]
    snap = {sick_bdf: DeviceHealth(
        bdf=sick_bdf,
        xgmi_links=[XgmiLinkHealth(index=0, status="up", errors=3)])}
    mgr._on_xgmi_telemetry(snap)
    assert topo.degraded == frozenset({sick_bdf})

    pick = pick2()
    chosen_bdfs = {inv.devices[g].primary.bdf for g in pick}
    assert sick_bdf not in chosen_bdfs, \
        "placement must avoid the degraded-fabric GPU"
    # health is NOT affected — the GPU still allocates
    for st_ in mgr.states.values():
        assert all(h for _, h in st_.snapshot())

    # recovery restores the hive
    mgr._on_xgmi_telemetry({sick_bdf: DeviceHealth(bdf=sick_bdf)})
    assert topo.degraded == frozenset()


def test_xgmi_metrics_exported(tmp_path):
    from prometheus_client import generate_latest
    from kata_xpu_device_plugin_amd.metrics import MetricsExporter
    from kata_xpu_device_plugin_amd.health.amdsmi_health import (
        AmdSmiPoller, DeviceHealth, XgmiLinkHealth)

    node = make_mock_node(str(tmp_path), n_gpus=1, kfd=False, hint=False)
    mgr = PluginManager(node.config())
    mgr.setup()
    snap = {"0000:0a:00.0": DeviceHealth(
        bdf="0000:0a:00.0",
        xgmi_links=[XgmiLinkHealth(index=0, status="up", errors=0),
                    XgmiLinkHealth(index=1, status="down", errors=2)])}
    mgr._amdsmi = AmdSmiPoller(999, lambda *a: None,
                               snapshot_fn=lambda: snap,
                               on_xgmi=mgr._on_xgmi_telemetry)
    mgr._amdsmi.poll_once()
    exp = MetricsExporter(mgr)
    text = generate_latest(exp.registry).decode()
    assert 'kxdp_xgmi_link_up{bdf="0000:0a:00.0",link="0"} 1.0' in text
    assert 'kxdp_xgmi_link_up{bdf="0000:0a:00.0",link="1"} 0.0' in text
    assert "# TYPE kxdp_xgmi_link_errors_total counter" in text
    assert 'kxdp_xgmi_link_errors_total{bdf="0000:0a:00.0",link="1"} 2.0' in text
    assert 'kxdp_xgmi_degraded{bdf="0000:0a:00.0"} 1.0' in text


def test_xgmi_degradation_survives_rescan(tmp_path):
    from kata_xpu_device_plugin_amd.health.amdsmi_health import (
        DeviceHealth, XgmiLinkHealth)
    node = make_mock_node(str(tmp_path), n_gpus=2, kfd=False, hint=True)
    mgr = PluginManager(node.config())
    mgr.setup()
    bdf = sorted(d.primary.bdf for d in mgr.inventory.devices.values())[0]
    mgr._on_xgmi_telemetry({bdf: DeviceHealth(
        bdf=bdf, xgmi_links=[XgmiLinkHealth(index=0, errors=1)])})
    assert bdf in mgr.topology.degraded
    node.add_gpu(MockGPU(bdf="0000:66:00.0", iommu_group="99"))
    assert mgr.rescan() is True
    assert bdf in mgr.topology.degraded, "degradation lost across rescan"


def test_ecc_block_attribution_metrics(tmp_path):
    from prometheus_client import generate_latest
    from kata_xpu_device_plugin_amd.metrics import MetricsExporter
    from kata_xpu_device_plugin_amd.health.amdsmi_health import (
        AmdSmiPoller, DeviceHealth)

    node = make_mock_node(str(tmp_path), n_gpus=1, kfd=False, hint=False)
    mgr = PluginManager(node.config())
    mgr.setup()
    snap = {"0000:0a:00.0": DeviceHealth(
        bdf="0000:0a:00.0", healthy=False,
        reasons=["2 uncorrectable in UMC"],
        ecc_by_block={"UMC": 2, "GFX": 0})}
    mgr._amdsmi = AmdSmiPoller(999, lambda *a: None, snapshot_fn=lambda: snap)
    mgr._amdsmi.poll_once()
    text = generate_latest(MetricsExporter(mgr).registry).decode()
    assert "# TYPE kxdp_ecc_uncorrectable_total counter" in text
    assert 'kxdp_ecc_uncorrectable_total{bdf="0000:0a:00.0",block="UMC"} 2.0' in text
    assert 'kxdp_ecc_uncorrectable_total{bdf="0000:0a:00.0",block="GFX"} 0.0' in text


def test_rpc_timing_breakdown(tmp_path):
    """KXDP_RPC_TIMING: per-RPC handler durations + loop lag dumped as a
    JSON distribution on shutdown (tail attribution)."""
    import json, glob
    node = make_mock_node(str(tmp_path), n_gpus=2)
    cfg = node.config(rpc_timing_path=os.path.join(str(tmp_path), "timing"))
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start()
    try:
        reg = stub.wait_for_registration(1)[0]
        ps = stub.plugin_stub(reg.endpoint)
        for _ in range(5):
            pref = ps.GetPreferredAllocation(api.PreferredAllocationRequest(
                container_requests=[api.ContainerPreferredAllocationRequest(
                    available_device_ids=["70", "71"], allocation_size=1)]))
            ps.Allocate(api.AllocateRequest(container_requests=[
                api.ContainerAllocateRequest(
                    devices_ids=list(pref.container_responses[0].device_ids))]))
        time.sleep(0.05)
    finally:
        mgr.stop()
        stub.stop()
    files = glob.glob(os.path.join(str(tmp_path), "timing.*.json"))
    assert files, "timing dump missing"
    doc = json.load(open(files[0]))
    assert doc["allocate_us"]["n"] == 5
    assert doc["preferred_us"]["n"] == 5
    assert doc["allocate_us"]["p50"] > 0
    assert doc["loop_lag_us"] is None or doc["loop_lag_us"]["n"] >= 1


def test_cpu_affinity_parse_and_apply(tmp_path):
    """cpu_affinity spec parsing ("0", "0-1", "0,2") + application via
    sched_setaffinity; invalid specs are logged, never fatal."""
    before = os.sched_getaffinity(0)
    try:
        PluginManager._pin_cpus("0")
        assert os.sched_getaffinity(0) == {0}
        all_spec = ",".join(str(c) for c in sorted(before))
        PluginManager._pin_cpus(all_spec)
        assert os.sched_getaffinity(0) == before
        PluginManager._pin_cpus("not-a-cpu")   # must not raise
        PluginManager._pin_cpus("")            # no-op
        assert os.sched_getaffinity(0) == before
    finally:
        os.sched_setaffinity(0, before)


def test_metrics_scrape_concurrent_with_rescan(tmp_path):
    """/metrics scrapes race rescan's plugin-dict mutation (dynamic
    start/retire): no 'dict changed size' crashes, scrape always returns."""
    from prometheus_client import generate_latest
    from kata_xpu_device_plugin_amd.metrics import MetricsExporter

    node = make_mock_node(str(tmp_path), n_gpus=2, kfd=False, hint=True)
    mgr = PluginManager(node.config())
    mgr.setup()
    exp = MetricsExporter(mgr)
    stop = threading.Event()
    errs = []

    def scraper():
        try:
            while not stop.is_set():
                assert b"kxdp_devices" in generate_latest(exp.registry)
        except Exception as e:  # pragma: no cover
            errs.append(e)

    t = threading.Thread(target=scraper)
    t.start()
    try:
        for cycle in range(6):
            for k in range(2):
                node.add_gpu(MockGPU(bdf=f"0000:61:02.{k}", device_id=0x75B3,
                                     iommu_group=str(140 + k),
                                     physfn_bdf="0000:61:00.0"))
            mgr.rescan()
            for k in range(2):
                node.remove_gpu(f"0000:61:02.{k}")
            mgr.rescan()
    finally:
        stop.set()
        t.join(timeout=5)
        mgr.stop()
    assert not errs
