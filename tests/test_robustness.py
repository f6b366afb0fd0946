"""Robustness / end-to-end edge cases: multi-model nodes, concurrent
health flapping under churn, rescans, multi-container Allocate."""
import os
import threading
import time

import grpc
import pytest

from kata_xpu_device_plugin_amd.cdi.spec import read_spec
from kata_xpu_device_plugin_amd.plugin import api
from kata_xpu_device_plugin_amd.plugin.manager import PluginManager
from kata_xpu_device_plugin_amd.testing.kubelet_stub import KubeletStub
from kata_xpu_device_plugin_amd.testing.mocknode import MockGPU, make_mock_node


@pytest.fixture
def multi_model(tmp_path):
    """Node with 6 MI355X + 2 MI300X (two resource names → two plugins,
    like the reference's per-model plugin split, device_plugin.go:83-119)."""
    node = make_mock_node(str(tmp_path), n_gpus=6, kfd=False, hint=False)
    for k in range(2):
        node.add_gpu(MockGPU(bdf=f"0000:e{k}:00.0", device_id=0x74A1,
                             iommu_group=str(90 + k), numa_node=1))
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start(register=True)
    yield node, cfg, stub, mgr
    mgr.stop()
    stub.stop()


def test_two_models_two_plugins(multi_model):
    node, cfg, stub, mgr = multi_model
    regs = stub.wait_for_registration(2)
    names = sorted(r.resource_name for r in regs)
    assert names == ["amd.com/INSTINCT_MI300X", "amd.com/INSTINCT_MI355X"]
    # distinct sockets
    assert len({r.endpoint for r in regs}) == 2
    by_name = {r.resource_name: r for r in regs}
    ps300 = stub.plugin_stub(by_name["amd.com/INSTINCT_MI300X"].endpoint)
    stream = ps300.ListAndWatch(api.Empty())
    devs = next(stream).devices
    assert [d.id for d in devs] == ["90", "91"]
    stream.cancel()
    # cross-model allocate must fail (id belongs to the other plugin)
    with pytest.raises(grpc.RpcError):
        ps300.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["70"])]))


def test_multi_container_request(multi_model):
    node, cfg, stub, mgr = multi_model
    regs = stub.wait_for_registration(2)
    r = next(x for x in regs if x.resource_name.endswith("MI355X"))
    ps = stub.plugin_stub(r.endpoint)
    resp = ps.Allocate(api.AllocateRequest(container_requests=[
        api.ContainerAllocateRequest(devices_ids=["70", "71"]),
        api.ContainerAllocateRequest(devices_ids=["72"]),
    ]))
    assert len(resp.container_responses) == 2
    assert len(resp.container_responses[0].cdi_devices) == 2
    assert len(resp.container_responses[1].cdi_devices) == 1


def test_cdi_spec_covers_all_models(multi_model):
    node, cfg, stub, mgr = multi_model
    spec = read_spec(mgr.cdi_spec_path)
    assert sorted(spec.device_names(), key=int) == [
        "70", "71", "72", "73", "74", "75", "90", "91"]


def test_rescan_rewrites_cdi(multi_model):
    node, cfg, stub, mgr = multi_model
    node.add_gpu(MockGPU(bdf="0000:f5:00.0", iommu_group="95"))
    mgr.rescan()
    spec = read_spec(mgr.cdi_spec_path)
    assert "95" in spec.device_names()
    # served immediately on the existing plugin
    st = mgr.states["amd.com/INSTINCT_MI355X"]
    assert "95" in st.device_ids()


def test_health_flap_during_churn(tmp_path):
    """Concurrent Allocate churn + vfio node flapping: no crashes, flapped
    device rejected only while its sysfs state is intact (health is a
    ListAndWatch concern; Allocate revalidates sysfs, not /dev)."""
    node = make_mock_node(str(tmp_path), n_gpus=4)
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start()
    try:
        r = stub.wait_for_registration(1)[0]
        ps = stub.plugin_stub(r.endpoint)
        stream = ps.ListAndWatch(api.Empty())
        next(stream)
        stop = threading.Event()
        errs = []

        def churn():
            try:
                while not stop.is_set():
                    ps.Allocate(api.AllocateRequest(container_requests=[
                        api.ContainerAllocateRequest(devices_ids=["70"])]))
            except Exception as e:  # pragma: no cover
                errs.append(e)

        t = threading.Thread(target=churn)
        t.start()
        seen_unhealthy = seen_healthy = False
        for _ in range(6):
            node.remove_vfio_node("72")
            time.sleep(0.05)
            node.add_vfio_node("72")
            time.sleep(0.05)
        deadline = time.monotonic() + 5
        while time.monotonic() < deadline and not (seen_unhealthy and seen_healthy):
            upd = next(stream)
            h = {d.id: d.health for d in upd.devices}
            if h.get("72") == api.UNHEALTHY:
                seen_unhealthy = True
            if seen_unhealthy and h.get("72") == api.HEALTHY:
                seen_healthy = True
        stop.set()
        t.join()
        stream.cancel()
        assert not errs
        assert seen_unhealthy and seen_healthy
    finally:
        mgr.stop()
        stub.stop()


def test_plugin_restart_keeps_serving(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=2)
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start()
    try:
        r = stub.wait_for_registration(1)[0]
        plugin = next(iter(mgr.plugins.values()))
        for _ in range(3):
            plugin.restart()
        regs = stub.wait_for_registration(4)  # initial + 3 restarts
        assert len(regs) >= 4
        # fresh channel post-restart
        ch = grpc.insecure_channel(f"unix://{plugin.socket_path}")
        grpc.channel_ready_future(ch).result(timeout=5)
        ps = api.DevicePluginStub(ch)
        resp = ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["70"])]))
        assert resp.container_responses[0].cdi_devices[0].name == "amd.com/gpu=70"
        ch.close()
    finally:
        mgr.stop()
        stub.stop()


def test_allocate_empty_request(tmp_path):
    node = make_mock_node(str(tmp_path), n_gpus=1)
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start()
    try:
        r = stub.wait_for_registration(1)[0]
        ps = stub.plugin_stub(r.endpoint)
        resp = ps.Allocate(api.AllocateRequest())
        assert len(resp.container_responses) == 0
        resp = ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest()]))
        assert len(resp.container_responses) == 1
        assert len(resp.container_responses[0].cdi_devices) == 0
    finally:
        mgr.stop()
        stub.stop()


def test_cdi_spec_self_heal(tmp_path):
    """Deleting the CDI spec file regenerates it (runtime resolution must
    never dangle)."""
    node = make_mock_node(str(tmp_path), n_gpus=2)
    cfg = node.config()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start(register=False)
    try:
        path = mgr.cdi_spec_path
        assert os.path.exists(path)
        os.unlink(path)
        deadline = time.monotonic() + 5
        while not os.path.exists(path) and time.monotonic() < deadline:
            time.sleep(0.05)
        assert os.path.exists(path), "CDI spec must be regenerated"
        spec = read_spec(path)
        assert spec.device_names() == ["70", "71"]
    finally:
        mgr.stop()


def test_mixed_pf_vf_node(tmp_path):
    """Hybrid node: 4 whole GPUs vfio-bound for passthrough + 1 PF kept on
    amdgpu whose 4 VFs are vfio-bound — two resources, correct split
    (BASELINE.json config #5 shape)."""
    node = make_mock_node(str(tmp_path), n_gpus=4, kfd=False, hint=False)
    # PF on amdgpu (not schedulable itself)
    node.add_gpu(MockGPU(bdf="0000:60:00.0", driver="amdgpu", iommu_group="50",
                         sriov_totalvfs=4))
    for k in range(4):
        node.add_gpu(MockGPU(bdf=f"0000:60:02.{k}", device_id=0x75B3,
                             iommu_group=str(110 + k), physfn_bdf="0000:60:00.0"))
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start()
    try:
        regs = stub.wait_for_registration(2)
        names = sorted(r.resource_name for r in regs)
        assert names == ["amd.com/INSTINCT_MI355X", "amd.com/INSTINCT_MI355X_VF"]
        vf_state = mgr.states["amd.com/INSTINCT_MI355X_VF"]
        assert vf_state.device_ids() == ["110", "111", "112", "113"]
        # PF group 50 is not schedulable anywhere
        for st_ in mgr.states.values():
            assert st_.device(str(50)) is None
        # allocate a VF end-to-end
        r = next(x for x in regs if x.resource_name.endswith("_VF"))
        ps = stub.plugin_stub(r.endpoint)
        resp = ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["110"])]))
        assert resp.container_responses[0].cdi_devices[0].name == "amd.com/gpu=110"
    finally:
        mgr.stop()
        stub.stop()


def test_rescan_serves_new_resource_name_without_restart(tmp_path):
    """First-time SR-IOV enable creates a brand-new *_VF resource name:
    rescan() must create, start and register its plugin in place — and
    retire it again when the VFs disappear (round 1
    logged 'restart the daemon' instead)."""
    node = make_mock_node(str(tmp_path), n_gpus=2, kfd=False, hint=True)
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start(register=True)
    try:
        stub.wait_for_registration(1)
        assert sorted(mgr.plugins) == ["amd.com/INSTINCT_MI355X"]

        # VFs appear (e.g. tools/sriov enable): new resource name.
        for k in range(2):
            node.add_gpu(MockGPU(bdf=f"0000:61:02.{k}", device_id=0x75B3,
                                 iommu_group=str(120 + k),
                                 physfn_bdf="0000:61:00.0"))
        assert mgr.rescan() is True
        regs = stub.wait_for_registration(2)
        names = sorted(r.resource_name for r in regs)
        assert names == ["amd.com/INSTINCT_MI355X", "amd.com/INSTINCT_MI355X_VF"]
        assert sorted(mgr.plugins) == names

        # The dynamic plugin serves for real: Allocate a VF end-to-end.
        vf_reg = next(r for r in regs if r.resource_name.endswith("_VF"))
        ps = stub.plugin_stub(vf_reg.endpoint)
        resp = ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["120"])]))
        assert resp.container_responses[0].cdi_devices[0].name == "amd.com/gpu=120"

        # Topology was reloaded (advisor r1): the VFs have locality — both
        # share their PF's (pseudo-)hive, so a 2-VF pod scores as one hive.
        assert mgr.topology.hive_of.get("0000:61:02.0") is not None
        assert mgr.topology.hive_of["0000:61:02.0"] == \
            mgr.topology.hive_of["0000:61:02.1"]
        pref = ps.GetPreferredAllocation(api.PreferredAllocationRequest(
            container_requests=[api.ContainerPreferredAllocationRequest(
                available_device_ids=["120", "121"], allocation_size=2)]))
        assert sorted(pref.container_responses[0].device_ids) == ["120", "121"]

        # VFs disappear again → the *_VF plugin is retired.
        vf_socket = mgr.plugins["amd.com/INSTINCT_MI355X_VF"].socket_path
        for k in range(2):
            node.remove_gpu(f"0000:61:02.{k}")
        assert mgr.rescan() is True
        assert sorted(mgr.plugins) == ["amd.com/INSTINCT_MI355X"]
        assert not os.path.exists(vf_socket)
        # PF resource keeps serving untouched
        pf_reg = next(r for r in regs if not r.resource_name.endswith("_VF"))
        ps_pf = stub.plugin_stub(pf_reg.endpoint)
        assert len(ps_pf.GetPreferredAllocation(api.PreferredAllocationRequest(
            container_requests=[api.ContainerPreferredAllocationRequest(
                available_device_ids=["70", "71"], allocation_size=1)]
        )).container_responses[0].device_ids) == 1
    finally:
        mgr.stop()
        stub.stop()


def _wait_until(pred, timeout=5.0, what="condition"):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if pred():
            return
        time.sleep(0.05)
    raise AssertionError(f"timed out waiting for {what}")


def test_watcher_kubelet_dir_created_after_start(tmp_path):
    """a kubelet dir created AFTER the watcher started
    must still be watched — a subsequent socket wipe triggers the restart
    callback, and kubelet.sock creation triggers re-registration. Round 1
    armed the watch only if the dir existed at startup."""
    from kata_xpu_device_plugin_amd.config import Config
    from kata_xpu_device_plugin_amd.health.watcher import NodeWatcher

    kdir = os.path.join(str(tmp_path), "var", "lib", "kubelet",
                        "device-plugins")
    cfg = Config(
        sysfs_root=os.path.join(str(tmp_path), "sys"),
        dev_root=os.path.join(str(tmp_path), "dev"),
        cdi_dir=os.path.join(str(tmp_path), "cdi"),
        kubelet_socket_dir=kdir,
        pci_ids_paths=(),
    )
    removed, restarted = [], []
    w = NodeWatcher(
        cfg, {},
        on_socket_removed=removed.append,
        on_kubelet_restarted=lambda: restarted.append(1),
        plugin_socket_names={"kxdp-test.sock"},
    )
    w.start()
    w.wait_ready()
    try:
        assert not os.path.isdir(kdir)
        os.makedirs(kdir)                      # kubelet dir appears late
        _wait_until(lambda: "kubelet" in w._armed, what="late dir armed")
        # (arming a late dir may fire one reconcile restart for the
        # not-yet-recreated socket — correct behavior; count from here)
        sock = os.path.join(kdir, "kxdp-test.sock")
        open(sock, "w").close()
        time.sleep(0.3)
        base = len(removed)
        os.unlink(sock)                        # the wipe we must catch
        _wait_until(lambda: len(removed) > base and
                    removed[-1] == "kxdp-test.sock",
                    what="socket-wipe callback")
        open(os.path.join(kdir, "kubelet.sock"), "w").close()
        _wait_until(lambda: restarted, what="kubelet-restart callback")
    finally:
        w.stop()


def test_kubelet_dir_wipe_full_recovery(tmp_path):
    """The whole kubelet dir vanishes mid-run (kubelet reinstall; modeled
    as an atomic rename — inotify watches follow the inode, so round 1's
    watcher silently kept watching the MOVED dir): the plugin restarts,
    recreates dir + socket, and the RE-ARMED watch still catches a later
    plain socket wipe."""
    node = make_mock_node(str(tmp_path), n_gpus=1)
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start(register=True)
    plugin = next(iter(mgr.plugins.values()))
    try:
        stub.wait_for_registration(1)
        stub.stop()                       # free kubelet.sock before the wipe
        os.rename(cfg.kubelet_socket_dir, cfg.kubelet_socket_dir + ".old")
        _wait_until(lambda: plugin.serving and
                    os.path.exists(plugin.socket_path), timeout=10,
                    what="socket recreation after dir wipe")
        time.sleep(0.8)                   # watcher re-armed on the new dir
        os.unlink(plugin.socket_path)     # later plain wipe must be caught
        _wait_until(lambda: plugin.serving and
                    os.path.exists(plugin.socket_path), timeout=10,
                    what="restart after post-wipe socket removal")
    finally:
        mgr.stop()


def test_cdi_dir_wipe_selfheal(tmp_path):
    """The entire CDI dir vanishes (tmp-cleaner; atomic rename so the wipe
    cannot race the self-heal): dir AND spec are regenerated. Round 1 only
    healed a removed spec FILE inside a surviving dir."""
    import shutil
    node = make_mock_node(str(tmp_path), n_gpus=2)
    cfg = node.config()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start(register=False)
    try:
        path = mgr.cdi_spec_path
        assert path and os.path.exists(path)
        os.rename(cfg.cdi_dir, cfg.cdi_dir + ".old")
        shutil.rmtree(cfg.cdi_dir + ".old")
        _wait_until(lambda: os.path.exists(path), timeout=6.0,
                    what="CDI spec regeneration after dir wipe")
        spec = read_spec(path)
        assert spec.device_names() == ["70", "71"]
    finally:
        mgr.stop()


def test_daemon_survives_missing_kubelet(tmp_path):
    """kubelet not up at daemon start (node-boot race): the plugin must
    serve anyway and register as soon as kubelet appears."""
    node = make_mock_node(str(tmp_path), n_gpus=1)
    cfg = node.config()
    cfg.grpc_timeout_s = 0.5  # fail the initial dial fast
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start(register=True)  # no kubelet.sock exists yet
    stub = None
    try:
        plugin = next(iter(mgr.plugins.values()))
        assert plugin.serving, "plugin must serve despite missing kubelet"
        # kubelet comes up late
        stub = KubeletStub(cfg.kubelet_socket_dir)
        stub.start()
        regs = stub.wait_for_registration(1, timeout=15)
        assert regs[0].resource_name == "amd.com/INSTINCT_MI355X"
    finally:
        mgr.stop()
        if stub:
            stub.stop()


def test_daemon_main_sighup_rescan(tmp_path):
    """SIGHUP triggers a live rescan in the real daemon process: a GPU
    added after startup becomes allocatable without restart."""
    import signal
    import subprocess
    import sys

    node = make_mock_node(str(tmp_path), n_gpus=2)
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    env = dict(os.environ,
               KXDP_SYSFS_ROOT=cfg.sysfs_root,
               KXDP_DEV_ROOT=cfg.dev_root,
               KXDP_CDI_DIR=cfg.cdi_dir,
               KXDP_KUBELET_DIR=cfg.kubelet_socket_dir,
               KXDP_TOPOLOGY_HINT=cfg.topology_hint_path,
               KXDP_AMDSMI_HEALTH="false",
               PYTHONPATH=os.path.dirname(os.path.dirname(
                   os.path.abspath(__file__))))
    proc = subprocess.Popen(
        [sys.executable, "-m", "kata_xpu_device_plugin_amd"],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True)
    try:
        regs = stub.wait_for_registration(1, timeout=20)
        ps = stub.plugin_stub(regs[0].endpoint)
        stream = ps.ListAndWatch(api.Empty())
        assert len(next(stream).devices) == 2

        node.add_gpu(MockGPU(bdf="0000:77:00.0", iommu_group="99"))
        proc.send_signal(signal.SIGHUP)
        upd = next(stream)   # rescan pushes the new inventory
        assert sorted(d.id for d in upd.devices) == ["70", "71", "99"]
        stream.cancel()
        resp = ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["99"])]))
        assert resp.container_responses[0].cdi_devices[0].name == "amd.com/gpu=99"
        proc.send_signal(signal.SIGTERM)
        assert proc.wait(timeout=15) == 0
    finally:
        if proc.poll() is None:
            proc.kill()
        stub.stop()


def test_daemon_main_sigterm(tmp_path):
    """End-to-end daemon process: `python -m kata_xpu_device_plugin_amd`
    starts against a mock node, serves, and shuts down cleanly on SIGTERM
    (the reference has no signal handling — device_plugin.go:114)."""
    import signal
    import subprocess
    import sys

    node = make_mock_node(str(tmp_path), n_gpus=2)
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    env = dict(os.environ,
               KXDP_SYSFS_ROOT=cfg.sysfs_root,
               KXDP_DEV_ROOT=cfg.dev_root,
               KXDP_CDI_DIR=cfg.cdi_dir,
               KXDP_KUBELET_DIR=cfg.kubelet_socket_dir,
               KXDP_TOPOLOGY_HINT=cfg.topology_hint_path,
               KXDP_AMDSMI_HEALTH="false",
               PYTHONPATH=os.path.dirname(os.path.dirname(
                   os.path.abspath(__file__))))
    proc = subprocess.Popen(
        [sys.executable, "-m", "kata_xpu_device_plugin_amd"],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True)
    try:
        regs = stub.wait_for_registration(1, timeout=20)
        assert regs[0].resource_name == "amd.com/INSTINCT_MI355X"
        ps = stub.plugin_stub(regs[0].endpoint)
        resp = ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["70"])]))
        assert resp.container_responses[0].cdi_devices[0].name == "amd.com/gpu=70"
        proc.send_signal(signal.SIGTERM)
        rc = proc.wait(timeout=15)
        assert rc == 0, proc.stderr.read()[-2000:]
        # socket cleaned up on shutdown
        assert not os.path.exists(
            os.path.join(cfg.kubelet_socket_dir, regs[0].endpoint))
    finally:
        if proc.poll() is None:
            proc.kill()
        stub.stop()


def test_shutdown_under_load(tmp_path):
    """Stopping the daemon while health flaps and allocations are in
    flight must be clean: no hangs, no exceptions escaping, bounded time."""
    node = make_mock_node(str(tmp_path), n_gpus=4)
    cfg = node.config()
    stub = KubeletStub(cfg.kubelet_socket_dir)
    stub.start()
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start()
    r = stub.wait_for_registration(1)[0]
    ps = stub.plugin_stub(r.endpoint)
    stream = ps.ListAndWatch(api.Empty())
    next(stream)
    stop = threading.Event()
    errs = []

    def churn():
        while not stop.is_set():
            try:
                ps.Allocate(api.AllocateRequest(container_requests=[
                    api.ContainerAllocateRequest(devices_ids=["70"])]))
            except grpc.RpcError:
                return  # server going down: expected
            except Exception as e:  # pragma: no cover
                errs.append(e)
                return

    def flap():
        while not stop.is_set():
            node.remove_vfio_node("72")
            node.add_vfio_node("72")
            time.sleep(0.01)

    threads = [threading.Thread(target=churn) for _ in range(3)]
    threads.append(threading.Thread(target=flap))
    for t in threads:
        t.start()
    time.sleep(0.4)
    t0 = time.monotonic()
    mgr.stop()          # under full load
    elapsed = time.monotonic() - t0
    stop.set()
    for t in threads:
        t.join(timeout=5)
        assert not t.is_alive()
    stub.stop()
    assert not errs
    assert elapsed < 10, f"shutdown took {elapsed:.1f}s"


def test_watcher_mode_kubelet_flow(tmp_path):
    """Modern-kubelet flow end-to-end: watch the plugins_registry dir,
    discover the registration socket, GetInfo → dial the advertised
    endpoint → full DevicePlugin round-trip → NotifyRegistrationStatus."""
    import tempfile
    from kata_xpu_device_plugin_amd.plugin.watcher_registration import (
        InfoRequest, RegistrationStatus, WatcherRegistrationStub)
    from kata_xpu_device_plugin_amd.utils import inotify

    node = make_mock_node(str(tmp_path), n_gpus=2)
    registry = tempfile.mkdtemp(prefix="kxdp-reg3-")
    cfg = node.config(registration_mode="watcher",
                     plugins_registry_dir=registry)
    ino = inotify.Inotify()
    ino.add_watch(registry, inotify.IN_CREATE)
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start(register=True)
    try:
        # kubelet's plugin watcher sees the socket appear
        deadline = time.monotonic() + 5
        seen = set()
        while time.monotonic() < deadline and not seen:
            seen = {e.name for e in ino.read_events(timeout=0.5) if e.created}
        plugin = next(iter(mgr.plugins.values()))
        assert plugin.socket_name in seen, seen
        # GetInfo on the registry socket
        ch = grpc.insecure_channel(
            f"unix://{os.path.join(registry, plugin.socket_name)}")
        grpc.channel_ready_future(ch).result(timeout=5)
        info = WatcherRegistrationStub(ch).GetInfo(InfoRequest())
        assert info.type == "DevicePlugin"
        # dial the advertised endpoint, full DevicePlugin round-trip
        ch2 = grpc.insecure_channel(f"unix://{info.endpoint}")
        grpc.channel_ready_future(ch2).result(timeout=5)
        ps = api.DevicePluginStub(ch2)
        stream = ps.ListAndWatch(api.Empty())
        assert len(next(stream).devices) == 2
        stream.cancel()
        resp = ps.Allocate(api.AllocateRequest(container_requests=[
            api.ContainerAllocateRequest(devices_ids=["70"])]))
        assert resp.container_responses[0].cdi_devices[0].name == "amd.com/gpu=70"
        WatcherRegistrationStub(ch).NotifyRegistrationStatus(
            RegistrationStatus(plugin_registered=True))
        assert plugin.watcher_servicer.last_status == (True, "")
        ch.close()
        ch2.close()
    finally:
        ino.close()
        mgr.stop()


def test_rescan_dynamic_plugin_watcher_mode(tmp_path):
    """Dynamic rescan in registration_mode=watcher: the new *_VF plugin
    exposes its pluginregistration socket under plugins_registry/ (the
    modern-kubelet discovery path needs no kubelet.sock registration)."""
    import tempfile as _tf
    node = make_mock_node(str(tmp_path), n_gpus=1, kfd=False, hint=True)
    registry = _tf.mkdtemp(prefix="kxdp-reg-")
    cfg = node.config(registration_mode="watcher",
                      plugins_registry_dir=registry)
    mgr = PluginManager(cfg)
    mgr.setup()
    mgr.start(register=True)
    try:
        socks = set(os.listdir(registry))
        assert len(socks) == 1
        for k in range(2):
            node.add_gpu(MockGPU(bdf=f"0000:61:02.{k}", device_id=0x75B3,
                                 iommu_group=str(130 + k),
                                 physfn_bdf="0000:61:00.0"))
        assert mgr.rescan() is True
        new_socks = set(os.listdir(registry)) - socks
        assert len(new_socks) == 1
        assert "_vf" in next(iter(new_socks))
        # the registry socket serves the pluginregistration service
        import grpc as _grpc
        from kata_xpu_device_plugin_amd.plugin.watcher_registration import (
            InfoRequest, WatcherRegistrationStub)
        ch = _grpc.insecure_channel(
            f"unix://{os.path.join(registry, next(iter(new_socks)))}")
        info = WatcherRegistrationStub(ch).GetInfo(InfoRequest(), timeout=5)
        assert info.name == "amd.com/INSTINCT_MI355X_VF"
        ch.close()
    finally:
        mgr.stop()
