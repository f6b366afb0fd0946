"""Experimental DRA plugin tests: prepare/unprepare round-trips over real
sockets, idempotency, error paths, ResourceSlice shape."""
import asyncio
import os
import tempfile
import threading
import time

import grpc
import grpc.aio
import pytest

from kata_xpu_device_plugin_amd.discovery import scan_node
from kata_xpu_device_plugin_amd.plugin import dra
from kata_xpu_device_plugin_amd.testing.mocknode import make_mock_node
from kata_xpu_device_plugin_amd.topology.hive import load_topology


@pytest.fixture
def dra_setup(tmp_path):
    node = make_mock_node(str(tmp_path))
    cfg = node.config()
    inv = scan_node(cfg)
    allocations = {"uid-1": ["70", "71"], "uid-2": ["74"]}
    servicer = dra.DRAServicer(cfg, inv, pool_name="mi355x",
                               resolver=allocations.get)

    sock = os.path.join(tempfile.mkdtemp(prefix="kxdp-dra-"), "dra.sock")
    ready = threading.Event()
    stop_evt = {}

    def serve():
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)

        async def main():
            server = grpc.aio.server()
            dra.add_dra_servicer(server, servicer)
            server.add_insecure_port(f"unix://{sock}")
            await server.start()
            stop_evt["ev"] = asyncio.Event()
            ready.set()
            await stop_evt["ev"].wait()
            await server.stop(0.2)

        loop.run_until_complete(main())
        loop.close()

    t = threading.Thread(target=serve, daemon=True)
    t.start()
    assert ready.wait(5)
    ch = grpc.insecure_channel(f"unix://{sock}")
    grpc.channel_ready_future(ch).result(timeout=5)
    yield node, cfg, inv, servicer, dra.DRAStub(ch)
    ch.close()
    loop_ev = stop_evt.get("ev")
    if loop_ev is not None:
        loop_ev._loop.call_soon_threadsafe(loop_ev.set)
    t.join(timeout=5)


def test_prepare_returns_cdi_ids(dra_setup):
    node, cfg, inv, servicer, stub = dra_setup
    resp = stub.NodePrepareResources(dra.NodePrepareResourcesRequest(claims=[
        dra.Claim(namespace="ml", uid="uid-1", name="train-claim"),
    ]))
    r = resp.claims["uid-1"]
    assert r.error == ""
    assert [d.device_name for d in r.devices] == ["70", "71"]
    assert [list(d.cdi_device_ids) for d in r.devices] == \
        [["amd.com/gpu=70"], ["amd.com/gpu=71"]]
    assert all(d.pool_name == "mi355x" for d in r.devices)


def test_prepare_idempotent(dra_setup):
    node, cfg, inv, servicer, stub = dra_setup
    req = dra.NodePrepareResourcesRequest(claims=[
        dra.Claim(namespace="ml", uid="uid-2", name="c")])
    r1 = stub.NodePrepareResources(req).claims["uid-2"]
    r2 = stub.NodePrepareResources(req).claims["uid-2"]
    assert [d.device_name for d in r1.devices] == \
        [d.device_name for d in r2.devices] == ["74"]
    assert servicer.store.get("uid-2") == ["74"]


def test_prepare_unknown_claim_errors(dra_setup):
    node, cfg, inv, servicer, stub = dra_setup
    resp = stub.NodePrepareResources(dra.NodePrepareResourcesRequest(claims=[
        dra.Claim(namespace="ml", uid="ghost", name="g")]))
    assert "no allocation known" in resp.claims["ghost"].error


def test_prepare_foreign_device_errors(dra_setup):
    node, cfg, inv, servicer, stub = dra_setup
    servicer.resolver = lambda uid: ["999"]
    resp = stub.NodePrepareResources(dra.NodePrepareResourcesRequest(claims=[
        dra.Claim(namespace="ml", uid="uid-x", name="x")]))
    assert "not on this node" in resp.claims["uid-x"].error
    assert servicer.store.get("uid-x") is None


def test_unprepare(dra_setup):
    node, cfg, inv, servicer, stub = dra_setup
    stub.NodePrepareResources(dra.NodePrepareResourcesRequest(claims=[
        dra.Claim(namespace="ml", uid="uid-1", name="c")]))
    assert servicer.store.get("uid-1") is not None
    resp = stub.NodeUnprepareResources(dra.NodeUnprepareResourcesRequest(claims=[
        dra.Claim(namespace="ml", uid="uid-1", name="c"),
        dra.Claim(namespace="ml", uid="never-prepared", name="n"),
    ]))
    assert resp.claims["uid-1"].error == ""
    assert resp.claims["never-prepared"].error == ""  # safe for unknown
    assert servicer.store.get("uid-1") is None


def test_resource_slice_shape(tmp_path):
    node = make_mock_node(str(tmp_path), hives=[[0, 1, 2, 3], [4, 5, 6, 7]])
    cfg = node.config()
    inv = scan_node(cfg)
    topo = load_topology(cfg, inv)
    obj = dra.resource_slice_obj(inv, topo, "node-a", "mi355x")
    assert obj["spec"]["driver"] == "gpu.amd.com"
    assert len(obj["spec"]["devices"]) == 8
    d0 = obj["spec"]["devices"][0]
    assert d0["name"] == "70"
    attrs = d0["basic"]["attributes"]
    assert attrs["amd.com/pciDeviceId"]["string"] == "75a3"
    assert attrs["amd.com/xgmiHive"]["string"].startswith("hive-")
    assert attrs["amd.com/numaNode"]["int"] == 0
    assert attrs["amd.com/isSriovVf"]["bool"] is False


def test_claim_store_survives_restart(tmp_path):
    """DRA requirement: a restarted driver must still unprepare (and
    idempotently re-prepare) claims the previous process prepared."""
    path = str(tmp_path / "claims.json")
    s1 = dra.ClaimStore(path)
    s1.put("uid-a", ["70", "71"])
    s1.put("uid-b", ["74"])
    s1.pop("uid-b")
    # "restart"
    s2 = dra.ClaimStore(path)
    assert s2.get("uid-a") == ["70", "71"]
    assert s2.get("uid-b") is None
    s2.pop("uid-a")
    s3 = dra.ClaimStore(path)
    assert s3.all() == {}


def test_claim_store_corrupt_file_starts_empty(tmp_path):
    path = tmp_path / "claims.json"
    path.write_text("{not json")
    s = dra.ClaimStore(str(path))
    assert s.all() == {}
    s.put("u", ["70"])
    assert dra.ClaimStore(str(path)).get("u") == ["70"]
