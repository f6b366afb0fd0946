"""Pod-resources client tests (reference C11 is dead code; ours is live)."""
import os
import tempfile
from concurrent import futures

import grpc
import pytest

from kata_xpu_device_plugin_amd.plugin import podresources as pr


def _tag(field_no, wire_type=2):
    return bytes([(field_no << 3) | wire_type])


def _ld(field_no, payload: bytes) -> bytes:
    return _tag(field_no) + bytes([len(payload)]) + payload


def test_wire_format():
    """Pin field numbers to k8s podresources v1 api.proto."""
    resp = pr.ListPodResourcesResponse(pod_resources=[
        pr.PodResources(
            name="p", namespace="ns",
            containers=[pr.ContainerResources(
                name="c",
                devices=[pr.ContainerDevices(
                    resource_name="amd.com/INSTINCT_MI355X",
                    device_ids=["70"],
                )],
            )],
        )
    ])
    dev = _ld(1, b"amd.com/INSTINCT_MI355X") + _ld(2, b"70")
    ctr = _ld(1, b"c") + _ld(2, dev)
    pod = _ld(1, b"p") + _ld(2, b"ns") + _ld(3, ctr)
    assert resp.SerializeToString() == _ld(1, pod)


class _Lister:
    def List(self, request, context):
        return pr.ListPodResourcesResponse(pod_resources=[
            pr.PodResources(
                name="train-0", namespace="ml",
                containers=[pr.ContainerResources(
                    name="worker",
                    devices=[
                        pr.ContainerDevices(
                            resource_name="amd.com/INSTINCT_MI355X",
                            device_ids=["70", "71"],
                        ),
                        pr.ContainerDevices(
                            resource_name="cpu-thing/other",
                            device_ids=["x"],
                        ),
                    ],
                )],
            )
        ])


@pytest.fixture
def lister_socket():
    d = tempfile.mkdtemp(prefix="kxdp-pr-")
    sock = os.path.join(d, "kubelet.sock")
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
    pr.add_lister_servicer(server, _Lister())
    server.add_insecure_port(f"unix://{sock}")
    server.start()
    yield sock
    server.stop(0)


def test_list_round_trip(lister_socket):
    client = pr.PodResourcesClient(lister_socket)
    assignments = client.list()
    assert len(assignments) == 2
    a = assignments[0]
    assert (a.namespace, a.pod, a.container) == ("ml", "train-0", "worker")
    assert a.device_ids == ["70", "71"]


def test_assignments_filtering(lister_socket):
    client = pr.PodResourcesClient(lister_socket)
    m = client.assignments("amd.com/")
    assert m == {"70": "ml/train-0/worker", "71": "ml/train-0/worker"}


class _ListerWithAllocatable(_Lister):
    def GetAllocatableResources(self, request, context):
        return pr.AllocatableResourcesResponse(devices=[
            pr.ContainerDevices(resource_name="amd.com/INSTINCT_MI355X",
                                device_ids=["70", "71", "72"]),
            pr.ContainerDevices(resource_name="other.io/dev",
                                device_ids=["z"]),
        ])


def test_allocatable_roundtrip():
    """GetAllocatableResources over a real unix socket: the capacity view
    complementing List (kubelet podresources v1)."""
    sock_dir = tempfile.mkdtemp(prefix="kxdp-podres-")
    path = os.path.join(sock_dir, "kubelet.sock")
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
    pr.add_lister_servicer(server, _ListerWithAllocatable())
    server.add_insecure_port(f"unix://{path}")
    server.start()
    try:
        client = pr.PodResourcesClient(socket_path=path, timeout_s=5)
        alloc = client.allocatable()
        assert alloc == {"amd.com/INSTINCT_MI355X": ["70", "71", "72"]}
    finally:
        server.stop(grace=None)


def test_allocatable_wire_format():
    """AllocatableResourcesResponse field numbers pinned (devices=1)."""
    resp = pr.AllocatableResourcesResponse(devices=[
        pr.ContainerDevices(resource_name="amd.com/X", device_ids=["1"])])
    dev = _ld(1, b"amd.com/X") + _ld(2, b"1")
    assert resp.SerializeToString() == _ld(1, dev)
