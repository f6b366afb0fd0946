"""Wire-format tests pinning the v1beta1 schema.

Since the message classes are built from a hand-written descriptor (no
protoc available), these tests assert the exact wire bytes against
hand-encoded protobuf so a field-number regression can never silently
break kubelet compatibility.
"""
from kata_xpu_device_plugin_amd.plugin import api


def _tag(field_no, wire_type=2):
    return bytes([(field_no << 3) | wire_type])


def _ld(field_no, payload: bytes) -> bytes:
    assert len(payload) < 128
    return _tag(field_no) + bytes([len(payload)]) + payload


def test_register_request_wire():
    msg = api.RegisterRequest(
        version="v1beta1",
        endpoint="kxdp.sock",
        resource_name="amd.com/INSTINCT_MI355X",
        options=api.DevicePluginOptions(get_preferred_allocation_available=True),
    )
    expected = (
        _ld(1, b"v1beta1")
        + _ld(2, b"kxdp.sock")
        + _ld(3, b"amd.com/INSTINCT_MI355X")
        + _ld(4, _tag(2, 0) + b"\x01")  # options: bool field 2 = true
    )
    assert msg.SerializeToString() == expected


def test_device_wire():
    d = api.Device(id="70", health=api.HEALTHY,
                   topology=api.TopologyInfo(nodes=[api.NUMANode(id=1)]))
    expected = (
        _ld(1, b"70")
        + _ld(2, b"Healthy")
        + _ld(3, _ld(1, _tag(1, 0) + b"\x01"))
    )
    assert d.SerializeToString() == expected


def test_allocate_request_wire():
    req = api.AllocateRequest(
        container_requests=[api.ContainerAllocateRequest(devices_ids=["70", "71"])]
    )
    expected = _ld(1, _ld(1, b"70") + _ld(1, b"71"))
    assert req.SerializeToString() == expected


def test_container_allocate_response_wire():
    resp = api.ContainerAllocateResponse(
        envs={"K": "V"},
        cdi_devices=[api.CDIDevice(name="amd.com/gpu=70")],
    )
    env_entry = _ld(1, b"K") + _ld(2, b"V")
    expected = _ld(1, env_entry) + _ld(5, _ld(1, b"amd.com/gpu=70"))
    assert resp.SerializeToString() == expected
    # roundtrip
    back = api.ContainerAllocateResponse.FromString(expected)
    assert dict(back.envs) == {"K": "V"}
    assert back.cdi_devices[0].name == "amd.com/gpu=70"


def test_preferred_allocation_wire():
    req = api.PreferredAllocationRequest(
        container_requests=[
            api.ContainerPreferredAllocationRequest(
                available_device_ids=["70", "71"],
                must_include_device_ids=["70"],
                allocation_size=2,
            )
        ]
    )
    inner = _ld(1, b"70") + _ld(1, b"71") + _ld(2, b"70") + _tag(3, 0) + b"\x02"
    assert req.SerializeToString() == _ld(1, inner)


def test_device_spec_and_mount_fields():
    ds = api.DeviceSpec(container_path="/dev/vfio/70", host_path="/dev/vfio/70",
                        permissions="rw")
    assert ds.SerializeToString() == (
        _ld(1, b"/dev/vfio/70") + _ld(2, b"/dev/vfio/70") + _ld(3, b"rw")
    )
    m = api.Mount(container_path="/a", host_path="/b", read_only=True)
    assert m.SerializeToString() == _ld(1, b"/a") + _ld(2, b"/b") + _tag(3, 0) + b"\x01"


def test_constants():
    assert api.VERSION == "v1beta1"
    assert api.HEALTHY == "Healthy"
    assert api.UNHEALTHY == "Unhealthy"
