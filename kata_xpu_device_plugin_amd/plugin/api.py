"""Kubernetes DevicePlugin v1beta1 API, built without protoc.

This module reconstructs the kubelet DevicePlugin v1beta1 protobuf schema
(`k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1/api.proto` — the API the
reference consumes via its Go dependency, SURVEY.md §0 key deps) as a
programmatic FileDescriptorProto: the environment has the protobuf runtime
and grpcio but no protoc / grpc_tools, and wire compatibility only requires
matching **field numbers and types**, which are pinned here and covered by
wire-format tests (tests/test_api.py).

Exposes:
* message classes (Empty, Device, AllocateRequest, ...)
* service wiring: ``add_device_plugin_servicer`` / ``add_registration_servicer``
  for grpc servers, ``DevicePluginStub`` / ``RegistrationStub`` for clients
  (the stubs double as the in-process "kubelet" in tests and bench —
  SURVEY.md §4 integration tier).
"""
from __future__ import annotations

import grpc
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

VERSION = "v1beta1"
HEALTHY = "Healthy"
UNHEALTHY = "Unhealthy"
KUBELET_SOCKET_NAME = "kubelet.sock"

_PKG = "v1beta1"

_F = descriptor_pb2.FieldDescriptorProto

# (name, number, type, label, type_name) — type_name for message fields.
def _field(msg, name, number, ftype, label=_F.LABEL_OPTIONAL, type_name=None):
    f = msg.field.add()
    f.name = name
    f.number = number
    f.type = ftype
    f.label = label
    if type_name:
        f.type_name = f".{_PKG}.{type_name}"
    return f


def _map_field(fdp, msg, name, number):
    """Add a map<string,string> field: nested MapEntry + repeated field."""
    entry = msg.nested_type.add()
    entry.name = "".join(p.capitalize() for p in name.split("_")) + "Entry"
    entry.options.map_entry = True
    k = entry.field.add(); k.name = "key"; k.number = 1
    k.type = _F.TYPE_STRING; k.label = _F.LABEL_OPTIONAL
    v = entry.field.add(); v.name = "value"; v.number = 2
    v.type = _F.TYPE_STRING; v.label = _F.LABEL_OPTIONAL
    f = msg.field.add()
    f.name = name
    f.number = number
    f.type = _F.TYPE_MESSAGE
    f.label = _F.LABEL_REPEATED
    f.type_name = f".{_PKG}.{msg.name}.{entry.name}"


def _build_file_descriptor() -> descriptor_pb2.FileDescriptorProto:
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "kxdp/deviceplugin_v1beta1.proto"
    fdp.package = _PKG
    fdp.syntax = "proto3"

    def msg(name):
        m = fdp.message_type.add()
        m.name = name
        return m

    m = msg("Empty")  # noqa: F841

    m = msg("DevicePluginOptions")
    _field(m, "pre_start_required", 1, _F.TYPE_BOOL)
    _field(m, "get_preferred_allocation_available", 2, _F.TYPE_BOOL)

    m = msg("RegisterRequest")
    _field(m, "version", 1, _F.TYPE_STRING)
    _field(m, "endpoint", 2, _F.TYPE_STRING)
    _field(m, "resource_name", 3, _F.TYPE_STRING)
    _field(m, "options", 4, _F.TYPE_MESSAGE, type_name="DevicePluginOptions")

    m = msg("ListAndWatchResponse")
    _field(m, "devices", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED, "Device")

    m = msg("NUMANode")
    _field(m, "id", 1, _F.TYPE_INT64)

    m = msg("TopologyInfo")
    _field(m, "nodes", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED, "NUMANode")

    m = msg("Device")
    _field(m, "id", 1, _F.TYPE_STRING)
    _field(m, "health", 2, _F.TYPE_STRING)
    _field(m, "topology", 3, _F.TYPE_MESSAGE, type_name="TopologyInfo")

    m = msg("ContainerPreferredAllocationRequest")
    _field(m, "available_device_ids", 1, _F.TYPE_STRING, _F.LABEL_REPEATED)
    _field(m, "must_include_device_ids", 2, _F.TYPE_STRING, _F.LABEL_REPEATED)
    _field(m, "allocation_size", 3, _F.TYPE_INT32)

    m = msg("PreferredAllocationRequest")
    _field(m, "container_requests", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
           "ContainerPreferredAllocationRequest")

    m = msg("ContainerPreferredAllocationResponse")
    _field(m, "device_ids", 1, _F.TYPE_STRING, _F.LABEL_REPEATED)

    m = msg("PreferredAllocationResponse")
    _field(m, "container_responses", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
           "ContainerPreferredAllocationResponse")

    m = msg("ContainerAllocateRequest")
    _field(m, "devices_ids", 1, _F.TYPE_STRING, _F.LABEL_REPEATED)

    m = msg("AllocateRequest")
    _field(m, "container_requests", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
           "ContainerAllocateRequest")

    m = msg("Mount")
    _field(m, "container_path", 1, _F.TYPE_STRING)
    _field(m, "host_path", 2, _F.TYPE_STRING)
    _field(m, "read_only", 3, _F.TYPE_BOOL)

    m = msg("DeviceSpec")
    _field(m, "container_path", 1, _F.TYPE_STRING)
    _field(m, "host_path", 2, _F.TYPE_STRING)
    _field(m, "permissions", 3, _F.TYPE_STRING)

    m = msg("CDIDevice")
    _field(m, "name", 1, _F.TYPE_STRING)

    m = msg("ContainerAllocateResponse")
    _map_field(fdp, m, "envs", 1)
    _field(m, "mounts", 2, _F.TYPE_MESSAGE, _F.LABEL_REPEATED, "Mount")
    _field(m, "devices", 3, _F.TYPE_MESSAGE, _F.LABEL_REPEATED, "DeviceSpec")
    _map_field(fdp, m, "annotations", 4)
    _field(m, "cdi_devices", 5, _F.TYPE_MESSAGE, _F.LABEL_REPEATED, "CDIDevice")

    m = msg("AllocateResponse")
    _field(m, "container_responses", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
           "ContainerAllocateResponse")

    m = msg("PreStartContainerRequest")
    _field(m, "devices_ids", 1, _F.TYPE_STRING, _F.LABEL_REPEATED)

    m = msg("PreStartContainerResponse")  # noqa: F841
    return fdp


_pool = descriptor_pool.DescriptorPool()
_pool.Add(_build_file_descriptor())


def _cls(name: str):
    return message_factory.GetMessageClass(_pool.FindMessageTypeByName(f"{_PKG}.{name}"))


Empty = _cls("Empty")
DevicePluginOptions = _cls("DevicePluginOptions")
RegisterRequest = _cls("RegisterRequest")
ListAndWatchResponse = _cls("ListAndWatchResponse")
NUMANode = _cls("NUMANode")
TopologyInfo = _cls("TopologyInfo")
Device = _cls("Device")
ContainerPreferredAllocationRequest = _cls("ContainerPreferredAllocationRequest")
PreferredAllocationRequest = _cls("PreferredAllocationRequest")
ContainerPreferredAllocationResponse = _cls("ContainerPreferredAllocationResponse")
PreferredAllocationResponse = _cls("PreferredAllocationResponse")
ContainerAllocateRequest = _cls("ContainerAllocateRequest")
AllocateRequest = _cls("AllocateRequest")
Mount = _cls("Mount")
DeviceSpec = _cls("DeviceSpec")
CDIDevice = _cls("CDIDevice")
ContainerAllocateResponse = _cls("ContainerAllocateResponse")
AllocateResponse = _cls("AllocateResponse")
PreStartContainerRequest = _cls("PreStartContainerRequest")
PreStartContainerResponse = _cls("PreStartContainerResponse")


# ---------------------------------------------------------------------------
# gRPC service wiring (the 5 DevicePlugin RPCs + Registration — reference
# surface: generic_device_plugin.go:222-386 and :200-219).
# ---------------------------------------------------------------------------

_DEVICE_PLUGIN_SERVICE = f"{_PKG}.DevicePlugin"
_REGISTRATION_SERVICE = f"{_PKG}.Registration"


def add_device_plugin_servicer(server: grpc.Server, servicer) -> None:
    handlers = {
        "GetDevicePluginOptions": grpc.unary_unary_rpc_method_handler(
            servicer.GetDevicePluginOptions,
            request_deserializer=Empty.FromString,
            response_serializer=DevicePluginOptions.SerializeToString,
        ),
        "ListAndWatch": grpc.unary_stream_rpc_method_handler(
            servicer.ListAndWatch,
            request_deserializer=Empty.FromString,
            response_serializer=ListAndWatchResponse.SerializeToString,
        ),
        "GetPreferredAllocation": grpc.unary_unary_rpc_method_handler(
            servicer.GetPreferredAllocation,
            request_deserializer=PreferredAllocationRequest.FromString,
            response_serializer=PreferredAllocationResponse.SerializeToString,
        ),
        "Allocate": grpc.unary_unary_rpc_method_handler(
            servicer.Allocate,
            request_deserializer=AllocateRequest.FromString,
            response_serializer=AllocateResponse.SerializeToString,
        ),
        "PreStartContainer": grpc.unary_unary_rpc_method_handler(
            servicer.PreStartContainer,
            request_deserializer=PreStartContainerRequest.FromString,
            response_serializer=PreStartContainerResponse.SerializeToString,
        ),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(_DEVICE_PLUGIN_SERVICE, handlers),)
    )


def add_registration_servicer(server: grpc.Server, servicer) -> None:
    handlers = {
        "Register": grpc.unary_unary_rpc_method_handler(
            servicer.Register,
            request_deserializer=RegisterRequest.FromString,
            response_serializer=Empty.SerializeToString,
        ),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(_REGISTRATION_SERVICE, handlers),)
    )


class DevicePluginStub:
    """Client stub — what kubelet uses against the plugin socket."""

    def __init__(self, channel: grpc.Channel):
        p = f"/{_DEVICE_PLUGIN_SERVICE}/"
        self.GetDevicePluginOptions = channel.unary_unary(
            p + "GetDevicePluginOptions",
            request_serializer=Empty.SerializeToString,
            response_deserializer=DevicePluginOptions.FromString,
        )
        self.ListAndWatch = channel.unary_stream(
            p + "ListAndWatch",
            request_serializer=Empty.SerializeToString,
            response_deserializer=ListAndWatchResponse.FromString,
        )
        self.GetPreferredAllocation = channel.unary_unary(
            p + "GetPreferredAllocation",
            request_serializer=PreferredAllocationRequest.SerializeToString,
            response_deserializer=PreferredAllocationResponse.FromString,
        )
        self.Allocate = channel.unary_unary(
            p + "Allocate",
            request_serializer=AllocateRequest.SerializeToString,
            response_deserializer=AllocateResponse.FromString,
        )
        self.PreStartContainer = channel.unary_unary(
            p + "PreStartContainer",
            request_serializer=PreStartContainerRequest.SerializeToString,
            response_deserializer=PreStartContainerResponse.FromString,
        )


class RegistrationStub:
    """Client stub — what the plugin uses against kubelet.sock
    (reference: Register, generic_device_plugin.go:200-219)."""

    def __init__(self, channel: grpc.Channel):
        self.Register = channel.unary_unary(
            f"/{_REGISTRATION_SERVICE}/Register",
            request_serializer=RegisterRequest.SerializeToString,
            response_deserializer=Empty.FromString,
        )
