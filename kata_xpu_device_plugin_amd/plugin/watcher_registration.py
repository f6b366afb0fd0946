"""Kubelet plugin-watcher registration (pluginregistration.v1).

Modern kubelets discover device plugins through the *plugin watcher*: the
plugin serves a `Registration` service (GetInfo / NotifyRegistrationStatus)
on a socket under ``/var/lib/kubelet/plugins_registry/`` and kubelet dials
it — no self-registration race on restarts. The reference supports only
legacy v1beta1 self-registration (`generic_device_plugin.go:200-219`);
this build serves BOTH (config: ``registration_mode``): legacy is the
default for parity, watcher mode is available for current kubelets.

Schema reconstructed from
``k8s.io/kubelet/pkg/apis/pluginregistration/v1/api.proto`` the same way
as plugin/api.py; field numbers pinned by tests.
"""
from __future__ import annotations

import grpc
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

from ..utils.log import get_logger

log = get_logger(__name__)

_PKG = "pluginregistration"
_F = descriptor_pb2.FieldDescriptorProto

DEVICE_PLUGIN_TYPE = "DevicePlugin"


def _build_fdp() -> descriptor_pb2.FileDescriptorProto:
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "kxdp/pluginregistration_v1.proto"
    fdp.package = _PKG
    fdp.syntax = "proto3"

    def msg(name):
        m = fdp.message_type.add()
        m.name = name
        return m

    def fld(m, name, number, ftype, label=_F.LABEL_OPTIONAL):
        f = m.field.add()
        f.name, f.number, f.type, f.label = name, number, ftype, label

    m = msg("PluginInfo")
    fld(m, "type", 1, _F.TYPE_STRING)
    fld(m, "name", 2, _F.TYPE_STRING)
    fld(m, "endpoint", 3, _F.TYPE_STRING)
    fld(m, "supported_versions", 4, _F.TYPE_STRING, _F.LABEL_REPEATED)

    m = msg("RegistrationStatus")
    fld(m, "plugin_registered", 1, _F.TYPE_BOOL)
    fld(m, "error", 2, _F.TYPE_STRING)

    m = msg("RegistrationStatusResponse")

    m = msg("InfoRequest")
    return fdp


_pool = descriptor_pool.DescriptorPool()
_pool.Add(_build_fdp())


def _cls(name):
    return message_factory.GetMessageClass(_pool.FindMessageTypeByName(f"{_PKG}.{name}"))


PluginInfo = _cls("PluginInfo")
RegistrationStatus = _cls("RegistrationStatus")
RegistrationStatusResponse = _cls("RegistrationStatusResponse")
InfoRequest = _cls("InfoRequest")

_SERVICE = "pluginregistration.Registration"


class WatcherRegistrationServicer:
    """Answers kubelet's plugin-watcher probes for one device plugin."""

    def __init__(self, resource_name: str, endpoint: str, versions=("v1beta1",)):
        self.resource_name = resource_name
        self.endpoint = endpoint  # ABSOLUTE path of the DevicePlugin socket
        self.versions = list(versions)
        self.last_status = None

    async def GetInfo(self, request, context):
        return PluginInfo(
            type=DEVICE_PLUGIN_TYPE,
            name=self.resource_name,
            endpoint=self.endpoint,
            supported_versions=self.versions,
        )

    async def NotifyRegistrationStatus(self, request, context):
        self.last_status = (request.plugin_registered, request.error)
        if request.plugin_registered:
            log.info("kubelet accepted plugin %s (watcher mode)", self.resource_name)
        else:
            log.error("kubelet REJECTED plugin %s: %s",
                      self.resource_name, request.error)
        return RegistrationStatusResponse()


def add_watcher_registration_servicer(server: grpc.Server, servicer) -> None:
    handlers = {
        "GetInfo": grpc.unary_unary_rpc_method_handler(
            servicer.GetInfo,
            request_deserializer=InfoRequest.FromString,
            response_serializer=PluginInfo.SerializeToString,
        ),
        "NotifyRegistrationStatus": grpc.unary_unary_rpc_method_handler(
            servicer.NotifyRegistrationStatus,
            request_deserializer=RegistrationStatus.FromString,
            response_serializer=RegistrationStatusResponse.SerializeToString,
        ),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(_SERVICE, handlers),)
    )


class WatcherRegistrationStub:
    """Client side (kubelet's role) — used by tests."""

    def __init__(self, channel: grpc.Channel):
        self.GetInfo = channel.unary_unary(
            f"/{_SERVICE}/GetInfo",
            request_serializer=InfoRequest.SerializeToString,
            response_deserializer=PluginInfo.FromString,
        )
        self.NotifyRegistrationStatus = channel.unary_unary(
            f"/{_SERVICE}/NotifyRegistrationStatus",
            request_serializer=RegistrationStatus.SerializeToString,
            response_deserializer=RegistrationStatusResponse.FromString,
        )
