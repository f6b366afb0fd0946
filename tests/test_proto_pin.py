"""Wire-schema pinning.

The four kubelet-facing protobuf schemas in this repo are reconstructed
programmatically (no protoc in the image). This test parses the .proto
transcriptions checked into tests/data/ — field names, numbers, types,
labels and service surfaces of the upstream k8s.io/kubelet APIs — and
asserts every message/field/service each module declares matches them.

Subset semantics: a module may declare FEWER fields than upstream (proto3
skips unknown fields on the wire — e.g. podresources only consumes List),
but every field it DOES declare must agree with the upstream number, type,
label and (normalized) name, and no declared field number may collide with
a differently-named upstream field.
"""
import os
import re

import pytest

from google.protobuf import descriptor_pb2

_F = descriptor_pb2.FieldDescriptorProto

DATA = os.path.join(os.path.dirname(os.path.abspath(__file__)), "data")

_SCALAR = {
    "string": _F.TYPE_STRING,
    "bool": _F.TYPE_BOOL,
    "int32": _F.TYPE_INT32,
    "int64": _F.TYPE_INT64,
    "uint32": _F.TYPE_UINT32,
    "uint64": _F.TYPE_UINT64,
    "bytes": _F.TYPE_BYTES,
    "double": _F.TYPE_DOUBLE,
    "float": _F.TYPE_FLOAT,
}


def _norm(name: str) -> str:
    return name.lower().replace("_", "")


class ProtoText:
    """Messages/services parsed from a (simple, option-free) proto file."""

    def __init__(self, path):
        self.package = None
        self.messages = {}   # name -> {number: (name, kind, repeated)}
        #   kind: scalar token | ("msg", TypeName) | ("map", valtoken)
        self.services = {}   # name -> {method: (req, resp, resp_stream)}
        self._parse(open(path).read())

    def _parse(self, text):
        text = re.sub(r"//[^\n]*", "", text)
        pkg = re.search(r"\bpackage\s+([\w.]+)\s*;", text)
        self.package = pkg.group(1)
        for m in re.finditer(r"\bmessage\s+(\w+)\s*\{([^{}]*)\}", text):
            name, body = m.group(1), m.group(2)
            fields = {}
            for f in re.finditer(
                    r"(repeated\s+)?(map\s*<\s*(\w+)\s*,\s*(\w+)\s*>|[\w.]+)"
                    r"\s+(\w+)\s*=\s*(\d+)\s*;", body):
            # groups: 1 repeated, 2 type-or-map, 3 map key, 4 map val,
            #         5 field name, 6 number
                rep = bool(f.group(1))
                if f.group(3):
                    assert f.group(3) == "string", "only string keys used"
                    kind = ("map", f.group(4))
                    rep = True  # maps are repeated entries on the wire
                elif f.group(2) in _SCALAR:
                    kind = f.group(2)
                else:
                    kind = ("msg", f.group(2))
                fields[int(f.group(6))] = (f.group(5), kind, rep)
            self.messages[name] = fields
        for s in re.finditer(r"\bservice\s+(\w+)\s*\{(.*?)\n\}", text,
                             re.DOTALL):
            name, body = s.group(1), s.group(2)
            methods = {}
            for r in re.finditer(
                    r"\brpc\s+(\w+)\s*\(\s*(stream\s+)?(\w+)\s*\)\s*"
                    r"returns\s*\(\s*(stream\s+)?(\w+)\s*\)", body):
                methods[r.group(1)] = (r.group(3), r.group(5),
                                       bool(r.group(4)))
            self.services[name] = methods


def _assert_fdp_matches(fdp, proto: ProtoText, allow_missing_messages=()):
    """Every message/field in the module FileDescriptorProto must agree
    with the proto transcription."""
    for msg in fdp.message_type:
        assert msg.name in proto.messages, \
            f"message {msg.name} not in upstream proto"
        upstream = proto.messages[msg.name]
        map_entries = {n.name for n in msg.nested_type
                       if n.options.map_entry}
        for f in msg.field:
            assert f.number in upstream, \
                f"{msg.name}.{f.name}={f.number} has no upstream field"
            u_name, u_kind, u_rep = upstream[f.number]
            assert _norm(u_name) == _norm(f.name), \
                f"{msg.name} #{f.number}: name {f.name!r} vs upstream {u_name!r}"
            is_map = (f.type == _F.TYPE_MESSAGE and
                      f.type_name.rsplit(".", 1)[-1] in map_entries)
            if isinstance(u_kind, tuple) and u_kind[0] == "map":
                assert is_map, f"{msg.name}.{f.name}: upstream is a map"
                entry = next(n for n in msg.nested_type
                             if n.name == f.type_name.rsplit(".", 1)[-1])
                val = entry.field[1]
                if u_kind[1] in _SCALAR:
                    assert val.type == _SCALAR[u_kind[1]]
                else:
                    assert val.type == _F.TYPE_MESSAGE
                    assert val.type_name.rsplit(".", 1)[-1] == u_kind[1], \
                        f"{msg.name}.{f.name}: map value type mismatch"
            elif isinstance(u_kind, tuple):  # message field
                assert f.type == _F.TYPE_MESSAGE, \
                    f"{msg.name}.{f.name}: expected message type"
                assert f.type_name.rsplit(".", 1)[-1] == u_kind[1], \
                    f"{msg.name}.{f.name}: type {f.type_name} vs {u_kind[1]}"
                assert (f.label == _F.LABEL_REPEATED) == u_rep
            else:
                assert f.type == _SCALAR[u_kind], \
                    f"{msg.name}.{f.name}: scalar type mismatch ({u_kind})"
                assert (f.label == _F.LABEL_REPEATED) == u_rep, \
                    f"{msg.name}.{f.name}: label mismatch"


# --- DevicePlugin v1beta1 ---------------------------------------------------

def test_deviceplugin_schema_pinned():
    from kata_xpu_device_plugin_amd.plugin import api
    proto = ProtoText(os.path.join(DATA, "deviceplugin_v1beta1.proto"))
    assert proto.package == "v1beta1" == api._PKG
    _assert_fdp_matches(api._build_file_descriptor(), proto)
    # full message coverage: this module implements the WHOLE surface
    declared = {m.name for m in api._build_file_descriptor().message_type}
    assert declared == set(proto.messages)
    # service paths
    assert api._DEVICE_PLUGIN_SERVICE == "v1beta1.DevicePlugin"
    assert api._REGISTRATION_SERVICE == "v1beta1.Registration"
    dp = proto.services["DevicePlugin"]
    assert dp["ListAndWatch"] == ("Empty", "ListAndWatchResponse", True)
    assert dp["Allocate"] == ("AllocateRequest", "AllocateResponse", False)
    assert dp["GetPreferredAllocation"] == (
        "PreferredAllocationRequest", "PreferredAllocationResponse", False)
    assert dp["GetDevicePluginOptions"] == ("Empty", "DevicePluginOptions",
                                            False)
    assert dp["PreStartContainer"] == ("PreStartContainerRequest",
                                       "PreStartContainerResponse", False)
    assert proto.services["Registration"]["Register"] == (
        "RegisterRequest", "Empty", False)


# --- pluginregistration v1 --------------------------------------------------

def test_pluginregistration_schema_pinned():
    from kata_xpu_device_plugin_amd.plugin import watcher_registration as wr
    proto = ProtoText(os.path.join(DATA, "pluginregistration_v1.proto"))
    assert proto.package == "pluginregistration" == wr._PKG
    _assert_fdp_matches(wr._build_fdp(), proto)
    declared = {m.name for m in wr._build_fdp().message_type}
    assert declared == set(proto.messages)
    assert wr._SERVICE == "pluginregistration.Registration"
    svc = proto.services["Registration"]
    assert svc["GetInfo"] == ("InfoRequest", "PluginInfo", False)
    assert svc["NotifyRegistrationStatus"] == (
        "RegistrationStatus", "RegistrationStatusResponse", False)


# --- podresources v1 --------------------------------------------------------

def test_podresources_schema_pinned():
    from kata_xpu_device_plugin_amd.plugin import podresources as pr
    proto = ProtoText(os.path.join(DATA, "podresources_v1.proto"))
    assert proto.package == "v1"
    _assert_fdp_matches(pr._build_fdp(), proto)
    # the client MUST dial the upstream package's path, whatever the
    # internal pool package is named
    assert pr.KUBELET_SERVICE_PATH == "/v1.PodResourcesLister/List"
    assert proto.services["PodResourcesLister"]["List"] == (
        "ListPodResourcesRequest", "ListPodResourcesResponse", False)


# --- DRA v1beta1 ------------------------------------------------------------

def test_dra_schema_pinned():
    from kata_xpu_device_plugin_amd.plugin import dra
    proto = ProtoText(os.path.join(DATA, "dra_v1beta1.proto"))
    assert proto.package == "k8s.io.kubelet.pkg.apis.dra.v1beta1" == dra._PKG
    _assert_fdp_matches(dra._build_fdp(), proto)
    declared = {m.name for m in dra._build_fdp().message_type}
    assert declared == set(proto.messages)
    assert dra._SERVICE == "k8s.io.kubelet.pkg.apis.dra.v1beta1.DRAPlugin"
    svc = proto.services["DRAPlugin"]
    assert svc["NodePrepareResources"] == (
        "NodePrepareResourcesRequest", "NodePrepareResourcesResponse", False)
    assert svc["NodeUnprepareResources"] == (
        "NodeUnprepareResourcesRequest", "NodeUnprepareResourcesResponse",
        False)
