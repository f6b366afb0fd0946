"""EXPERIMENTAL: kubelet DRA (Dynamic Resource Allocation) plugin.

Kubernetes ≥1.31 is migrating accelerators from device plugins to DRA
(resource.k8s.io): the scheduler allocates devices from ResourceSlices
against ResourceClaims, and kubelet asks the node-local DRA driver to
*prepare* a claim — returning CDI device IDs to inject. This module
implements the kubelet-facing half for this driver:

* ``DRAPlugin`` gRPC service (NodePrepareResources /
  NodeUnprepareResources) on a socket under the kubelet plugins dir,
  discovered through the same plugin-watcher registration as
  `watcher_registration.py` (type ``DRAPlugin``);
* claim preparation backed by the SAME discovery/topology/CDI layers as
  the v1beta1 plugin: device names are IOMMU group ids, prepared claims
  resolve to ``amd.com/gpu=<group>`` CDI ids;
* a ``ResourceSlice``-shaped node inventory dump
  (`resource_slice_obj()`) for the control-plane publisher (publishing
  to the API server needs a cluster and is out of scope here).

STATUS: experimental (needs a DRA-enabled cluster for end-to-end
verification). The protobuf schema mirrors
``k8s.io/kubelet/pkg/apis/dra/v1beta1/api.proto`` (k8s 1.32) and is
pinned against the transcription checked into
``tests/data/dra_v1beta1.proto`` by ``tests/test_proto_pin.py`` —
every message, field number/type/label and the DRAPlugin service path
are asserted there (same pinning as the DevicePlugin/pluginregistration/
podresources schemas). The preparation logic and tests are real.
"""
from __future__ import annotations

import json
import os
import threading
from typing import Dict, List, Optional

import grpc
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

from ..config import Config
from ..discovery.sysfs import NodeInventory
from ..utils.log import get_logger
from ..cdi.spec import qualified_name

log = get_logger(__name__)

_PKG = "k8s.io.kubelet.pkg.apis.dra.v1beta1"
_F = descriptor_pb2.FieldDescriptorProto


def _build_fdp() -> descriptor_pb2.FileDescriptorProto:
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "kxdp/dra_v1beta1.proto"
    fdp.package = _PKG
    fdp.syntax = "proto3"

    def msg(name):
        m = fdp.message_type.add()
        m.name = name
        return m

    def fld(m, name, number, ftype, label=_F.LABEL_OPTIONAL, type_name=None):
        f = m.field.add()
        f.name, f.number, f.type, f.label = name, number, ftype, label
        if type_name:
            f.type_name = f".{_PKG}.{type_name}"

    def map_field(m, name, number, value_type):
        entry = m.nested_type.add()
        entry.name = "".join(p.capitalize() for p in name.split("_")) + "Entry"
        entry.options.map_entry = True
        k = entry.field.add()
        k.name, k.number, k.type, k.label = "key", 1, _F.TYPE_STRING, _F.LABEL_OPTIONAL
        v = entry.field.add()
        v.name, v.number, v.type, v.label = ("value", 2, _F.TYPE_MESSAGE,
                                             _F.LABEL_OPTIONAL)
        v.type_name = f".{_PKG}.{value_type}"
        f = m.field.add()
        f.name, f.number, f.type, f.label = (name, number, _F.TYPE_MESSAGE,
                                             _F.LABEL_REPEATED)
        f.type_name = f".{_PKG}.{m.name}.{entry.name}"

    m = msg("Claim")
    fld(m, "namespace", 1, _F.TYPE_STRING)
    fld(m, "uid", 2, _F.TYPE_STRING)
    fld(m, "name", 3, _F.TYPE_STRING)

    m = msg("NodePrepareResourcesRequest")
    fld(m, "claims", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED, "Claim")

    m = msg("Device")
    fld(m, "request_names", 1, _F.TYPE_STRING, _F.LABEL_REPEATED)
    fld(m, "pool_name", 2, _F.TYPE_STRING)
    fld(m, "device_name", 3, _F.TYPE_STRING)
    fld(m, "cdi_device_ids", 4, _F.TYPE_STRING, _F.LABEL_REPEATED)

    m = msg("NodePrepareResourceResponse")
    fld(m, "devices", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED, "Device")
    fld(m, "error", 2, _F.TYPE_STRING)

    m = msg("NodePrepareResourcesResponse")
    map_field(m, "claims", 1, "NodePrepareResourceResponse")

    m = msg("NodeUnprepareResourcesRequest")
    fld(m, "claims", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED, "Claim")

    m = msg("NodeUnprepareResourceResponse")
    fld(m, "error", 1, _F.TYPE_STRING)

    m = msg("NodeUnprepareResourcesResponse")
    map_field(m, "claims", 1, "NodeUnprepareResourceResponse")
    return fdp


_pool = descriptor_pool.DescriptorPool()
_pool.Add(_build_fdp())


def _cls(name):
    return message_factory.GetMessageClass(_pool.FindMessageTypeByName(f"{_PKG}.{name}"))


Claim = _cls("Claim")
NodePrepareResourcesRequest = _cls("NodePrepareResourcesRequest")
NodePrepareResourcesResponse = _cls("NodePrepareResourcesResponse")
NodePrepareResourceResponse = _cls("NodePrepareResourceResponse")
DRADevice = _cls("Device")
NodeUnprepareResourcesRequest = _cls("NodeUnprepareResourcesRequest")
NodeUnprepareResourcesResponse = _cls("NodeUnprepareResourcesResponse")
NodeUnprepareResourceResponse = _cls("NodeUnprepareResourceResponse")

_SERVICE = f"{_PKG}.DRAPlugin"
DRIVER_NAME = "gpu.amd.com"


class ClaimStore:
    """Prepared-claim bookkeeping (uid → device names), optionally durable.

    DRA semantics require NodePrepareResources to be idempotent,
    NodeUnprepareResources to be safe for unknown claims, and prepared
    state to SURVIVE a driver restart (kubelet may unprepare a claim the
    old process prepared). With `path` set, every mutation is persisted
    atomically as JSON and reloaded on construction.
    """

    def __init__(self, path: Optional[str] = None):
        self._lock = threading.Lock()
        self._claims: Dict[str, List[str]] = {}
        self._path = path
        if path and os.path.exists(path):
            try:
                with open(path) as f:
                    raw = json.load(f)
                self._claims = {str(k): [str(d) for d in v]
                                for k, v in raw.items()}
            except (OSError, ValueError) as e:
                log.warning("unreadable claim store %s: %s (starting empty)",
                            path, e)

    def _persist_locked(self) -> None:
        if not self._path:
            return
        os.makedirs(os.path.dirname(self._path) or ".", exist_ok=True)
        tmp = self._path + ".tmp"
        with open(tmp, "w") as f:
            json.dump(self._claims, f)
        os.replace(tmp, self._path)

    def put(self, uid: str, devices: List[str]) -> None:
        with self._lock:
            self._claims[uid] = list(devices)
            self._persist_locked()

    def get(self, uid: str) -> Optional[List[str]]:
        with self._lock:
            v = self._claims.get(uid)
            return list(v) if v is not None else None

    def pop(self, uid: str) -> Optional[List[str]]:
        with self._lock:
            v = self._claims.pop(uid, None)
            if v is not None:
                self._persist_locked()
            return v

    def all(self) -> Dict[str, List[str]]:
        with self._lock:
            return {k: list(v) for k, v in self._claims.items()}


class DRAServicer:
    """NodePrepareResources / NodeUnprepareResources backed by the shared
    inventory + CDI layers.

    Claim→device resolution: this environment has no API server, so the
    scheduler's per-claim device allocation is injected by the caller via
    `resolver(claim_uid) -> [device names]` — in a cluster this would read
    the ResourceClaim's status.allocation. Tests and the stub exercise the
    full wire + preparation path.
    """

    def __init__(self, cfg: Config, inventory: NodeInventory, pool_name: str,
                 resolver=None, store_path: Optional[str] = None):
        self.cfg = cfg
        self.inventory = inventory
        self.pool_name = pool_name
        self.resolver = resolver
        self.store = ClaimStore(store_path)

    def _prepare_one(self, claim) -> object:
        uid = claim.uid
        prior = self.store.get(uid)
        if prior is not None:
            names = prior  # idempotent re-prepare
        elif self.resolver is not None:
            names = self.resolver(uid)
        else:
            names = None
        if not names:
            return NodePrepareResourceResponse(
                error=f"no allocation known for claim {claim.namespace}/"
                      f"{claim.name} ({uid})")
        missing = [n for n in names if n not in self.inventory.devices]
        if missing:
            return NodePrepareResourceResponse(
                error=f"allocated device(s) not on this node: {missing}")
        devices = [
            DRADevice(
                request_names=[],
                pool_name=self.pool_name,
                device_name=n,
                cdi_device_ids=[qualified_name(self.cfg.cdi_kind, n)],
            )
            for n in names
        ]
        self.store.put(uid, names)
        return NodePrepareResourceResponse(devices=devices)

    async def NodePrepareResources(self, request, context):
        resp = NodePrepareResourcesResponse()
        for claim in request.claims:
            resp.claims[claim.uid].CopyFrom(self._prepare_one(claim))
        return resp

    async def NodeUnprepareResources(self, request, context):
        resp = NodeUnprepareResourcesResponse()
        for claim in request.claims:
            self.store.pop(claim.uid)
            resp.claims[claim.uid].CopyFrom(NodeUnprepareResourceResponse())
        return resp


def add_dra_servicer(server, servicer) -> None:
    handlers = {
        "NodePrepareResources": grpc.unary_unary_rpc_method_handler(
            servicer.NodePrepareResources,
            request_deserializer=NodePrepareResourcesRequest.FromString,
            response_serializer=NodePrepareResourcesResponse.SerializeToString,
        ),
        "NodeUnprepareResources": grpc.unary_unary_rpc_method_handler(
            servicer.NodeUnprepareResources,
            request_deserializer=NodeUnprepareResourcesRequest.FromString,
            response_serializer=NodeUnprepareResourcesResponse.SerializeToString,
        ),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(_SERVICE, handlers),)
    )


class DRAStub:
    """kubelet-side client (tests)."""

    def __init__(self, channel):
        p = f"/{_SERVICE}/"
        self.NodePrepareResources = channel.unary_unary(
            p + "NodePrepareResources",
            request_serializer=NodePrepareResourcesRequest.SerializeToString,
            response_deserializer=NodePrepareResourcesResponse.FromString,
        )
        self.NodeUnprepareResources = channel.unary_unary(
            p + "NodeUnprepareResources",
            request_serializer=NodeUnprepareResourcesRequest.SerializeToString,
            response_deserializer=NodeUnprepareResourcesResponse.FromString,
        )


def resource_slice_obj(inventory: NodeInventory, topo, node_name: str,
                       pool_name: str) -> dict:
    """ResourceSlice-shaped inventory (resource.k8s.io/v1beta1) for a
    control-plane publisher: one named device per IOMMU group with
    hive/NUMA/model attributes the scheduler can select on."""
    devices = []
    for gid in inventory.device_ids():
        dev = inventory.devices[gid]
        bdf = dev.primary.bdf
        attrs = {
            "amd.com/pciDeviceId": {"string": f"{dev.model_device_id:04x}"},
            "amd.com/bdf": {"string": bdf},
            "amd.com/xgmiHive": {"string": topo.hive(bdf) if topo else ""},
            "amd.com/isSriovVf": {"bool": dev.is_vf},
        }
        if dev.numa_node >= 0:
            attrs["amd.com/numaNode"] = {"int": dev.numa_node}
        devices.append({"name": gid, "basic": {"attributes": attrs}})
    return {
        "apiVersion": "resource.k8s.io/v1beta1",
        "kind": "ResourceSlice",
        "metadata": {"name": f"{node_name}-{DRIVER_NAME}"},
        "spec": {
            "driver": DRIVER_NAME,
            "nodeName": node_name,
            "pool": {"name": pool_name, "generation": 1,
                     "resourceSliceCount": 1},
            "devices": devices,
        },
    }
