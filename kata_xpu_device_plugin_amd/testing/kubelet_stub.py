"""In-process kubelet stub for tests and the benchmark.

Plays the kubelet side of the DevicePlugin protocol ("run the
plugin's gRPC server on a temp socket, play kubelet by calling
Register/ListAndWatch/Allocate directly"): serves the Registration service
on ``kubelet.sock`` and exposes client helpers against registered plugin
endpoints.
"""
from __future__ import annotations

import os
import threading
from concurrent import futures
from dataclasses import dataclass
from typing import Dict, List, Optional

import grpc

from ..plugin import api


@dataclass
class Registration:
    version: str
    endpoint: str
    resource_name: str
    preferred_allocation: bool


class KubeletStub:
    """Registration server + plugin-side client helpers."""

    def __init__(self, socket_dir: str):
        self.socket_dir = socket_dir
        self.socket_path = os.path.join(socket_dir, api.KUBELET_SOCKET_NAME)
        self.registrations: List[Registration] = []
        self.registered = threading.Event()
        self._server: Optional[grpc.Server] = None
        self._channels: Dict[str, grpc.Channel] = {}

    # -- Registration service -----------------------------------------
    def Register(self, request, context):
        self.registrations.append(
            Registration(
                version=request.version,
                endpoint=request.endpoint,
                resource_name=request.resource_name,
                preferred_allocation=request.options.get_preferred_allocation_available,
            )
        )
        self.registered.set()
        return api.Empty()

    def start(self) -> None:
        os.makedirs(self.socket_dir, exist_ok=True)
        if os.path.exists(self.socket_path):
            os.unlink(self.socket_path)
        server = grpc.server(futures.ThreadPoolExecutor(max_workers=4))
        api.add_registration_servicer(server, self)
        server.add_insecure_port(f"unix://{self.socket_path}")
        server.start()
        self._server = server

    def stop(self) -> None:
        for ch in self._channels.values():
            ch.close()
        self._channels.clear()
        if self._server is not None:
            self._server.stop(grace=0.5).wait()
            self._server = None
        if os.path.exists(self.socket_path):
            try:
                os.unlink(self.socket_path)
            except OSError:
                pass

    def __enter__(self):
        self.start()
        return self

    def __exit__(self, *exc):
        self.stop()

    # -- plugin clients -------------------------------------------------
    def plugin_stub(self, endpoint: str, timeout: float = 5.0) -> api.DevicePluginStub:
        """Dial a registered plugin endpoint (socket name relative to the
        device-plugin dir, like kubelet does)."""
        if endpoint not in self._channels:
            path = os.path.join(self.socket_dir, endpoint)
            ch = grpc.insecure_channel(f"unix://{path}")
            grpc.channel_ready_future(ch).result(timeout=timeout)
            self._channels[endpoint] = ch
        return api.DevicePluginStub(self._channels[endpoint])

    def wait_for_registration(self, n: int = 1, timeout: float = 5.0) -> List[Registration]:
        import time
        deadline = time.monotonic() + timeout
        while len(self.registrations) < n and time.monotonic() < deadline:
            time.sleep(0.01)
        if len(self.registrations) < n:
            raise TimeoutError(
                f"only {len(self.registrations)}/{n} plugins registered"
            )
        return list(self.registrations)
