"""Stub kubelet: Registration server + DevicePlugin client.

The reference has no kubelet-side test harness (SURVEY.md §4 lists the gRPC
serving paths as untested); this stub closes that gap and doubles as the
bench.py driver (BASELINE.json configs 1-5).
"""

from __future__ import annotations

import os
import threading
import time
from concurrent.futures import ThreadPoolExecutor
from typing import List

from ..protos import deviceplugin as dp


class StubKubelet:
    """Serves v1beta1.Registration on <dir>/kubelet.sock and can dial back
    any registered plugin endpoint like the real kubelet does."""

    def __init__(self, device_plugin_path: str):
        self.device_plugin_path = device_plugin_path
        self.socket_path = os.path.join(device_plugin_path, "kubelet.sock")
        self.registrations: List = []  # RegisterRequest
        self._reg_event = threading.Event()
        self._server = None
        self._channels = []

    # Registration servicer
    def Register(self, request, context):
        self.registrations.append(request)
        self._reg_event.set()
        return dp.Empty()

    def start(self) -> "StubKubelet":
        import grpc

        os.makedirs(self.device_plugin_path, exist_ok=True)
        if os.path.exists(self.socket_path):
            os.unlink(self.socket_path)
        self._server = grpc.server(ThreadPoolExecutor(max_workers=4))
        dp.add_registration_servicer(self._server, self)
        self._server.add_insecure_port(f"unix://{self.socket_path}")
        self._server.start()
        return self

    def stop(self) -> None:
        for ch in self._channels:
            ch.close()
        self._channels.clear()
        if self._server is not None:
            self._server.stop(grace=0.5).wait()
            self._server = None
        if os.path.exists(self.socket_path):
            os.unlink(self.socket_path)

    def restart(self) -> None:
        """Simulate a kubelet restart (socket re-created with a new inode)."""
        self.stop()
        self._reg_event.clear()
        self.registrations.clear()
        self.start()

    def wait_for_registration(self, timeout: float = 10.0):
        if not self._reg_event.wait(timeout):
            raise TimeoutError("no plugin registered in time")
        return self.registrations[-1]

    def wait_for_registrations(self, count: int, timeout: float = 10.0) -> List:
        deadline = time.monotonic() + timeout
        while len(self.registrations) < count:
            if time.monotonic() > deadline:
                raise TimeoutError(
                    f"expected {count} registrations, got {len(self.registrations)}"
                )
            time.sleep(0.01)
        return list(self.registrations)

    def connect(self, endpoint: str) -> dp.DevicePluginStub:
        """Dial a plugin endpoint (basename under the device-plugin dir)."""
        import grpc

        path = os.path.join(self.device_plugin_path, endpoint)
        channel = grpc.insecure_channel(f"unix://{path}")
        self._channels.append(channel)
        return dp.DevicePluginStub(channel)
