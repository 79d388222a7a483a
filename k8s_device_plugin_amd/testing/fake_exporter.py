"""Fake amd-metrics-exporter MetricsService for health-path tests.

The reference leaves the exporter client untested; this fake
serves the exact metricssvc schema on an injectable unix socket so tests can
flip a single GPU Unhealthy and verify in-stream propagation
(BASELINE.json config 5).
"""

from __future__ import annotations

import os
import threading
from concurrent.futures import ThreadPoolExecutor
from typing import Dict

from ..protos import metricssvc as ms


class FakeExporter:
    def __init__(self, socket_path: str):
        self.socket_path = socket_path
        self._health: Dict[str, str] = {}  # device_id -> "healthy"/"unhealthy"
        self._lock = threading.Lock()
        self._server = None

    def set_health(self, device_id: str, health: str) -> None:
        with self._lock:
            self._health[device_id] = health

    # MetricsService servicer
    def List(self, request, context):
        resp = ms.GPUStateResponse()
        with self._lock:
            for i, (dev, health) in enumerate(sorted(self._health.items())):
                resp.GPUState.add(ID=str(i), Health=health, Device=dev)
        return resp

    def GetGPUState(self, request, context):
        resp = ms.GPUStateResponse()
        with self._lock:
            for i, (dev, health) in enumerate(sorted(self._health.items())):
                if str(i) in request.ID or dev in request.ID:
                    resp.GPUState.add(ID=str(i), Health=health, Device=dev)
        return resp

    def start(self) -> "FakeExporter":
        import grpc

        os.makedirs(os.path.dirname(self.socket_path), exist_ok=True)
        if os.path.exists(self.socket_path):
            os.unlink(self.socket_path)
        self._server = grpc.server(ThreadPoolExecutor(max_workers=2))
        ms.add_metrics_servicer(self._server, self)
        self._server.add_insecure_port(f"unix://{self.socket_path}")
        self._server.start()
        return self

    def stop(self) -> None:
        if self._server is not None:
            self._server.stop(grace=0.5).wait()
            self._server = None
        if os.path.exists(self.socket_path):
            os.unlink(self.socket_path)
