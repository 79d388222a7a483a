"""Per-GPU health from the amd-metrics-exporter gRPC service.

Parity with the reference health bridge (reference:
internal/pkg/exporter/health.go:36-106): short-lived connection to the
exporter's unix socket, 5 s query timeout, device keyed by GPUState.Device
(PCI bus ID), merge rule = exporter verdict per device when present, else
the node-wide default.
"""

from __future__ import annotations

import logging
import os
from typing import Dict, Optional, Sequence

from ..protos import deviceplugin as dp
from ..protos import metricssvc as ms

log = logging.getLogger(__name__)


def get_gpu_health(
    socket_path: str = ms.EXPORTER_SOCKET,
    timeout: float = ms.QUERY_TIMEOUT_S,
) -> Optional[Dict[str, str]]:
    """{device_id: Healthy|Unhealthy} from the exporter, or None when the
    exporter is unavailable (callers then fall back to the node default)."""
    if not os.path.exists(socket_path):
        return None
    import grpc

    try:
        with grpc.insecure_channel(f"unix://{socket_path}") as channel:
            stub = ms.MetricsServiceStub(channel)
            resp = stub.List(ms.Empty(), timeout=timeout)
    except grpc.RpcError as e:
        log.warning("metrics exporter query failed: %s", e)
        return None

    out: Dict[str, str] = {}
    for gpu in resp.GPUState:
        # the exporter reports lowercase health strings (reference:
        # health.go:74-80)
        if gpu.Health.lower() == dp.UNHEALTHY.lower():
            out[gpu.Device] = dp.UNHEALTHY
        else:
            out[gpu.Device] = dp.HEALTHY
    return out


def populate_per_gpu_health(
    devices: Sequence,  # Sequence[dp.Device]
    default_health: str,
    socket_path: str = ms.EXPORTER_SOCKET,
    timeout: float = ms.QUERY_TIMEOUT_S,
) -> None:
    """Set .health on each device: exporter verdict if known, else default
    (reference: health.go:86-106)."""
    health_map = get_gpu_health(socket_path, timeout)
    for dev in devices:
        if health_map is None:
            dev.health = default_health
        else:
            dev.health = health_map.get(dev.ID, default_health)
