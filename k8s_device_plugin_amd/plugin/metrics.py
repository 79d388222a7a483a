"""Optional Prometheus metrics endpoint.

The reference ships no metrics at all (its labeller even disables the
controller-runtime metrics server — SURVEY.md §5).  This build exposes the
native server's RPC counters plus manager-level events when
`--metrics-port` is set; with the port unset nothing is started, matching
the reference's default posture.
"""

from __future__ import annotations

import logging
from typing import Optional

log = logging.getLogger(__name__)

_COUNTER_HELP = {
    "allocate_total": "Allocate RPCs served",
    "allocate_handler_ns_total": "cumulative Allocate handler time (ns)",
    "preferred_handler_ns_total": "cumulative GetPreferredAllocation handler time (ns)",
    "preferred_allocation_total": "GetPreferredAllocation RPCs served",
    "list_and_watch_streams_total": "ListAndWatch streams opened",
    "options_total": "GetDevicePluginOptions RPCs served",
    "prestart_total": "PreStartContainer RPCs served",
    "unknown_method_total": "RPCs to unknown methods",
    "list_pushes_total": "device-list updates pushed to streams",
    "connections_total": "kubelet connections accepted",
}


class _ManagerCollector:
    """Prometheus collector reading live stats from the plugin manager."""

    def __init__(self, manager):
        self.manager = manager

    def collect(self):
        from prometheus_client.core import CounterMetricFamily, GaugeMetricFamily

        g = GaugeMetricFamily(
            "amdgpu_dp_advertised_devices",
            "devices advertised per resource",
            labels=["resource"],
        )
        for resource, inst in self.manager.plugins.items():
            g.add_metric([resource], len(inst.plugin.devices))
        yield g

        df = GaugeMetricFamily(
            "amdgpu_dp_deep_probe_failed_gpus",
            "physical GPUs currently pinned Unhealthy by the deep probe",
            labels=["resource"],
        )
        for resource, inst in self.manager.plugins.items():
            df.add_metric([resource], len(inst.plugin._deep_failed))
        yield df

        counters = {}
        for resource, inst in self.manager.plugins.items():
            if inst.native:
                try:
                    stats = inst.server._srv.stats()
                except Exception:
                    continue
                for key, value in stats.items():
                    counters.setdefault(key, []).append((resource, value))
        for key, entries in counters.items():
            c = CounterMetricFamily(
                f"amdgpu_dp_{key}",
                _COUNTER_HELP.get(key, key),
                labels=["resource"],
            )
            for resource, value in entries:
                c.add_metric([resource], value)
            yield c


def start_metrics_server(manager, port: int) -> Optional[object]:
    """Start the Prometheus HTTP endpoint; returns the registry or None."""
    if port <= 0:
        return None
    try:
        import prometheus_client
        from prometheus_client.core import CollectorRegistry
    except ImportError:
        log.warning("prometheus_client not installed; metrics disabled")
        return None

    registry = CollectorRegistry()
    registry.register(_ManagerCollector(manager))
    prometheus_client.start_http_server(port, registry=registry)
    log.info("metrics endpoint on :%d/metrics", port)
    return registry
