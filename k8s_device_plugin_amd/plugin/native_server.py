"""Native-server glue: feeds the C++ fast server pre-serialized state.

The C++ side (native/fastserver.cpp) owns the wire and the
preferred-allocation search; this module computes everything once in
Python — device list bytes, per-device Allocate fragments, allocator
tables — and pushes updates on each heartbeat.  Wire behavior is
conformance-tested against the Python grpc client in
tests/test_fastserver.py.
"""

from __future__ import annotations

import logging
from typing import Dict, Optional

from ..protos import deviceplugin as dp
from .server import AMDGPUPlugin

log = logging.getLogger(__name__)


def _container_response_bytes(specs, cdi_name=None) -> bytes:
    """Serialized ContainerAllocateResponse holding only `specs` (and the
    optional CDI device name) — tagged field fragments, concatenable per
    protobuf rules."""
    car = dp.ContainerAllocateResponse()
    for host_path in specs:
        s = car.devices.add()
        s.host_path = s.container_path = host_path
        s.permissions = "rw"
    if cdi_name:
        car.cdi_devices.add().name = cdi_name
    return car.SerializeToString()


class NativePluginServer:
    """Drop-in replacement for the grpc.Server serving one plugin."""

    def __init__(self, plugin: AMDGPUPlugin, socket_path: str):
        from ..native import load_fastserver

        mod = load_fastserver(required=True)
        self.plugin = plugin
        self.socket_path = socket_path
        self._srv = mod.Server(socket_path)

    def _cdi_name(self, device_id):
        if not getattr(self.plugin, "cdi_enabled", False):
            return None
        from .cdi import cdi_device_name

        return cdi_device_name(device_id)

    def start(self) -> None:
        p = self.plugin
        self._srv.set_options_response(
            p.GetDevicePluginOptions(dp.Empty(), None).SerializeToString()
        )
        self._srv.set_kfd_spec(_container_response_bytes(["/dev/kfd"]))
        specs: Dict[str, bytes] = {}
        for d in p.devices.values():
            specs[d.id] = _container_response_bytes(
                [f"/dev/dri/card{d.card}", f"/dev/dri/renderD{d.render_d}"],
                cdi_name=self._cdi_name(d.id),
            )
        self._srv.set_device_specs(specs)
        self._srv.set_list_response(
            dp.ListAndWatchResponse(devices=p._device_list()).SerializeToString()
        )
        if not p.allocator_init_error and p.allocator.initialized:
            groups, node_of_id, weights = p.allocator.export_state()
            self._srv.set_allocator_state(groups, node_of_id, weights)
        if p.prestart_probe:
            self._srv.set_prestart_paths(
                {d.id: p.render_device_path(d) for d in p.devices.values()}
            )
        self._srv.start()

    def heartbeat(self) -> None:
        """Recompute health and push the fresh list to every open stream.

        Also re-walks sysfs: if the device set changed (hot-unplug,
        partition-mode change), the Allocate fragments and allocator
        tables are rebuilt — the reference only refreshes its cache when
        the kubelet re-opens ListAndWatch (plugin.go:231)."""
        p = self.plugin
        from ..topology import KFDTopology, discover_gpus

        topo = KFDTopology.load(p.paths)
        fresh = discover_gpus(p.paths, topology=topo, strict=False)
        if set(fresh) != set(p.devices) or any(
            fresh[i].render_d != p.devices[i].render_d for i in fresh
        ):
            log.info("device set changed (%d -> %d); rebuilding serving state",
                     len(p.devices), len(fresh))
            p.start()  # re-discover + re-init allocator
            specs = {
                d.id: _container_response_bytes(
                    [f"/dev/dri/card{d.card}", f"/dev/dri/renderD{d.render_d}"],
                    cdi_name=self._cdi_name(d.id),
                )
                for d in p.devices.values()
            }
            self._srv.set_device_specs(specs)
            if not p.allocator_init_error and p.allocator.initialized:
                groups, node_of_id, weights = p.allocator.export_state()
                self._srv.set_allocator_state(groups, node_of_id, weights)
            if p.prestart_probe:
                self._srv.set_prestart_paths(
                    {d.id: p.render_device_path(d) for d in p.devices.values()}
                )
            if p.cdi_enabled:
                # keep the CDI spec in step with the device set
                try:
                    from .cdi import CDI_SPEC_DIR, write_cdi_spec

                    write_cdi_spec(
                        p.devices.values(),
                        spec_dir=p.cdi_spec_dir or CDI_SPEC_DIR,
                    )
                except OSError as e:
                    log.warning("CDI spec refresh failed: %s", e)

        p.maybe_deep_check()  # --deep-probe-every applies to this path too
        devs = p.refreshed_device_list(topology=topo)
        self._srv.push_list_update(
            dp.ListAndWatchResponse(devices=devs).SerializeToString()
        )

    def stop(self, grace: Optional[float] = None):
        self._srv.stop()

        class _Done:
            def wait(self, timeout=None):
                return True

        return _Done()
