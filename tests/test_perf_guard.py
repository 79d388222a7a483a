"""Coarse performance regression guards (generous bounds; catch gross
regressions, not jitter)."""

import time

import grpc

from k8s_device_plugin_amd.plugin import AMDGPUPlugin
from k8s_device_plugin_amd.plugin.native_server import NativePluginServer
from k8s_device_plugin_amd.protos import deviceplugin as dp


def test_native_allocate_latency_guard(tmp_path, fake_mi355x_8):
    sock = str(tmp_path / "perf.sock")
    plugin = AMDGPUPlugin(resource="gpu", paths=fake_mi355x_8.paths)
    plugin.start()
    srv = NativePluginServer(plugin, sock)
    srv.start()
    try:
        ch = grpc.insecure_channel(f"unix://{sock}")
        stub = dp.DevicePluginStub(ch)
        req = dp.AllocateRequest()
        req.container_requests.add().devices_ids.append(
            sorted(plugin.devices)[0]
        )
        for _ in range(50):
            stub.Allocate(req, timeout=5)
        lat = []
        for _ in range(300):
            t0 = time.perf_counter()
            stub.Allocate(req, timeout=5)
            lat.append(time.perf_counter() - t0)
        lat.sort()
        p50 = lat[len(lat) // 2]
        # measured ~190us in this container; 5ms would indicate something
        # structurally wrong (per-request rediscovery, sysfs I/O, ...)
        assert p50 < 0.005, f"native Allocate p50 regressed to {p50*1e6:.0f}us"
        ch.close()
    finally:
        srv.stop()


def test_discovery_speed_guard(fake_mi355x_cpx):
    from k8s_device_plugin_amd.topology import KFDTopology, discover_gpus

    t0 = time.perf_counter()
    topo = KFDTopology.load(fake_mi355x_cpx.paths)
    devs = discover_gpus(fake_mi355x_cpx.paths, topology=topo)
    dt = time.perf_counter() - t0
    assert len(devs) == 64
    # one pass over a 64-partition tree (66 nodes, ~4k link files)
    assert dt < 5.0, f"CPX discovery took {dt:.1f}s"
