"""Prometheus metrics endpoint tests (native-server counters)."""

import urllib.request

from k8s_device_plugin_amd.plugin import AMDGPUPlugin, PluginManager
from k8s_device_plugin_amd.plugin.metrics import start_metrics_server
from k8s_device_plugin_amd.protos import deviceplugin as dp
from k8s_device_plugin_amd.testing.stub_kubelet import StubKubelet


def _free_port():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def test_metrics_counters(tmp_path, fake_mi355x_8):
    dp_dir = str(tmp_path / "dp")
    kubelet = StubKubelet(dp_dir).start()
    mgr = PluginManager(
        lambda r: AMDGPUPlugin(resource=r, paths=fake_mi355x_8.paths),
        device_plugin_path=dp_dir,
    )
    port = _free_port()
    try:
        mgr.run(["gpu"])
        assert start_metrics_server(mgr, port) is not None
        reg = kubelet.wait_for_registration()
        stub = kubelet.connect(reg.endpoint)
        req = dp.AllocateRequest()
        req.container_requests.add().devices_ids.append(
            sorted(mgr.plugins["gpu"].plugin.devices)[0]
        )
        for _ in range(5):
            stub.Allocate(req, timeout=5)

        body = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/metrics", timeout=5
        ).read().decode()
        assert 'amdgpu_dp_advertised_devices{resource="gpu"} 8.0' in body
        assert 'amdgpu_dp_allocate_total{resource="gpu"} 5.0' in body
        assert 'amdgpu_dp_connections_total{resource="gpu"}' in body
    finally:
        mgr.stop()
        kubelet.stop()


def test_metrics_disabled_by_default():
    assert start_metrics_server(None, 0) is None
