"""Chaos test: kubelet restarts + exporter flaps + continuous allocations
for ~15 s; the plugin must keep serving and re-registering throughout."""

import random
import threading
import time

import grpc
import pytest

from k8s_device_plugin_amd.plugin import AMDGPUPlugin, PluginManager
from k8s_device_plugin_amd.protos import deviceplugin as dp
from k8s_device_plugin_amd.testing.fake_exporter import FakeExporter
from k8s_device_plugin_amd.testing.stub_kubelet import StubKubelet


@pytest.mark.timeout(120)
def test_chaos_restarts_and_flaps(tmp_path, fake_mi355x_8):
    dp_dir = str(tmp_path / "dp")
    exporter_sock = str(tmp_path / "metrics.sock")
    kubelet = StubKubelet(dp_dir).start()
    exporter = FakeExporter(exporter_sock).start()
    mgr = PluginManager(
        lambda r: AMDGPUPlugin(
            resource=r, paths=fake_mi355x_8.paths,
            exporter_socket=exporter_sock, exporter_timeout=1.0,
        ),
        device_plugin_path=dp_dir,
        watch_interval=0.1,
    )
    errors = []
    stop = threading.Event()

    try:
        mgr.run(["gpu"])
        reg = kubelet.wait_for_registration()
        ids = sorted(mgr.plugins["gpu"].plugin.devices)

        def allocator_worker():
            rng = random.Random(1)
            ch = grpc.insecure_channel(
                f"unix://{dp_dir}/{reg.endpoint}"
            )
            stub = dp.DevicePluginStub(ch)
            while not stop.is_set():
                req = dp.AllocateRequest()
                req.container_requests.add().devices_ids.extend(
                    rng.sample(ids, rng.randint(1, 4))
                )
                try:
                    stub.Allocate(req, timeout=5)
                except grpc.RpcError as e:  # pragma: no cover
                    errors.append(e)
                time.sleep(0.002)
            ch.close()

        def heartbeat_worker():
            rng = random.Random(2)
            while not stop.is_set():
                exporter.set_health(rng.choice(ids),
                                    rng.choice(["healthy", "unhealthy"]))
                mgr.heartbeat_all()
                time.sleep(0.05)

        threads = [
            threading.Thread(target=allocator_worker, daemon=True),
            threading.Thread(target=heartbeat_worker, daemon=True),
        ]
        for t in threads:
            t.start()

        # three kubelet restarts while traffic flows
        for _ in range(3):
            time.sleep(3.0)
            kubelet.restart()
            kubelet.wait_for_registration(timeout=15)

        time.sleep(2.0)
        stop.set()
        for t in threads:
            t.join(timeout=10)
        assert not errors, f"{len(errors)} allocation errors: {errors[:3]}"
    finally:
        stop.set()
        mgr.stop()
        kubelet.stop()
        exporter.stop()
