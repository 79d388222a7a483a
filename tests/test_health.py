"""Health module unit tests: exporter merge rules, timeout fallback."""

import threading
import time

from k8s_device_plugin_amd.health import get_gpu_health, populate_per_gpu_health
from k8s_device_plugin_amd.protos import deviceplugin as dp
from k8s_device_plugin_amd.protos import metricssvc as ms
from k8s_device_plugin_amd.testing.fake_exporter import FakeExporter


def _devs(ids):
    return [dp.Device(ID=i, health="Healthy") for i in ids]


def test_no_socket_returns_none(tmp_path):
    assert get_gpu_health(str(tmp_path / "missing.sock")) is None


def test_merge_rules(tmp_path):
    sock = str(tmp_path / "m.sock")
    ex = FakeExporter(sock).start()
    try:
        ex.set_health("a", "healthy")
        ex.set_health("b", "unhealthy")
        devs = _devs(["a", "b", "c"])
        populate_per_gpu_health(devs, "Unhealthy", sock)
        health = {d.ID: d.health for d in devs}
        # exporter verdict per device when present, node default otherwise
        assert health == {"a": "Healthy", "b": "Unhealthy", "c": "Unhealthy"}
    finally:
        ex.stop()


def test_uppercase_unhealthy_normalized(tmp_path):
    sock = str(tmp_path / "u.sock")
    ex = FakeExporter(sock).start()
    try:
        ex.set_health("a", "UNHEALTHY")
        devs = _devs(["a"])
        populate_per_gpu_health(devs, "Healthy", sock)
        assert devs[0].health == "Unhealthy"
    finally:
        ex.stop()


class _SlowExporter(FakeExporter):
    def List(self, request, context):
        time.sleep(5.0)
        return super().List(request, context)


def test_slow_exporter_falls_back_within_timeout(tmp_path):
    """A hung exporter must not stall the heartbeat past the query timeout
    (reference behavior: 5 s timeout then node default, health.go:37,66)."""
    sock = str(tmp_path / "slow.sock")
    ex = _SlowExporter(sock).start()
    try:
        ex.set_health("a", "unhealthy")
        devs = _devs(["a"])
        t0 = time.monotonic()
        populate_per_gpu_health(devs, "Healthy", sock, timeout=1.0)
        elapsed = time.monotonic() - t0
        assert elapsed < 3.0, f"timeout not honored: {elapsed:.1f}s"
        assert devs[0].health == "Healthy"  # fell back to node default
    finally:
        ex.stop()


def test_exporter_get_gpu_state_stub(tmp_path):
    """The GetGPUState RPC (defined for wire parity; the plugin itself uses
    List, like the reference) round-trips against the fake exporter."""
    import grpc

    from k8s_device_plugin_amd.protos import metricssvc as ms

    sock = str(tmp_path / "g.sock")
    ex = FakeExporter(sock).start()
    try:
        ex.set_health("0000:0c:00.0", "unhealthy")
        with grpc.insecure_channel(f"unix://{sock}") as ch:
            stub = ms.MetricsServiceStub(ch)
            resp = stub.GetGPUState(ms.GPUGetRequest(ID=["0"]), timeout=5)
            assert resp.GPUState[0].Device == "0000:0c:00.0"
            assert resp.GPUState[0].Health == "unhealthy"
    finally:
        ex.stop()


def test_heartbeat_ticker_survives_bad_callback():
    from k8s_device_plugin_amd.health import HeartbeatTicker
    import time

    hits = []
    t = HeartbeatTicker(0.05)
    t.subscribe(lambda: (_ for _ in ()).throw(RuntimeError("boom")))
    t.subscribe(lambda: hits.append(1))
    t.start()
    time.sleep(0.4)
    t.stop()
    assert len(hits) >= 3, "good callback starved by a failing one"


def test_heartbeat_ticker_zero_pulse_noop():
    from k8s_device_plugin_amd.health import HeartbeatTicker

    t = HeartbeatTicker(0)
    t.start()
    assert t._thread is None
    t.stop()
