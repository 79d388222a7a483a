"""Deep-probe performance floors + heartbeat/prestart wiring (CPU side).

The GPU-side numerics of the probe itself are covered by tests/test_gpu.py;
here a stubbed probe module exercises the floor logic and the plugin
integration: a GPU that answers ioctls but misses its MFMA/HBM floor must
flip Unhealthy on the heartbeat path and abort deep PreStartContainer
(before floors, a degraded pipe still reported
healthy=true).
"""

import pytest

import k8s_device_plugin_amd.native as native
from k8s_device_plugin_amd.plugin import AMDGPUPlugin
from k8s_device_plugin_amd.protos import deviceplugin as dp
from k8s_device_plugin_amd.testing.fakesysfs import build_mi355x_node


class _StubProbeMod:
    def __init__(self, mfma=2000.0, hbm=6200.0, ok=True):
        self.mfma, self.hbm, self.ok = mfma, hbm, ok
        self.calls = []

    def run_probe(self, device=0, hbm_bytes=0):
        self.calls.append(device)
        return {
            "wave_ok": self.ok,
            "mfma_ok": self.ok,
            "lds_ok": self.ok,
            "hbm_copy_ok": self.ok,
            "mfma_tflops": self.mfma,
            "hbm_gbps": self.hbm,
            "healthy": self.ok,
        }


@pytest.fixture
def stub_probe(monkeypatch):
    def _install(mod):
        monkeypatch.setattr(native, "load_healthprobe",
                            lambda required=None: mod)
        return mod
    return _install


def test_floor_pass(stub_probe):
    mod = stub_probe(_StubProbeMod(mfma=2000, hbm=6200))
    res = native.deep_health_probe()
    assert res["healthy"] and res["floor_violations"] == []
    assert res["floors"]["mfma_tflops"] == native.DEFAULT_MFMA_FLOOR_TFLOPS


def test_floor_mfma_violation(stub_probe):
    stub_probe(_StubProbeMod(mfma=900, hbm=6200))
    res = native.deep_health_probe()
    assert not res["healthy"]
    assert any("mfma" in v for v in res["floor_violations"])


def test_floor_hbm_violation(stub_probe):
    stub_probe(_StubProbeMod(mfma=2000, hbm=3000))
    res = native.deep_health_probe()
    assert not res["healthy"]
    assert any("hbm" in v for v in res["floor_violations"])


def test_floor_env_override(stub_probe, monkeypatch):
    stub_probe(_StubProbeMod(mfma=900, hbm=3000))
    monkeypatch.setenv(native.MFMA_FLOOR_ENV, "0")
    monkeypatch.setenv(native.HBM_FLOOR_ENV, "0")
    res = native.deep_health_probe()
    assert res["healthy"] and res["floor_violations"] == []
    monkeypatch.setenv(native.MFMA_FLOOR_ENV, "2100")
    res = native.deep_health_probe()
    assert not res["healthy"]


def test_explicit_floor_args(stub_probe):
    stub_probe(_StubProbeMod(mfma=1000, hbm=5000))
    res = native.deep_health_probe(mfma_floor_tflops=900, hbm_floor_gbps=4000)
    assert res["healthy"]
    res = native.deep_health_probe(mfma_floor_tflops=1100, hbm_floor_gbps=4000)
    assert not res["healthy"]


class _Ctx:
    def is_active(self):
        return True


def _dev_root(tmp_path):
    d = tmp_path / "dev"
    d.mkdir(exist_ok=True)
    (d / "kfd").touch()
    return str(d)



def test_heartbeat_deep_check_flips_unhealthy(tmp_path, stub_probe):
    fs = build_mi355x_node(str(tmp_path / "n"), n_gpus=2)
    mod = stub_probe(_StubProbeMod(mfma=900))  # below floor

    plugin = AMDGPUPlugin(resource="gpu", paths=fs.paths, deep_probe_every=2,
                          dev_root=_dev_root(tmp_path))
    plugin.start()
    stream = plugin.ListAndWatch(dp.Empty(), _Ctx())
    first = next(stream)
    assert all(d.health == "Healthy" for d in first.devices)

    plugin.heartbeat()  # beat 1: deep check not due yet
    resp = next(stream)
    assert all(d.health == "Healthy" for d in resp.devices)
    assert mod.calls == []

    plugin.heartbeat()  # beat 2: deep check fires, floors fail
    resp = next(stream)
    assert all(d.health == "Unhealthy" for d in resp.devices)
    assert sorted(set(mod.calls)) == [0, 1]  # one probe per physical GPU

    # recovery: probe passes again -> healthy on the next due beat
    mod.mfma = 2000.0
    plugin.heartbeat()
    next(stream)
    plugin.heartbeat()
    resp = next(stream)
    assert all(d.health == "Healthy" for d in resp.devices)
    plugin.stop()


def test_deep_check_pins_all_partitions_of_failed_gpu(tmp_path, stub_probe):
    fs = build_mi355x_node(str(tmp_path / "n"), n_gpus=2, partitions_per_gpu=4)

    class _FirstGpuBad(_StubProbeMod):
        def run_probe(self, device=0, hbm_bytes=0):
            r = super().run_probe(device, hbm_bytes)
            r["mfma_tflops"] = 900.0 if device == 0 else 2000.0
            return r

    stub_probe(_FirstGpuBad())
    plugin = AMDGPUPlugin(resource="gpu", paths=fs.paths, deep_probe_every=1,
                          dev_root=_dev_root(tmp_path))
    plugin.start()
    stream = plugin.ListAndWatch(dp.Empty(), _Ctx())
    next(stream)
    plugin.heartbeat()
    resp = next(stream)
    health = {d.ID: d.health for d in resp.devices}
    assert sum(1 for h in health.values() if h == "Unhealthy") == 4
    assert sum(1 for h in health.values() if h == "Healthy") == 4
    # the unhealthy four are exactly one physical GPU's partitions
    bad = {i for i, h in health.items() if h == "Unhealthy"}
    assert len({plugin.devices[i].dev_id for i in bad}) == 1
    plugin.stop()


def test_prestart_deep_aborts_on_floor_failure(tmp_path, stub_probe):
    import grpc

    fs = build_mi355x_node(str(tmp_path / "n"), n_gpus=1)
    stub_probe(_StubProbeMod(mfma=900))

    plugin = AMDGPUPlugin(resource="gpu", paths=fs.paths,
                          prestart_probe=True, prestart_deep=True,
                          dev_root=fs.paths.root + "/dev")
    plugin.start()
    # fake /dev/dri files so the path-exists check passes
    import os

    dev = next(iter(plugin.devices.values()))
    dri = os.path.join(fs.paths.root, "dev", "dri")
    os.makedirs(dri, exist_ok=True)
    open(os.path.join(dri, f"renderD{dev.render_d}"), "w").close()

    class _AbortCtx:
        code = None

        def abort(self, code, msg):
            self.code = code
            raise RuntimeError(f"aborted: {msg}")

    ctx = _AbortCtx()
    req = dp.PreStartContainerRequest(devices_ids=[dev.id])
    with pytest.raises(RuntimeError, match="deep pre-start"):
        plugin.PreStartContainer(req, ctx)
    assert ctx.code == grpc.StatusCode.FAILED_PRECONDITION
    plugin.stop()


def test_deep_check_skipped_without_dev_kfd(tmp_path, stub_probe):
    """CPU/fake environments (no /dev/kfd) must not condemn synthetic
    devices just because the probe cannot run."""
    fs = build_mi355x_node(str(tmp_path / "n"), n_gpus=2)
    mod = stub_probe(_StubProbeMod(mfma=900))  # would fail floors IF run
    plugin = AMDGPUPlugin(resource="gpu", paths=fs.paths, deep_probe_every=1,
                          dev_root=str(tmp_path / "emptydev"))
    plugin.start()
    stream = plugin.ListAndWatch(dp.Empty(), _Ctx())
    next(stream)
    plugin.heartbeat()
    resp = next(stream)
    assert all(d.health == "Healthy" for d in resp.devices)
    assert mod.calls == []  # probe never invoked
    plugin.stop()


def test_native_heartbeat_triggers_deep_check(tmp_path, stub_probe):
    """--deep-probe-every must apply to the NATIVE serving path too (the
    default daemon path; gap found by the r02 soak rehearsal)."""
    import grpc

    from k8s_device_plugin_amd.plugin.native_server import NativePluginServer

    fs = build_mi355x_node(str(tmp_path / "n"), n_gpus=2)
    mod = stub_probe(_StubProbeMod(mfma=900))  # below floor
    plugin = AMDGPUPlugin(resource="gpu", paths=fs.paths, deep_probe_every=1,
                          dev_root=_dev_root(tmp_path))
    plugin.start()
    srv = NativePluginServer(plugin, str(tmp_path / "s.sock"))
    srv.start()
    try:
        ch = grpc.insecure_channel(f"unix://{tmp_path}/s.sock")
        stub = dp.DevicePluginStub(ch)
        call = stub.ListAndWatch(dp.Empty())
        it = iter(call)
        first = next(it)
        assert all(d.health == "Healthy" for d in first.devices)
        srv.heartbeat()
        resp = next(it)
        assert all(d.health == "Unhealthy" for d in resp.devices)
        assert sorted(set(mod.calls)) == [0, 1]
        call.cancel()
        ch.close()
    finally:
        srv.stop()
