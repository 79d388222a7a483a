"""PreStartContainer probe tests (both serving implementations).

The reference's PreStartContainer is an unadvertised no-op
(plugin.go:219-224); with --prestart-probe this build verifies each
requested device node answers before the container starts.
"""

import os

import grpc
import pytest

from k8s_device_plugin_amd.plugin import AMDGPUPlugin
from k8s_device_plugin_amd.plugin.native_server import NativePluginServer
from k8s_device_plugin_amd.protos import deviceplugin as dp


class _Ctx:
    """Minimal grpc context for direct servicer calls."""

    def __init__(self):
        self.aborted = None

    def abort(self, code, details):
        self.aborted = (code, details)
        raise RuntimeError(f"abort: {code} {details}")

    def is_active(self):
        return True


def _mk_plugin(fs, tmp_path, create_devices):
    dev_root = str(tmp_path / "dev")
    plugin = AMDGPUPlugin(resource="gpu", paths=fs.paths,
                          prestart_probe=True, dev_root=dev_root)
    plugin.start()
    if create_devices:
        os.makedirs(os.path.join(dev_root, "dri"), exist_ok=True)
        for d in plugin.devices.values():
            with open(plugin.render_device_path(d), "w") as f:
                f.write("")
    return plugin


def test_options_advertise_prestart(fake_mi355x_8, tmp_path):
    plugin = _mk_plugin(fake_mi355x_8, tmp_path, create_devices=True)
    opts = plugin.GetDevicePluginOptions(dp.Empty(), None)
    assert opts.pre_start_required


def test_python_prestart_pass_and_fail(fake_mi355x_8, tmp_path):
    plugin = _mk_plugin(fake_mi355x_8, tmp_path, create_devices=True)
    ids = sorted(plugin.devices)
    req = dp.PreStartContainerRequest()
    req.devices_ids.extend(ids[:2])
    assert plugin.PreStartContainer(req, _Ctx()) is not None

    # remove one device node -> probe must fail
    os.unlink(plugin.render_device_path(plugin.devices[ids[0]]))
    ctx = _Ctx()
    with pytest.raises(RuntimeError):
        plugin.PreStartContainer(req, ctx)
    assert ctx.aborted[0] == grpc.StatusCode.FAILED_PRECONDITION


def test_native_prestart_pass_and_fail(fake_mi355x_8, tmp_path):
    plugin = _mk_plugin(fake_mi355x_8, tmp_path, create_devices=True)
    sock = str(tmp_path / "ps.sock")
    srv = NativePluginServer(plugin, sock)
    srv.start()
    try:
        ch = grpc.insecure_channel(f"unix://{sock}")
        stub = dp.DevicePluginStub(ch)
        ids = sorted(plugin.devices)
        req = dp.PreStartContainerRequest()
        req.devices_ids.extend(ids[:2])
        assert stub.PreStartContainer(req, timeout=5) is not None

        os.unlink(plugin.render_device_path(plugin.devices[ids[1]]))
        with pytest.raises(grpc.RpcError) as ei:
            stub.PreStartContainer(req, timeout=5)
        assert ei.value.code() == grpc.StatusCode.FAILED_PRECONDITION
        assert ids[1] in ei.value.details()
        ch.close()
    finally:
        srv.stop()


def test_prestart_noop_when_disabled(fake_mi355x_8):
    plugin = AMDGPUPlugin(resource="gpu", paths=fake_mi355x_8.paths)
    plugin.start()
    opts = plugin.GetDevicePluginOptions(dp.Empty(), None)
    assert not opts.pre_start_required
    req = dp.PreStartContainerRequest()
    req.devices_ids.append("whatever")
    assert plugin.PreStartContainer(req, _Ctx()) is not None
