"""CDI (Container Device Interface) tests: spec generation and CDI device
names in Allocate responses (both serving implementations)."""

import json

import grpc
import pytest

from k8s_device_plugin_amd.plugin import AMDGPUPlugin
from k8s_device_plugin_amd.plugin.cdi import (
    build_cdi_spec,
    cdi_device_name,
    write_cdi_spec,
)
from k8s_device_plugin_amd.plugin.native_server import NativePluginServer
from k8s_device_plugin_amd.protos import deviceplugin as dp
from k8s_device_plugin_amd.topology import discover_gpus


def test_cdi_spec_contents(tmp_path, fake_mi355x_8):
    devices = discover_gpus(fake_mi355x_8.paths)
    path = write_cdi_spec(devices.values(), spec_dir=str(tmp_path / "cdi"))
    spec = json.load(open(path))
    assert spec["kind"] == "amd.com/gpu"
    assert spec["cdiVersion"] == "0.7.0"
    # 0.7 annotations: per-device topology hints
    for dev in spec["devices"]:
        ann = dev["annotations"]
        assert "cdi.amd.com/numa-node" in ann
        assert "cdi.amd.com/physical-gpu" in ann
    assert len(spec["devices"]) == 8
    assert spec["containerEdits"]["deviceNodes"][0]["path"] == "/dev/kfd"
    d0 = next(d for d in spec["devices"] if d["name"] == "0000:0c:00.0")
    nodes = [n["path"] for n in d0["containerEdits"]["deviceNodes"]]
    assert nodes == ["/dev/dri/card0", "/dev/dri/renderD128"]


def test_cdi_name_format():
    assert cdi_device_name("0000:0c:00.0") == "amd.com/gpu=0000:0c:00.0"


def test_allocate_with_cdi_python(fake_mi355x_8):
    plugin = AMDGPUPlugin(resource="gpu", paths=fake_mi355x_8.paths,
                          cdi_enabled=True)
    plugin.start()
    ids = sorted(plugin.devices)[:2]
    req = dp.AllocateRequest()
    req.container_requests.add().devices_ids.extend(ids)
    resp = plugin.Allocate(req, None)
    car = resp.container_responses[0]
    assert len(car.devices) == 5  # kfd + 2 per device (still present)
    assert [c.name for c in car.cdi_devices] == [
        f"amd.com/gpu={i}" for i in ids
    ]


def test_allocate_with_cdi_native(tmp_path, fake_mi355x_8):
    sock = str(tmp_path / "cdi.sock")
    plugin = AMDGPUPlugin(resource="gpu", paths=fake_mi355x_8.paths,
                          cdi_enabled=True)
    plugin.start()
    srv = NativePluginServer(plugin, sock)
    srv.start()
    try:
        ch = grpc.insecure_channel(f"unix://{sock}")
        stub = dp.DevicePluginStub(ch)
        ids = sorted(plugin.devices)[:2]
        req = dp.AllocateRequest()
        req.container_requests.add().devices_ids.extend(ids)
        resp = stub.Allocate(req, timeout=5)
        car = resp.container_responses[0]
        assert len(car.devices) == 5
        assert sorted(c.name for c in car.cdi_devices) == sorted(
            f"amd.com/gpu={i}" for i in ids
        )
        # semantic parity with the python servicer (field ordering differs)
        py = plugin.Allocate(req, None).container_responses[0]
        assert sorted(d.host_path for d in car.devices) == sorted(
            d.host_path for d in py.devices
        )
        ch.close()
    finally:
        srv.stop()


def test_cdi_spec_refreshed_on_device_change(tmp_path):
    """Removing a GPU between heartbeats must rewrite the CDI spec."""
    import os
    import shutil

    from k8s_device_plugin_amd.testing.fakesysfs import build_mi355x_node

    fs = build_mi355x_node(str(tmp_path / "r"), n_gpus=4)
    cdi_dir = str(tmp_path / "cdi")
    plugin = AMDGPUPlugin(resource="gpu", paths=fs.paths, cdi_enabled=True,
                          cdi_spec_dir=cdi_dir)
    plugin.start()
    write_cdi_spec(plugin.devices.values(), spec_dir=cdi_dir)
    srv = NativePluginServer(plugin, str(tmp_path / "c.sock"))
    srv.start()
    try:
        gone = sorted(plugin.devices)[3]
        node_id = plugin.devices[gone].node_id
        shutil.rmtree(os.path.join(fs.paths.amdgpu_pci, gone))
        shutil.rmtree(os.path.join(fs.paths.kfd_topology_nodes, str(node_id)))
        srv.heartbeat()
        spec = json.load(open(os.path.join(cdi_dir, "amd.com-gpu.json")))
        assert len(spec["devices"]) == 3
        assert gone not in [d["name"] for d in spec["devices"]]
    finally:
        srv.stop()
