"""End-to-end DevicePlugin gRPC tests against a stub kubelet.

Covers the serving paths the reference leaves untested:
registration, ListAndWatch streaming + health flips, Allocate device specs,
GetPreferredAllocation, kubelet-restart re-registration, heterogeneous
bucketing.  BASELINE.json configs 1 and 5.
"""

import queue
import threading
import time

import pytest

from k8s_device_plugin_amd.plugin import AMDGPUPlugin, PluginManager
from k8s_device_plugin_amd.protos import deviceplugin as dp
from k8s_device_plugin_amd.testing.fake_exporter import FakeExporter
from k8s_device_plugin_amd.testing.fakesysfs import FakeSysfs, build_mi355x_node
from k8s_device_plugin_amd.testing.stub_kubelet import StubKubelet


@pytest.fixture(params=["native", "python"])
def harness(request, tmp_path, fake_mi355x_8):
    """StubKubelet + PluginManager over the 8-GPU fake node, exercised
    against BOTH serving implementations."""
    dp_dir = str(tmp_path / "device-plugins")
    exporter_sock = str(tmp_path / "exporter" / "metrics.sock")
    kubelet = StubKubelet(dp_dir).start()

    def factory(resource):
        return AMDGPUPlugin(
            resource=resource,
            paths=fake_mi355x_8.paths,
            exporter_socket=exporter_sock,
            exporter_timeout=2.0,
        )

    mgr = PluginManager(
        factory, device_plugin_path=dp_dir, watch_interval=0.1,
        server_impl=request.param,
    )
    yield kubelet, mgr, exporter_sock
    mgr.stop()
    kubelet.stop()


def open_stream(stub):
    """Open ListAndWatch; responses arrive on a queue via a reader thread."""
    call = stub.ListAndWatch(dp.Empty())
    q = queue.Queue()

    def reader():
        try:
            for resp in call:
                q.put(resp)
        except Exception:
            pass

    t = threading.Thread(target=reader, daemon=True)
    t.start()
    return call, q


def test_registration_and_initial_list(harness):
    kubelet, mgr, _ = harness
    mgr.run(["gpu"])
    reg = kubelet.wait_for_registration()
    assert reg.version == "v1beta1"
    assert reg.resource_name == "amd.com/gpu"
    assert reg.endpoint == "amd.com_gpu"
    assert reg.options.get_preferred_allocation_available

    stub = kubelet.connect(reg.endpoint)
    opts = stub.GetDevicePluginOptions(dp.Empty(), timeout=5)
    assert opts.get_preferred_allocation_available

    call, q = open_stream(stub)
    first = q.get(timeout=5)
    assert len(first.devices) == 8
    for d in first.devices:
        assert d.health == "Healthy"
        assert len(d.topology.nodes) == 1
        assert d.topology.nodes[0].ID in (0, 1)
    call.cancel()


def test_allocate_device_specs(harness):
    kubelet, mgr, _ = harness
    mgr.run(["gpu"])
    reg = kubelet.wait_for_registration()
    stub = kubelet.connect(reg.endpoint)
    call, q = open_stream(stub)
    first = q.get(timeout=5)
    ids = sorted(d.ID for d in first.devices)[:2]

    req = dp.AllocateRequest()
    req.container_requests.add().devices_ids.extend(ids)
    resp = stub.Allocate(req, timeout=5)
    assert len(resp.container_responses) == 1
    paths = [d.host_path for d in resp.container_responses[0].devices]
    # /dev/kfd once + card/renderD per device (reference plugin.go:368-391)
    assert paths[0] == "/dev/kfd"
    assert "/dev/dri/card0" in paths and "/dev/dri/renderD128" in paths
    assert "/dev/dri/card1" in paths and "/dev/dri/renderD129" in paths
    assert len(paths) == 1 + 2 * len(ids)
    for d in resp.container_responses[0].devices:
        assert d.permissions == "rw"
        assert d.container_path == d.host_path
    call.cancel()


def test_preferred_allocation_numa_packing(harness):
    kubelet, mgr, _ = harness
    mgr.run(["gpu"])
    reg = kubelet.wait_for_registration()
    stub = kubelet.connect(reg.endpoint)
    call, q = open_stream(stub)
    ids = sorted(d.ID for d in q.get(timeout=5).devices)

    req = dp.PreferredAllocationRequest()
    cr = req.container_requests.add()
    cr.available_deviceIDs.extend(ids[2:])  # numa0: 2, numa1: 4
    cr.allocation_size = 4
    resp = stub.GetPreferredAllocation(req, timeout=5)
    chosen = set(resp.container_responses[0].deviceIDs)
    assert chosen == set(ids[4:]), "expected the 4 NUMA-1 GPUs"
    call.cancel()


def test_health_flip_via_exporter(harness):
    """Inject exporter Unhealthy on GPU3 and verify the in-stream flip
    (BASELINE.json config 5)."""
    kubelet, mgr, exporter_sock = harness
    mgr.run(["gpu"])
    reg = kubelet.wait_for_registration()
    stub = kubelet.connect(reg.endpoint)
    call, q = open_stream(stub)
    first = q.get(timeout=5)
    ids = sorted(d.ID for d in first.devices)
    gpu3 = ids[3]

    exporter = FakeExporter(exporter_sock).start()
    try:
        for i in ids:
            exporter.set_health(i, "healthy")
        exporter.set_health(gpu3, "unhealthy")

        t0 = time.monotonic()
        mgr.heartbeat_all()
        resp = q.get(timeout=10)
        elapsed = time.monotonic() - t0
        health = {d.ID: d.health for d in resp.devices}
        assert health[gpu3] == "Unhealthy"
        assert all(h == "Healthy" for i, h in health.items() if i != gpu3)
        assert elapsed < 5.0, "health flip exceeded SLA"

        # recovery
        exporter.set_health(gpu3, "healthy")
        mgr.heartbeat_all()
        resp = q.get(timeout=10)
        assert {d.health for d in resp.devices} == {"Healthy"}
    finally:
        exporter.stop()
    call.cancel()


def test_health_without_exporter_falls_back(harness):
    kubelet, mgr, _ = harness
    mgr.run(["gpu"])
    reg = kubelet.wait_for_registration()
    stub = kubelet.connect(reg.endpoint)
    call, q = open_stream(stub)
    q.get(timeout=5)
    mgr.heartbeat_all()
    resp = q.get(timeout=10)
    # no exporter socket: node-level simple check (fake tree has GPUs -> Healthy)
    assert {d.health for d in resp.devices} == {"Healthy"}
    call.cancel()


def test_kubelet_restart_reregisters(harness):
    kubelet, mgr, _ = harness
    mgr.run(["gpu"])
    kubelet.wait_for_registration()
    kubelet.restart()
    reg = kubelet.wait_for_registration(timeout=10)
    assert reg.resource_name == "amd.com/gpu"


def test_heterogeneous_bucketing(tmp_path):
    """Each resource's plugin advertises only its partition bucket
    (reference: plugin.go:270-299)."""
    fs = FakeSysfs(str(tmp_path / "het"))
    fs.add_cpu_node(0)
    fs.add_physical_gpu(0, node_id=2, compute_partition="SPX", memory_partition="NPS1")
    fs.add_physical_gpu(1, node_id=3, compute_partition="SPX", memory_partition="NPS1")
    fs.add_physical_gpu(2, node_id=4, compute_partition="CPX", memory_partition="NPS4")
    fs.add_link(2, 3)
    fs.add_link(2, 4)
    fs.add_link(3, 4)

    dp_dir = str(tmp_path / "device-plugins")
    kubelet = StubKubelet(dp_dir).start()
    mgr = PluginManager(
        lambda res: AMDGPUPlugin(resource=res, paths=fs.paths),
        device_plugin_path=dp_dir,
    )
    try:
        mgr.run(["spx_nps1", "cpx_nps4"])
        kubelet.wait_for_registrations(2)
        stub_spx = kubelet.connect("amd.com_spx_nps1")
        call1, q1 = open_stream(stub_spx)
        spx = q1.get(timeout=5)
        assert len(spx.devices) == 2
        stub_cpx = kubelet.connect("amd.com_cpx_nps4")
        call2, q2 = open_stream(stub_cpx)
        cpx = q2.get(timeout=5)
        assert len(cpx.devices) == 1
        call1.cancel()
        call2.cancel()
    finally:
        mgr.stop()
        kubelet.stop()


def test_two_streams_both_get_heartbeats(harness):
    """kubelet reconnects can leave two ListAndWatch streams briefly open;
    every stream must receive health refreshes."""
    kubelet, mgr, _ = harness
    mgr.run(["gpu"])
    reg = kubelet.wait_for_registration()
    stub = kubelet.connect(reg.endpoint)
    call1, q1 = open_stream(stub)
    call2, q2 = open_stream(stub)
    q1.get(timeout=5)
    q2.get(timeout=5)
    mgr.heartbeat_all()
    r1 = q1.get(timeout=10)
    r2 = q2.get(timeout=10)
    assert len(r1.devices) == len(r2.devices) == 8
    call1.cancel()
    call2.cancel()


def test_register_fails_gracefully_without_kubelet(tmp_path, fake_mi355x_8):
    """No kubelet.sock: registration retries then returns False; the plugin
    keeps serving and registers when the kubelet appears."""
    dp_dir = str(tmp_path / "dp")
    import os

    os.makedirs(dp_dir, exist_ok=True)
    mgr = PluginManager(
        lambda r: AMDGPUPlugin(resource=r, paths=fake_mi355x_8.paths),
        device_plugin_path=dp_dir, watch_interval=0.1,
    )
    try:
        mgr.run(["gpu"])  # registration fails (no kubelet yet) but serving is up
        kubelet = StubKubelet(dp_dir).start()
        try:
            reg = kubelet.wait_for_registration(timeout=10)
            assert reg.resource_name == "amd.com/gpu"
        finally:
            kubelet.stop()
    finally:
        mgr.stop()


def test_native_unavailable_falls_back(tmp_path, fake_mi355x_8, monkeypatch):
    """If the fast server can't load, the manager serves via python grpc."""
    import k8s_device_plugin_amd.plugin.native_server as ns

    def boom(*a, **k):
        raise RuntimeError("simulated native load failure")

    monkeypatch.setattr(ns, "NativePluginServer", boom)
    dp_dir = str(tmp_path / "dp")
    kubelet = StubKubelet(dp_dir).start()
    mgr = PluginManager(
        lambda r: AMDGPUPlugin(resource=r, paths=fake_mi355x_8.paths),
        device_plugin_path=dp_dir, server_impl="native",
    )
    try:
        mgr.run(["gpu"])
        assert not mgr.plugins["gpu"].native
        reg = kubelet.wait_for_registration()
        stub = kubelet.connect(reg.endpoint)
        assert stub.GetDevicePluginOptions(dp.Empty(), timeout=5) is not None
    finally:
        mgr.stop()
        kubelet.stop()
