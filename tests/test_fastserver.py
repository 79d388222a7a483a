"""Conformance tests: the native C++ gRPC server (nghttp2) against the
standard Python grpc client — the same client/shapes the stub kubelet uses,
so passing here means the kubelet-facing wire behavior matches the Python
grpc server's."""

import queue
import threading
import time

import grpc
import pytest

from k8s_device_plugin_amd.allocator import AllocationError
from k8s_device_plugin_amd.plugin import AMDGPUPlugin
from k8s_device_plugin_amd.plugin.native_server import NativePluginServer
from k8s_device_plugin_amd.protos import deviceplugin as dp


@pytest.fixture
def native(tmp_path, fake_mi355x_8):
    sock = str(tmp_path / "native.sock")
    plugin = AMDGPUPlugin(resource="gpu", paths=fake_mi355x_8.paths)
    plugin.start()
    srv = NativePluginServer(plugin, sock)
    srv.start()
    channel = grpc.insecure_channel(f"unix://{sock}")
    stub = dp.DevicePluginStub(channel)
    yield plugin, srv, stub
    channel.close()
    srv.stop()


@pytest.fixture
def pyplugin(fake_mi355x_8):
    plugin = AMDGPUPlugin(resource="gpu", paths=fake_mi355x_8.paths)
    plugin.start()
    return plugin


def test_options(native):
    _, _, stub = native
    opts = stub.GetDevicePluginOptions(dp.Empty(), timeout=5)
    assert opts.get_preferred_allocation_available


def test_prestart(native):
    _, _, stub = native
    assert stub.PreStartContainer(
        dp.PreStartContainerRequest(), timeout=5
    ) is not None


def test_list_and_watch_initial_and_push(native):
    plugin, srv, stub = native
    call = stub.ListAndWatch(dp.Empty())
    q = queue.Queue()

    def reader():
        try:
            for r in call:
                q.put(r)
        except Exception:
            pass

    threading.Thread(target=reader, daemon=True).start()
    first = q.get(timeout=5)
    assert len(first.devices) == 8
    assert {d.health for d in first.devices} == {"Healthy"}
    assert first.devices[0].topology.nodes[0].ID in (0, 1)

    srv.heartbeat()
    second = q.get(timeout=5)
    assert len(second.devices) == 8
    srv.heartbeat()
    third = q.get(timeout=5)
    assert len(third.devices) == 8
    call.cancel()


def test_allocate_matches_python_server(native, pyplugin):
    """Byte-level: native Allocate == Python servicer Allocate."""
    plugin, _, stub = native
    ids = sorted(plugin.devices)
    for req_ids in ([ids[0]], ids[:3], ids):
        req = dp.AllocateRequest()
        req.container_requests.add().devices_ids.extend(req_ids)
        native_resp = stub.Allocate(req, timeout=5)
        py_resp = pyplugin.Allocate(req, None)
        assert native_resp.SerializeToString() == py_resp.SerializeToString()


def test_allocate_unknown_id_skipped(native):
    _, _, stub = native
    req = dp.AllocateRequest()
    req.container_requests.add().devices_ids.append("nope")
    resp = stub.Allocate(req, timeout=5)
    paths = [d.host_path for d in resp.container_responses[0].devices]
    assert paths == ["/dev/kfd"]


def test_multi_container_allocate(native):
    plugin, _, stub = native
    ids = sorted(plugin.devices)
    req = dp.AllocateRequest()
    req.container_requests.add().devices_ids.append(ids[0])
    req.container_requests.add().devices_ids.extend(ids[1:3])
    resp = stub.Allocate(req, timeout=5)
    assert len(resp.container_responses) == 2
    assert len(resp.container_responses[0].devices) == 3
    assert len(resp.container_responses[1].devices) == 5


def test_preferred_matches_python_policy(native, pyplugin):
    plugin, _, stub = native
    ids = sorted(plugin.devices)
    cases = [
        (ids, [], 4),
        (ids[2:], [], 4),
        (ids, [ids[1]], 2),
        (ids, [], 1),
        (ids[:5], [ids[4]], 3),
    ]
    for available, required, size in cases:
        req = dp.PreferredAllocationRequest()
        cr = req.container_requests.add()
        cr.available_deviceIDs.extend(available)
        cr.must_include_deviceIDs.extend(required)
        cr.allocation_size = size
        native_out = list(
            stub.GetPreferredAllocation(req, timeout=5)
            .container_responses[0].deviceIDs
        )
        py_out = pyplugin.allocator.allocate(available, required, size)
        assert native_out == py_out, (available, required, size)


def test_preferred_error_maps_to_grpc_status(native):
    plugin, _, stub = native
    ids = sorted(plugin.devices)
    req = dp.PreferredAllocationRequest()
    cr = req.container_requests.add()
    cr.available_deviceIDs.extend(ids[:2])
    cr.allocation_size = 5  # more than available
    with pytest.raises(grpc.RpcError) as ei:
        stub.GetPreferredAllocation(req, timeout=5)
    assert ei.value.code() == grpc.StatusCode.INVALID_ARGUMENT


def test_unknown_method_unimplemented(native, tmp_path):
    plugin, srv, _ = native
    channel = grpc.insecure_channel(f"unix://{srv.socket_path}")
    bogus = channel.unary_unary(
        "/v1beta1.DevicePlugin/DoesNotExist",
        request_serializer=dp.Empty.SerializeToString,
        response_deserializer=dp.Empty.FromString,
    )
    with pytest.raises(grpc.RpcError) as ei:
        bogus(dp.Empty(), timeout=5)
    assert ei.value.code() == grpc.StatusCode.UNIMPLEMENTED
    channel.close()


def test_cpx_preferred_parity(tmp_path):
    """Native vs Python policy on the 64-partition CPX tree."""
    from k8s_device_plugin_amd.testing.fakesysfs import build_mi355x_node

    fs = build_mi355x_node(str(tmp_path / "cpx"), partitions_per_gpu=8,
                           compute_partition="CPX", memory_partition="NPS2")
    sock = str(tmp_path / "cpx.sock")
    plugin = AMDGPUPlugin(resource="gpu", paths=fs.paths)
    plugin.start()
    srv = NativePluginServer(plugin, sock)
    srv.start()
    try:
        channel = grpc.insecure_channel(f"unix://{sock}")
        stub = dp.DevicePluginStub(channel)
        ids = sorted(plugin.devices)
        for size in (1, 8, 10, 30):
            req = dp.PreferredAllocationRequest()
            cr = req.container_requests.add()
            cr.available_deviceIDs.extend(ids)
            cr.allocation_size = size
            native_out = list(
                stub.GetPreferredAllocation(req, timeout=30)
                .container_responses[0].deviceIDs
            )
            py_out = plugin.allocator.allocate(ids, [], size)
            assert native_out == py_out, size
        channel.close()
    finally:
        srv.stop()


def test_heartbeat_rebuilds_on_device_change(tmp_path):
    """Removing a GPU from sysfs between heartbeats must rebuild the
    Allocate fragments and drop the device from the stream."""
    import shutil
    import queue
    import threading

    from k8s_device_plugin_amd.testing.fakesysfs import build_mi355x_node

    fs = build_mi355x_node(str(tmp_path / "hp"), n_gpus=4)
    sock = str(tmp_path / "hp.sock")
    plugin = AMDGPUPlugin(resource="gpu", paths=fs.paths)
    plugin.start()
    srv = NativePluginServer(plugin, sock)
    srv.start()
    try:
        channel = grpc.insecure_channel(f"unix://{sock}")
        stub = dp.DevicePluginStub(channel)
        call = stub.ListAndWatch(dp.Empty())
        q = queue.Queue()

        def reader():
            try:
                for r in call:
                    q.put(r)
            except Exception:
                pass

        threading.Thread(target=reader, daemon=True).start()
        first = q.get(timeout=5)
        assert len(first.devices) == 4
        gone = sorted(plugin.devices)[3]

        # remove one GPU's pci dir + kfd node
        import os

        shutil.rmtree(os.path.join(fs.paths.amdgpu_pci, gone))
        node_id = plugin.devices[gone].node_id
        shutil.rmtree(os.path.join(fs.paths.kfd_topology_nodes, str(node_id)))

        srv.heartbeat()
        second = q.get(timeout=5)
        assert len(second.devices) == 3
        assert gone not in {d.ID for d in second.devices}

        # allocate for a surviving device still works
        keep = sorted(plugin.devices)[0]
        req = dp.AllocateRequest()
        req.container_requests.add().devices_ids.append(keep)
        resp = stub.Allocate(req, timeout=5)
        assert len(resp.container_responses[0].devices) == 3
        call.cancel()
        channel.close()
    finally:
        srv.stop()


def test_stream_and_unary_same_connection(native):
    """kubelet multiplexes ListAndWatch and Allocate on one connection;
    both must progress concurrently on a single channel."""
    import queue
    import threading

    plugin, srv, stub = native
    call = stub.ListAndWatch(dp.Empty())
    q = queue.Queue()

    def reader():
        try:
            for r in call:
                q.put(r)
        except Exception:
            pass

    threading.Thread(target=reader, daemon=True).start()
    q.get(timeout=5)

    ids = sorted(plugin.devices)
    req = dp.AllocateRequest()
    req.container_requests.add().devices_ids.append(ids[0])
    for _ in range(20):
        resp = stub.Allocate(req, timeout=5)
        assert len(resp.container_responses[0].devices) == 3
    srv.heartbeat()
    assert len(q.get(timeout=5).devices) == 8
    resp = stub.Allocate(req, timeout=5)
    assert len(resp.container_responses[0].devices) == 3
    call.cancel()


def test_unknown_fields_tolerated(native):
    """A newer kubelet may add fields to AllocateRequest; the native parser
    must skip unknown varint/fixed/length-delimited fields."""
    plugin, srv, _ = native
    import grpc

    ids = sorted(plugin.devices)
    base = dp.AllocateRequest()
    car = base.container_requests.add()
    car.devices_ids.append(ids[0])
    # append unknown fields at both levels:
    #   container level: field 9 varint, field 10 bytes, field 11 fixed32
    inner = car.SerializeToString() + b"\x48\x2a" + b"\x52\x03abc" + b"\x5d\x01\x02\x03\x04"
    #   top level: rebuild with modified container + unknown field 7 bytes
    raw = b"\x0a" + bytes([len(inner)]) + inner + b"\x3a\x04zzzz"

    ch = grpc.insecure_channel(f"unix://{srv.socket_path}")
    call = ch.unary_unary(
        "/v1beta1.DevicePlugin/Allocate",
        request_serializer=lambda b: b,
        response_deserializer=dp.AllocateResponse.FromString,
    )
    resp = call(raw, timeout=5)
    assert len(resp.container_responses) == 1
    assert len(resp.container_responses[0].devices) == 3
    ch.close()
