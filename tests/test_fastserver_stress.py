"""Native server stress/robustness: concurrent clients, stream churn,
mid-stream disconnects, malformed frames."""

import queue
import socket
import threading
import time

import grpc
import pytest

from k8s_device_plugin_amd.plugin import AMDGPUPlugin
from k8s_device_plugin_amd.plugin.native_server import NativePluginServer
from k8s_device_plugin_amd.protos import deviceplugin as dp


@pytest.fixture
def native(tmp_path, fake_mi355x_8):
    sock = str(tmp_path / "n.sock")
    plugin = AMDGPUPlugin(resource="gpu", paths=fake_mi355x_8.paths)
    plugin.start()
    srv = NativePluginServer(plugin, sock)
    srv.start()
    yield plugin, srv, sock
    srv.stop()


def test_concurrent_allocate_many_channels(native):
    plugin, _, sock = native
    ids = sorted(plugin.devices)
    errors = []

    def worker(n):
        try:
            ch = grpc.insecure_channel(f"unix://{sock}")
            stub = dp.DevicePluginStub(ch)
            req = dp.AllocateRequest()
            req.container_requests.add().devices_ids.extend(ids[: 1 + n % 4])
            for _ in range(100):
                resp = stub.Allocate(req, timeout=10)
                assert len(resp.container_responses[0].devices) == 1 + 2 * (1 + n % 4)
            ch.close()
        except Exception as e:  # pragma: no cover
            errors.append(e)

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert not errors, errors


def test_stream_churn(native):
    """Open/cancel ListAndWatch repeatedly while heartbeats push."""
    plugin, srv, sock = native
    ch = grpc.insecure_channel(f"unix://{sock}")
    stub = dp.DevicePluginStub(ch)
    stop = threading.Event()

    def beater():
        while not stop.is_set():
            srv.heartbeat()
            time.sleep(0.005)

    t = threading.Thread(target=beater, daemon=True)
    t.start()
    try:
        for _ in range(30):
            call = stub.ListAndWatch(dp.Empty())
            it = iter(call)
            first = next(it)
            assert len(first.devices) == 8
            call.cancel()
        # a long-lived stream still works after the churn
        call = stub.ListAndWatch(dp.Empty())
        it = iter(call)
        next(it)
        srv.heartbeat()
        got = next(it)
        assert len(got.devices) == 8
        call.cancel()
    finally:
        stop.set()
        t.join(timeout=5)
    ch.close()


def test_abrupt_disconnect_mid_stream(native):
    """Kill the TCP-level connection while a stream is open; the server
    must keep serving other clients."""
    plugin, srv, sock = native
    ch1 = grpc.insecure_channel(f"unix://{sock}")
    stub1 = dp.DevicePluginStub(ch1)
    call = stub1.ListAndWatch(dp.Empty())
    next(iter(call))
    # abrupt close without goaway
    ch1.close()
    time.sleep(0.1)

    ch2 = grpc.insecure_channel(f"unix://{sock}")
    stub2 = dp.DevicePluginStub(ch2)
    opts = stub2.GetDevicePluginOptions(dp.Empty(), timeout=5)
    assert opts.get_preferred_allocation_available
    ch2.close()


def test_garbage_bytes_do_not_kill_server(native):
    plugin, srv, sock = native
    s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    s.connect(sock)
    s.sendall(b"GET / HTTP/1.1\r\nHost: x\r\n\r\n")  # not http/2
    time.sleep(0.1)
    s.close()

    s2 = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    s2.connect(sock)
    s2.sendall(b"\x00" * 64)
    s2.close()

    ch = grpc.insecure_channel(f"unix://{sock}")
    stub = dp.DevicePluginStub(ch)
    assert stub.GetDevicePluginOptions(dp.Empty(), timeout=5) is not None
    ch.close()


def test_restart_server_same_socket(tmp_path, fake_mi355x_8):
    sock = str(tmp_path / "r.sock")
    plugin = AMDGPUPlugin(resource="gpu", paths=fake_mi355x_8.paths)
    plugin.start()
    for _ in range(3):
        srv = NativePluginServer(plugin, sock)
        srv.start()
        ch = grpc.insecure_channel(f"unix://{sock}")
        stub = dp.DevicePluginStub(ch)
        assert stub.GetDevicePluginOptions(dp.Empty(), timeout=5) is not None
        ch.close()
        srv.stop()


def test_large_request_multi_frame(native):
    """A >100 KB Allocate request spans many DATA frames and needs
    connection/stream window updates — the server must reassemble it."""
    plugin, _, sock = native
    ch = grpc.insecure_channel(f"unix://{sock}")
    stub = dp.DevicePluginStub(ch)
    ids = sorted(plugin.devices)
    req = dp.AllocateRequest()
    # 1500 containers x ~80 B each ~= 120 KB on the wire
    for i in range(1500):
        cr = req.container_requests.add()
        cr.devices_ids.append(ids[i % len(ids)])
        cr.devices_ids.append(f"pad-{'x' * 64}-{i}")  # unknown: skipped
    assert len(req.SerializeToString()) > 100_000
    resp = stub.Allocate(req, timeout=30)
    assert len(resp.container_responses) == 1500
    for car in resp.container_responses:
        # kfd + 2 nodes for the one known device; unknown id skipped
        assert len(car.devices) == 3
    ch.close()


def test_large_response_multi_frame(native):
    """A large response (1500 containers x 3 specs) must stream out through
    the data provider across many frames."""
    plugin, _, sock = native
    ch = grpc.insecure_channel(f"unix://{sock}")
    stub = dp.DevicePluginStub(ch)
    ids = sorted(plugin.devices)
    req = dp.AllocateRequest()
    for i in range(1500):
        req.container_requests.add().devices_ids.extend(ids)
    resp = stub.Allocate(req, timeout=30)
    assert len(resp.container_responses) == 1500
    assert len(resp.container_responses[-1].devices) == 1 + 2 * len(ids)
    assert len(resp.SerializeToString()) > 500_000
    ch.close()


def test_fifty_concurrent_streams(native):
    """50 simultaneous ListAndWatch streams (one channel each, forcing
    separate connections is not possible with grpc pooling — mix grpc
    channels and raw go-wire connections): every stream must get the
    initial list AND a heartbeat push; no fd/slot exhaustion."""
    from k8s_device_plugin_amd.protos import deviceplugin as dp
    from k8s_device_plugin_amd.testing.goclient import GoWireClient

    plugin, srv, sock = native
    # 25 raw connections, each with one ListAndWatch stream
    raws = []
    for _ in range(25):
        c = GoWireClient(sock)
        sid = c.start_call("/v1beta1.DevicePlugin/ListAndWatch", b"")
        raws.append((c, c.conn.stream(sid)))
    # 25 grpc streams (share a channel/connection, distinct h2 streams)
    ch = grpc.insecure_channel(f"unix://{sock}")
    stub = dp.DevicePluginStub(ch)
    calls = [stub.ListAndWatch(dp.Empty()) for _ in range(25)]
    iters = [iter(c) for c in calls]
    for it in iters:
        first = next(it)
        assert len(first.devices) == 8
    for c, st in raws:
        assert c.conn.wait(lambda: len(st.grpc_messages()) >= 1, timeout=10)

    srv.heartbeat()  # one push must reach all 50 streams
    for it in iters:
        assert len(next(it).devices) == 8
    for c, st in raws:
        assert c.conn.wait(lambda: len(st.grpc_messages()) >= 2, timeout=10)

    for call in calls:
        call.cancel()
    ch.close()
    for c, _ in raws:
        c.close()


def test_connection_cap_protects_fds(native):
    """Beyond 256 connections the server refuses new ones instead of
    exhausting fds — and the kubelet's existing connection keeps
    working throughout."""
    import socket as socketmod

    plugin, srv, sock = native
    ch = grpc.insecure_channel(f"unix://{sock}")
    stub = dp.DevicePluginStub(ch)
    req = dp.AllocateRequest()
    req.container_requests.add().devices_ids.append(sorted(plugin.devices)[0])
    stub.Allocate(req, timeout=10)  # kubelet connection established

    flood = []
    try:
        for _ in range(300):
            s = socketmod.socket(socketmod.AF_UNIX, socketmod.SOCK_STREAM)
            try:
                s.connect(sock)
            except OSError:
                s.close()
                break
            flood.append(s)
        # regardless of how many the server kept, it must still serve
        for _ in range(20):
            resp = stub.Allocate(req, timeout=10)
            assert len(resp.container_responses[0].devices) == 3
    finally:
        for s in flood:
            s.close()
    ch.close()
    # and after the flood drains, fresh connections work again
    ch2 = grpc.insecure_channel(f"unix://{sock}")
    resp = dp.DevicePluginStub(ch2).Allocate(req, timeout=10)
    assert len(resp.container_responses[0].devices) == 3
    ch2.close()
