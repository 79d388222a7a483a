#!/usr/bin/env python3
"""Live-daemon soak driver: the real CLI daemon under sustained load.

Starts `python -m k8s_device_plugin_amd.cli` (native server) against a
stub kubelet, then hammers it from client threads while cycling streams,
restarting the kubelet, and (r02) letting the heartbeat run the deep GPU
probe with performance floors every few beats.  Prints one JSON summary.

Usage (GPU box): python scripts/soak.py --minutes 10 --pulse 2 --deep-every 3
CPU boxes work too (fake sysfs; deep probe skipped by the daemon).
"""

from __future__ import annotations

import argparse
import json
import os
import signal
import subprocess
import sys
import tempfile
import threading
import time
import urllib.request

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--minutes", type=float, default=10.0)
    ap.add_argument("--pulse", type=int, default=2)
    ap.add_argument("--deep-every", type=int, default=3)
    ap.add_argument("--metrics-port", type=int, default=19793)
    ap.add_argument("--kubelet-restart-s", type=float, default=240.0)
    ap.add_argument("--server", default="native", choices=["native", "python"])
    args = ap.parse_args()

    from k8s_device_plugin_amd.protos import deviceplugin as dp
    from k8s_device_plugin_amd.testing.stub_kubelet import StubKubelet
    from k8s_device_plugin_amd.topology import SysPaths, simple_health_check

    live = os.path.isdir("/sys/class/kfd") and simple_health_check(SysPaths("/"))
    tmp = tempfile.TemporaryDirectory(prefix="amdxdp-soak-")
    root = tmp.name
    if live:
        sysroot = "/"
    else:
        from k8s_device_plugin_amd.testing.fakesysfs import build_mi355x_node

        build_mi355x_node(os.path.join(root, "fakesys"), n_gpus=8)
        sysroot = os.path.join(root, "fakesys")

    dp_dir = os.path.join(root, "dp")
    os.makedirs(dp_dir)
    kubelet = StubKubelet(dp_dir).start()

    cdi_dir = os.path.join(root, "cdi")
    daemon = subprocess.Popen(
        [sys.executable, "-m", "k8s_device_plugin_amd.cli",
         "--sysroot", sysroot, "--kubelet-dir", dp_dir,
         "-pulse", str(args.pulse),
         "--deep-probe-every", str(args.deep_every),
         "--server", args.server,
         "--cdi", "--cdi-dir", cdi_dir,
         "--metrics-port", str(args.metrics_port), "-v"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        cwd=REPO,
    )
    daemon_log = []

    def log_reader():
        for line in daemon.stdout:
            daemon_log.append(line)

    threading.Thread(target=log_reader, daemon=True).start()

    reg = kubelet.wait_for_registration(timeout=60)
    sock = os.path.join(dp_dir, reg.endpoint)

    import grpc

    stop = threading.Event()
    stats = {"allocates": 0, "preferred": 0, "stream_updates": 0,
             "stream_cycles": 0, "kubelet_restarts": 0, "errors": []}
    lock = threading.Lock()

    def get_ids():
        ch = grpc.insecure_channel(f"unix://{sock}")
        stub = dp.DevicePluginStub(ch)
        call = stub.ListAndWatch(dp.Empty())
        first = next(iter(call))
        call.cancel()
        ch.close()
        return sorted(d.ID for d in first.devices if d.health == "Healthy")

    ids = get_ids()
    assert ids, "no healthy devices"

    def alloc_loop(n):
        ch = grpc.insecure_channel(f"unix://{sock}")
        stub = dp.DevicePluginStub(ch)
        req = dp.AllocateRequest()
        req.container_requests.add().devices_ids.extend(ids[: 1 + n % len(ids)])
        preq = dp.PreferredAllocationRequest()
        cr = preq.container_requests.add()
        cr.available_deviceIDs.extend(ids)
        cr.allocation_size = 1 + n % len(ids)
        while not stop.is_set():
            try:
                stub.Allocate(req, timeout=10)
                with lock:
                    stats["allocates"] += 1
                if n % 2 == 0:
                    stub.GetPreferredAllocation(preq, timeout=10)
                    with lock:
                        stats["preferred"] += 1
            except grpc.RpcError as e:
                if stop.is_set():
                    break
                with lock:
                    stats["errors"].append(f"alloc[{n}]: {e.code()}")
                time.sleep(0.5)
        ch.close()

    def stream_loop():
        while not stop.is_set():
            try:
                ch = grpc.insecure_channel(f"unix://{sock}")
                stub = dp.DevicePluginStub(ch)
                call = stub.ListAndWatch(dp.Empty())
                with lock:
                    stats["stream_cycles"] += 1
                t_end = time.monotonic() + 20
                for _ in call:
                    with lock:
                        stats["stream_updates"] += 1
                    if stop.is_set() or time.monotonic() > t_end:
                        break
                call.cancel()
                ch.close()
            except grpc.RpcError:
                if not stop.is_set():
                    time.sleep(0.5)

    threads = [threading.Thread(target=alloc_loop, args=(i,)) for i in range(4)]
    threads.append(threading.Thread(target=stream_loop))
    for t in threads:
        t.start()

    deadline = time.monotonic() + args.minutes * 60
    next_restart = time.monotonic() + args.kubelet_restart_s
    while time.monotonic() < deadline:
        time.sleep(1)
        if daemon.poll() is not None:
            stats["errors"].append(f"daemon died rc={daemon.returncode}")
            break
        if time.monotonic() >= next_restart:
            kubelet.stop()
            kubelet = StubKubelet(dp_dir).start()
            try:
                kubelet.wait_for_registration(timeout=30)
                with lock:
                    stats["kubelet_restarts"] += 1
            except Exception as e:
                stats["errors"].append(f"re-register failed: {e}")
            next_restart = time.monotonic() + args.kubelet_restart_s

    stop.set()
    for t in threads:
        t.join(timeout=10)

    metrics = {}
    try:
        with urllib.request.urlopen(
            f"http://127.0.0.1:{args.metrics_port}/metrics", timeout=5
        ) as r:
            for line in r.read().decode().splitlines():
                if line.startswith("amdgpu_dp_") and not line.startswith("#"):
                    k, _, v = line.rpartition(" ")
                    metrics[k] = float(v)
    except Exception as e:
        stats["errors"].append(f"metrics scrape: {e}")

    daemon.send_signal(signal.SIGTERM)
    try:
        daemon.wait(timeout=15)
    except subprocess.TimeoutExpired:
        daemon.kill()
        stats["errors"].append("daemon did not exit on SIGTERM")

    deep_lines = [l.strip() for l in daemon_log if "deep probe" in l.lower()]
    err_lines = [l.strip() for l in daemon_log
                 if "ERROR" in l or "Traceback" in l]
    cdi_ok = bool(
        [f for f in (os.listdir(cdi_dir) if os.path.isdir(cdi_dir) else [])]
    )
    out = {
        "live_sysfs": live,
        "server_impl": args.server,
        "minutes": args.minutes,
        "pulse_s": args.pulse,
        "deep_probe_every": args.deep_every,
        "devices": len(ids),
        **{k: v for k, v in stats.items() if k != "errors"},
        "error_count": len(stats["errors"]),
        "errors_sample": stats["errors"][:10],
        "daemon_exit": daemon.returncode,
        "cdi_spec_written": cdi_ok,
        "deep_probe_log_lines": len(deep_lines),
        "deep_probe_lines_sample": deep_lines[:6],
        "daemon_error_lines": len(err_lines),
        "daemon_error_sample": err_lines[:6],
        "metrics": metrics,
    }
    print(json.dumps(out, indent=2))
    kubelet.stop()
    tmp.cleanup()
    return 0 if not stats["errors"] and daemon.returncode == 0 else 1


if __name__ == "__main__":
    sys.exit(main())
