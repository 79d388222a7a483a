"""Device-plugin CLI end-to-end on a fake sysroot: registration with a stub
kubelet, heartbeat pulse, graceful shutdown."""

import os
import signal
import subprocess
import sys
import time

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_cli_device_plugin_registers_and_shuts_down(tmp_path, fake_mi355x_8):
    from k8s_device_plugin_amd.testing.stub_kubelet import StubKubelet

    dp_dir = str(tmp_path / "dp")
    kubelet = StubKubelet(dp_dir).start()
    proc = subprocess.Popen(
        [
            sys.executable, "-m", "k8s_device_plugin_amd.cli",
            "--pulse", "1",
            "--kubelet-dir", dp_dir,
            "--sysroot", fake_mi355x_8.paths.root,
        ],
        cwd=REPO,
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
        text=True,
    )
    try:
        reg = kubelet.wait_for_registration(timeout=30)
        assert reg.resource_name == "amd.com/gpu"
        stub = kubelet.connect(reg.endpoint)
        from k8s_device_plugin_amd.protos import deviceplugin as dp

        call = stub.ListAndWatch(dp.Empty())
        it = iter(call)
        first = next(it)
        assert len(first.devices) == 8
        # pulse=1: a health refresh must arrive within a few seconds
        second = next(it)
        assert len(second.devices) == 8
        assert {d.health for d in second.devices} == {"Healthy"}
        call.cancel()

        proc.send_signal(signal.SIGTERM)
        assert proc.wait(timeout=15) == 0
    finally:
        if proc.poll() is None:
            proc.kill()
            proc.wait()
        kubelet.stop()


def test_cli_bad_strategy(tmp_path, fake_mi355x_8):
    from k8s_device_plugin_amd.cli import device_plugin_main

    rc = device_plugin_main([
        "--resource_naming_strategy", "bogus",
        "--sysroot", fake_mi355x_8.paths.root,
    ])
    assert rc == 1


def test_cli_no_driver_exit2(tmp_path):
    from k8s_device_plugin_amd.cli import device_plugin_main
    import k8s_device_plugin_amd.cli as cli_mod

    # empty sysroot: /sys/class/kfd missing -> exit 2 after the gate.
    # Each monotonic() call advances 100 s so the 60 s wait window expires
    # immediately, and sleep is a no-op.
    ticks = iter(range(0, 100000, 100))
    orig_mono, orig_sleep = cli_mod.time.monotonic, cli_mod.time.sleep
    cli_mod.time.monotonic = lambda: float(next(ticks))
    cli_mod.time.sleep = lambda s: None
    try:
        rc = device_plugin_main(["--sysroot", str(tmp_path / "empty")])
    finally:
        cli_mod.time.monotonic = orig_mono
        cli_mod.time.sleep = orig_sleep
    assert rc == 2


def test_cli_dump(tmp_path, fake_mi355x_8, capsys):
    import json

    from k8s_device_plugin_amd.cli import device_plugin_main

    rc = device_plugin_main(["--dump", "--sysroot", fake_mi355x_8.paths.root])
    assert rc == 0
    out = json.loads(capsys.readouterr().out)
    assert len(out["devices"]) == 8
    assert out["homogeneous"] is True
    assert out["partition_configs"] == {"spx_nps1": 8}
    assert len(out["allocator"]["groups"]) == 8
    assert len(out["allocator"]["pair_weights"]) == 28


def test_cli_mixed_strategy_cpx(tmp_path, fake_mi355x_cpx):
    """Mixed naming on a CPX node: the daemon advertises amd.com/cpx_nps2
    with all 64 partitions."""
    from k8s_device_plugin_amd.testing.stub_kubelet import StubKubelet
    from k8s_device_plugin_amd.protos import deviceplugin as dp

    dp_dir = str(tmp_path / "dp")
    kubelet = StubKubelet(dp_dir).start()
    proc = subprocess.Popen(
        [sys.executable, "-m", "k8s_device_plugin_amd.cli",
         "--resource_naming_strategy", "mixed",
         "--kubelet-dir", dp_dir,
         "--sysroot", fake_mi355x_cpx.paths.root],
        cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
    )
    try:
        reg = kubelet.wait_for_registration(timeout=30)
        assert reg.resource_name == "amd.com/cpx_nps2"
        stub = kubelet.connect(reg.endpoint)
        call = stub.ListAndWatch(dp.Empty())
        first = next(iter(call))
        assert len(first.devices) == 64
        call.cancel()
        proc.send_signal(signal.SIGTERM)
        assert proc.wait(timeout=15) == 0
    finally:
        if proc.poll() is None:
            proc.kill()
            proc.wait()
        kubelet.stop()
