"""Multi-process (gloo, world_size=2) coverage of the bench/dist path —
the exact launch shape the driver uses for multi-GPU scaling runs."""

import json
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def test_bench_torchrun_world2():
    # bench uses live sysfs when present; a box exposing fewer than 2
    # schedulable GPUs cannot satisfy --gpus 2 (by design), so skip there
    import pytest

    from k8s_device_plugin_amd.topology import (
        SysPaths,
        discover_gpus,
        simple_health_check,
    )

    live = SysPaths("/")
    if os.path.isdir(live.kfd_class) and simple_health_check(live):
        healthy = [
            d for d in discover_gpus(live, strict=False).values() if d.kfd_backed
        ]
        if len(healthy) < 2:
            pytest.skip(f"live sysfs exposes only {len(healthy)} schedulable GPU(s)")

    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1",
            "--master-port", str(_free_port()),
            os.path.join(REPO, "bench.py"),
            "--gpus", "2", "--steps", "5", "--warmup", "1",
        ],
        capture_output=True, text=True, timeout=420, env=env, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-3000:]
    json_lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, f"exactly one JSON line expected:\n{out.stdout}"
    rec = json.loads(json_lines[0])
    assert rec["n_gpus"] == 2
    assert rec["steps"] == 5
    assert rec["scaling"] == "weak"
    assert rec["value"] > 0
    assert rec["ms_per_step"] > 0
