import os
import sys

import pytest

# Make the repo root importable regardless of how pytest is invoked.
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real AMD GPU (run on an MI355X box)"
    )
    # build the CPU-buildable native extensions if missing so a fresh
    # checkout passes without a separate build step (g++ only; the HIP
    # probe needs hipcc and is attempted best-effort)
    from k8s_device_plugin_amd.native import build as nb

    nb.build_drmctl()
    nb.build_fastserver()
    nb.build_h2tool()
    try:
        nb.build_healthprobe()
    except Exception:
        pass  # no hipcc on this machine; gpu-marked tests need it anyway


@pytest.fixture
def fake_mi355x_8(tmp_path):
    """8*MI355X SPX/NPS1 node, single xGMI hive."""
    from k8s_device_plugin_amd.testing.fakesysfs import build_mi355x_node

    return build_mi355x_node(str(tmp_path / "root"))


@pytest.fixture
def fake_mi355x_cpx(tmp_path):
    """8*MI355X in CPX/NPS2: 8 physical GPUs x 8 partitions = 64 devices."""
    from k8s_device_plugin_amd.testing.fakesysfs import build_mi355x_node

    return build_mi355x_node(
        str(tmp_path / "root"),
        partitions_per_gpu=8,
        compute_partition="CPX",
        memory_partition="NPS2",
    )
