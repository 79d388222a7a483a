"""Resource naming strategy tests (reference: cmd/k8s-device-plugin/main.go:42-91)."""

import pytest

from k8s_device_plugin_amd.plugin.resources import (
    StrategyError,
    get_resource_list,
    parse_strategy,
)
from k8s_device_plugin_amd.topology import discover_gpus
from k8s_device_plugin_amd.testing.fakesysfs import FakeSysfs, build_mi355x_node


def test_parse_strategy():
    assert parse_strategy("single") == "single"
    assert parse_strategy("mixed") == "mixed"
    with pytest.raises(StrategyError):
        parse_strategy("bogus")


def test_single_homogeneous(fake_mi355x_8):
    devs = discover_gpus(fake_mi355x_8.paths)
    assert get_resource_list(devs, "single") == ["gpu"]


def test_mixed_homogeneous_partitioned(fake_mi355x_cpx):
    devs = discover_gpus(fake_mi355x_cpx.paths)
    assert get_resource_list(devs, "mixed") == ["cpx_nps2"]


def test_mixed_unpartitioned(tmp_path):
    fs = FakeSysfs(str(tmp_path / "r"))
    fs.add_cpu_node(0)
    # GPU with no partition files at all -> "gpu" under both strategies
    pci = fs.add_physical_gpu(0, node_id=1)
    import os

    for f in ("current_compute_partition", "current_memory_partition"):
        os.unlink(os.path.join(fs.paths.amdgpu_pci, pci, f))
    devs = discover_gpus(fs.paths)
    assert get_resource_list(devs, "mixed") == ["gpu"]
    assert get_resource_list(devs, "single") == ["gpu"]


def test_heterogeneous(tmp_path):
    fs = FakeSysfs(str(tmp_path / "r"))
    fs.add_cpu_node(0)
    fs.add_physical_gpu(0, node_id=1, compute_partition="SPX", memory_partition="NPS1")
    fs.add_physical_gpu(1, node_id=2, compute_partition="CPX", memory_partition="NPS4")
    devs = discover_gpus(fs.paths)
    with pytest.raises(StrategyError):
        get_resource_list(devs, "single")
    assert get_resource_list(devs, "mixed") == ["cpx_nps4", "spx_nps1"]


def test_empty_devices():
    assert get_resource_list({}, "single") == []
