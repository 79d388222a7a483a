"""Partition-mode WRITE support (beyond the read-only reference:
internal/pkg/amdgpu/amdgpu.go:306-339).

All tests run against fake sysfs trees only — repartitioning is a
destructive node-wide operation; the module's double gate (allow=True AND
AMDXDP_ALLOW_REPARTITION=1) exists precisely so it can never fire on a
shared box by accident, and these tests verify the gate as much as the
mechanics.
"""

import os

import pytest

from k8s_device_plugin_amd.plugin import AMDGPUPlugin
from k8s_device_plugin_amd.protos import deviceplugin as dp
from k8s_device_plugin_amd.testing.fakesysfs import build_mi355x_node
from k8s_device_plugin_amd.topology import (
    PartitionError,
    SysPaths,
    apply_partition_mode,
    available_partition_modes,
    current_partition_modes,
    discover_gpus,
    set_partition_mode,
)


@pytest.fixture
def fs(tmp_path):
    return build_mi355x_node(str(tmp_path / "node"), n_gpus=2)


@pytest.fixture
def allow_env(monkeypatch):
    monkeypatch.setenv("AMDXDP_ALLOW_REPARTITION", "1")


def test_read_modes(fs):
    cur = current_partition_modes(fs.paths)
    assert len(cur) == 2
    assert all(v == ("SPX", "NPS1") for v in cur.values())
    comp, mem = available_partition_modes(fs.paths)
    assert comp == ["SPX", "DPX", "QPX", "CPX"]
    assert mem == ["NPS1", "NPS2"]


def test_write_refused_without_allow_flag(fs, allow_env):
    with pytest.raises(PartitionError, match="allow=True"):
        set_partition_mode(fs.paths, compute="CPX")
    # nothing changed
    assert all(v[0] == "SPX" for v in current_partition_modes(fs.paths).values())


def test_write_refused_without_env(fs, monkeypatch):
    monkeypatch.delenv("AMDXDP_ALLOW_REPARTITION", raising=False)
    with pytest.raises(PartitionError, match="AMDXDP_ALLOW_REPARTITION"):
        set_partition_mode(fs.paths, compute="CPX", allow=True)
    assert all(v[0] == "SPX" for v in current_partition_modes(fs.paths).values())


def test_write_validates_modes(fs, allow_env):
    with pytest.raises(PartitionError, match="unknown compute mode"):
        set_partition_mode(fs.paths, compute="XPX", allow=True)
    with pytest.raises(PartitionError, match="unknown memory mode"):
        set_partition_mode(fs.paths, memory="NPS3", allow=True)
    # NPS4 is a valid amdgpu mode but this (fake) hardware offers NPS1/NPS2
    with pytest.raises(PartitionError, match="not offered by hardware"):
        set_partition_mode(fs.paths, memory="NPS4", allow=True)
    with pytest.raises(PartitionError, match="nothing to do"):
        set_partition_mode(fs.paths, allow=True)
    with pytest.raises(PartitionError, match="unknown GPUs"):
        set_partition_mode(fs.paths, compute="CPX", allow=True,
                           pci_addrs=["9999:99:99.0"])


def test_set_compute_and_memory(fs, allow_env):
    modes = set_partition_mode(fs.paths, compute="CPX", memory="NPS2",
                               allow=True)
    assert all(v == ("CPX", "NPS2") for v in modes.values())
    # persisted in sysfs, visible to a fresh reader
    assert all(
        v == ("CPX", "NPS2") for v in current_partition_modes(fs.paths).values()
    )


def test_set_single_gpu(fs, allow_env):
    cur = current_partition_modes(fs.paths)
    target = sorted(cur)[0]
    modes = set_partition_mode(fs.paths, compute="DPX", allow=True,
                               pci_addrs=[target])
    assert modes[target][0] == "DPX"
    assert modes[sorted(cur)[1]][0] == "SPX"


def test_apply_rediscovers(fs, allow_env):
    modes, devices = apply_partition_mode(fs.paths, compute="CPX",
                                          memory="NPS2", allow=True)
    assert all(v == ("CPX", "NPS2") for v in modes.values())
    # rediscovery reflects the new mode on the physical GPUs
    assert all(d.compute_partition == "cpx" for d in devices.values())
    assert all(d.memory_partition == "nps2" for d in devices.values())


def test_full_pipeline_mode_flip_rebuilds_serving_state(tmp_path, allow_env):
    """SPX -> CPX end-to-end: write the mode, simulate the kernel's xcp
    fan-out (on real hardware the amdgpu driver re-creates the platform
    devices and kfd nodes; fake sysfs stands in for it), then one plugin
    heartbeat must rediscover the 16 logical devices, re-init the
    allocator, and push the new list to open streams."""
    root = str(tmp_path / "node")
    build_mi355x_node(root, n_gpus=2)
    paths = SysPaths(root)

    plugin = AMDGPUPlugin(resource="gpu", paths=paths)
    plugin.start()
    assert len(plugin.devices) == 2

    class _Ctx:
        def is_active(self):
            return True

    stream = plugin.ListAndWatch(dp.Empty(), _Ctx())
    first = next(stream)
    assert len(first.devices) == 2

    # operator flips the mode...
    set_partition_mode(paths, compute="CPX", memory="NPS1", allow=True)
    # ...the kernel re-enumerates (simulated: rebuild the tree CPX-style,
    # 8 partitions per GPU)
    import shutil

    shutil.rmtree(root)
    build_mi355x_node(root, n_gpus=2, partitions_per_gpu=8,
                      compute_partition="CPX", memory_partition="NPS1")

    # ...and the next heartbeat picks everything up without a restart
    plugin.heartbeat()
    resp = next(stream)
    assert len(resp.devices) == 16
    assert len(plugin.devices) == 16
    assert not plugin.allocator_init_error
    # allocator was re-inited over the new fan-out: a whole-GPU request
    # packs 8 partitions of ONE physical GPU (reference oracle semantics)
    ids = sorted(plugin.devices)
    resp2 = plugin.GetPreferredAllocation(
        dp.PreferredAllocationRequest(
            container_requests=[
                dp.ContainerPreferredAllocationRequest(
                    available_deviceIDs=ids,
                    must_include_deviceIDs=[],
                    allocation_size=8,
                )
            ]
        ),
        None,
    )
    chosen = list(resp2.container_responses[0].deviceIDs)
    assert len(chosen) == 8
    parents = {plugin.devices[i].dev_id for i in chosen}
    assert len(parents) == 1, f"not packed on one GPU: {chosen}"
    plugin.stop()


def test_cli_show_and_gated_set(fs, monkeypatch, capsys):
    from k8s_device_plugin_amd.cli import partition_main

    monkeypatch.delenv("AMDXDP_ALLOW_REPARTITION", raising=False)
    assert partition_main(["--sysroot", fs.paths.root]) == 0
    out = capsys.readouterr().out
    assert '"SPX"' in out and '"available_compute"' in out

    # set without env -> refused, exit 1
    assert partition_main(
        ["--sysroot", fs.paths.root, "--compute", "CPX", "--allow"]
    ) == 1

    monkeypatch.setenv("AMDXDP_ALLOW_REPARTITION", "1")
    assert partition_main(
        ["--sysroot", fs.paths.root, "--compute", "CPX", "--allow"]
    ) == 0
    assert all(
        v[0] == "CPX" for v in current_partition_modes(fs.paths).values()
    )


def test_mode_flip_updates_labels(tmp_path, allow_env):
    """Partition flip + labeller refresh: after SPX->CPX the recomputed
    labels must advertise the new partition config and per-partition
    values (the labeller's --refresh-interval path picks this up without
    a restart; the ROCm labeller needs a pod restart)."""
    import shutil

    from k8s_device_plugin_amd.labeller import generate_labels
    from k8s_device_plugin_amd.labeller.labels import LABEL_KINDS

    root = str(tmp_path / "node")
    build_mi355x_node(root, n_gpus=2)
    paths = SysPaths(root)
    enabled = {k: True for k in LABEL_KINDS}

    before = generate_labels(enabled, paths)
    assert before["amd.com/gpu.compute-memory-partition"] == "spx_nps1"
    assert before["amd.com/gpu.cu-count"] == "256"

    set_partition_mode(paths, compute="CPX", memory="NPS2", allow=True)
    shutil.rmtree(root)
    build_mi355x_node(root, n_gpus=2, partitions_per_gpu=8,
                      compute_partition="CPX", memory_partition="NPS2")

    after = generate_labels(enabled, paths)
    assert after["amd.com/gpu.compute-memory-partition"] == "cpx_nps2"
    # per-partition CU count: 256/8 = 32 per logical device
    assert after["amd.com/gpu.cu-count"] == "32"
    assert after["amd.com/gpu.vram"] == "36G"  # 288G / 8 partitions
