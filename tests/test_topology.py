"""kfd topology walker + GPU discovery tests (hermetic, fake /sys trees).

Mirrors the reference's fixture-driven test strategy (SURVEY.md §4;
reference: internal/pkg/amdgpu/amdgpu_test.go) over synthesized gfx950 trees.
"""

import pytest

from k8s_device_plugin_amd.topology import (
    KFDTopology,
    DriverUnavailableError,
    discover_gpus,
    is_homogeneous,
    unique_partition_config_count,
    is_compute_partition_supported,
    is_memory_partition_supported,
    count_gpus_from_topology,
    simple_health_check,
    parse_properties_text,
    SysPaths,
)
from k8s_device_plugin_amd.testing.fakesysfs import (
    FakeSysfs,
    MI355X_VRAM_BYTES,
    build_mi355x_node,
)


def test_parse_properties_text():
    props = parse_properties_text(
        "cpu_cores_count 0\nsimd_count 1024\nhive_id 7455128887705989632\n"
        "badline\nname two words\n"
    )
    assert props["cpu_cores_count"] == 0
    assert props["simd_count"] == 1024
    assert props["hive_id"] == 7455128887705989632
    assert "name" not in props


def test_walker_basic(fake_mi355x_8):
    topo = KFDTopology.load(fake_mi355x_8.paths)
    # 2 CPU nodes + 8 GPU nodes
    assert len(topo.nodes) == 10
    gpus = topo.gpu_nodes()
    assert len(gpus) == 8
    for node in gpus:
        assert node.vram_bytes == MI355X_VRAM_BYTES
        assert node.cu_count == 256
        assert node.hive_id != 0
        # 7 xGMI point-to-point links per GPU on an 8-GPU hive
        xgmi = [l for l in node.all_links() if l.type == 11]
        assert len(xgmi) == 7


def test_dev_id_decode(fake_mi355x_8):
    topo = KFDTopology.load(fake_mi355x_8.paths)
    m = topo.render_minor_to_dev_id()
    # GPU 0: bus 0x0c -> location_id 0x0c00 -> devID 0000:0c:00:0
    assert m[128] == "0000:0c:00:0"
    assert m[135] == "0000:13:00:0"
    nodemap = topo.render_minor_to_node_id()
    assert nodemap[128] == 2 and nodemap[135] == 9


def test_discover_physical(fake_mi355x_8):
    devs = discover_gpus(fake_mi355x_8.paths)
    assert len(devs) == 8
    d = devs["0000:0c:00.0"]
    assert d.card == 0 and d.render_d == 128
    assert d.dev_id == "0000:0c:00:0"
    assert d.compute_partition == "spx" and d.memory_partition == "nps1"
    assert d.numa_node == 0 and d.node_id == 2
    assert not d.is_partition
    # NUMA split across the node
    assert devs["0000:13:00.0"].numa_node == 1


def test_discover_cpx_fanout(fake_mi355x_cpx):
    devs = discover_gpus(fake_mi355x_cpx.paths)
    # 8 physical + 8*7 partitions = 64 schedulable devices
    assert len(devs) == 64
    partitions = [d for d in devs.values() if d.is_partition]
    assert len(partitions) == 56
    # partitions inherit partition type + numa from parent with same devID
    parent = devs["0000:0c:00.0"]
    children = [d for d in partitions if d.dev_id == parent.dev_id]
    assert len(children) == 7
    for c in children:
        assert c.compute_partition == "cpx"
        assert c.memory_partition == "nps2"
        assert c.numa_node == parent.numa_node
    assert is_homogeneous(devs)
    assert unique_partition_config_count(devs) == {"cpx_nps2": 64}


def test_discover_skips_invalid_renderd(tmp_path):
    fs = build_mi355x_node(str(tmp_path / "r"), n_gpus=2)
    # platform device whose renderD has no kfd node -> must be skipped
    fs.add_partition(99, node_id=0, parent_index=0, card=30, render_minor=250,
                     in_kfd=False)
    devs = discover_gpus(fs.paths)
    assert len(devs) == 2
    assert "amdgpu_xcp_99" not in devs


def test_discover_driver_unavailable(tmp_path):
    paths = SysPaths(str(tmp_path / "nothing"))
    with pytest.raises(DriverUnavailableError):
        discover_gpus(paths)
    assert discover_gpus(paths, strict=False) == {}


def test_heterogeneous_counts(tmp_path):
    fs = FakeSysfs(str(tmp_path / "r"))
    fs.add_cpu_node(0)
    fs.add_physical_gpu(0, node_id=1, compute_partition="SPX", memory_partition="NPS1")
    fs.add_physical_gpu(1, node_id=2, compute_partition="CPX", memory_partition="NPS4")
    devs = discover_gpus(fs.paths)
    assert not is_homogeneous(devs)
    assert unique_partition_config_count(devs) == {"spx_nps1": 1, "cpx_nps4": 1}


def test_partition_support_probes(fake_mi355x_8, tmp_path):
    assert is_compute_partition_supported(fake_mi355x_8.paths)
    assert is_memory_partition_supported(fake_mi355x_8.paths)
    fs = FakeSysfs(str(tmp_path / "r2"))
    fs.add_cpu_node(0)
    fs.add_physical_gpu(0, node_id=1, partition_caps=False)
    assert not is_compute_partition_supported(fs.paths)
    assert not is_memory_partition_supported(fs.paths)


def test_count_and_health(fake_mi355x_8, tmp_path):
    assert count_gpus_from_topology(fake_mi355x_8.paths) == 8
    assert simple_health_check(fake_mi355x_8.paths)
    fs = FakeSysfs(str(tmp_path / "cpuonly"))
    fs.add_cpu_node(0)
    assert count_gpus_from_topology(fs.paths) == 0
    assert not simple_health_check(fs.paths)
