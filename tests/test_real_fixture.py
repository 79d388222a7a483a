"""Parser regression against kfd content transcribed from a real MI355X
(tests/fixtures/real_mi355x; provenance in fixtures/README.md)."""

import os

from k8s_device_plugin_amd.topology import KFDTopology, SysPaths

FIXTURE = os.path.join(
    os.path.dirname(os.path.abspath(__file__)), "fixtures", "real_mi355x"
)


def test_parse_real_mi355x_node():
    topo = KFDTopology.load(SysPaths(FIXTURE))
    assert set(topo.nodes) == {0, 6}

    cpu = topo.nodes[0]
    assert not cpu.is_gpu
    assert cpu.properties["cpu_cores_count"] == 128

    gpu = topo.nodes[6]
    assert gpu.is_gpu
    assert gpu.properties["gfx_target_version"] == 90500
    assert gpu.properties["device_id"] == 0x75A3
    assert gpu.properties["num_xcc"] == 8
    assert gpu.properties["lds_size_in_kb"] == 160
    assert gpu.render_minor == 160
    assert gpu.hive_id == 11964924489695451825
    assert gpu.vram_bytes == 309220868096  # 288 GB HBM3E
    assert gpu.simd_count == 1024 and gpu.cu_count == 256
    # devID decode: location_id 62464 = 0xF400 -> bus 0xf4
    assert gpu.dev_id() == "0000:f4:00:0"
    link = gpu.io_links[0]
    assert link.type == 2 and link.node_to == 1 and link.max_bandwidth == 64000

    assert topo.render_minor_to_dev_id() == {160: "0000:f4:00:0"}
    assert topo.render_minor_to_node_id() == {160: 6}
