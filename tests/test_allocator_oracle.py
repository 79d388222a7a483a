"""Golden-port of the reference allocator's frozen policy oracles.

The reference pins its best-effort policy behavior with table-driven tests
over three checked-in kfd topologies and EXACT expected device-ID sets
(reference: internal/pkg/allocator/besteffort_policy_test.go:25-231,
device_test.go:43-67).  Round-1 covered equivalent synthetic scenarios;
this file replays the reference's own hard cases so "same grouping/BFS
semantics" is demonstrated, not asserted (Missing #2).

The topologies are stored as compact JSON link specs (tests/fixtures/
oracle/*.json) transcribed from the reference's testdata trees —
node id -> {drm_render_minor, cpu_cores_count, gfx_target_version,
[node_from, node_to, type] link tuples} — exactly the fields the policy
reads (reference: device.go:160-253).  A builder materializes them into
fake sysfs trees so the real KFDTopology.load path is exercised:
  - topo-mi210-xgmi-pcie: 8 whole MI210 GPUs, two 4-GPU xGMI islands
    bridged by PCIe, 2 NUMA nodes;
  - topo-mi300-cpx: 8 MI300X GPUs x 8 CPX partitions, but only 63 kfd
    nodes — the reference's synthetic device generator drops the last
    partition (endNodeId=64 truncation, device_test.go:56-58), making
    GPU8 a 7-partition group, which is what the anti-fragmentation
    ordering expectations hinge on;
  - topology-parsing-mi308: 4 GPUs x 8 partitions (structural cases).

Device synthesis mirrors testInfo.getTestDevices (device_test.go:43-67):
ids "test<i+1>" for the first partition of GPU i and "amdgpu_xcp_<i*8+j>"
for the rest, NodeId counting up from startNodeId, NumaNode = i // (devCount
/ numanodeCount), DevId = str(i).
"""

import json
import os

import pytest

from k8s_device_plugin_amd.allocator import BestEffortPolicy
from k8s_device_plugin_amd.allocator.weights import compute_pair_weights
from k8s_device_plugin_amd.topology import KFDTopology
from k8s_device_plugin_amd.topology.discovery import GPUDevice
from k8s_device_plugin_amd.topology.sysfs import SysPaths

FIXDIR = os.path.join(os.path.dirname(__file__), "fixtures", "oracle")


def _materialize(spec_file: str, root: str) -> SysPaths:
    """Expand a JSON link spec into a kfd-shaped sysfs tree."""
    with open(os.path.join(FIXDIR, spec_file)) as f:
        spec = json.load(f)
    nodes_dir = os.path.join(root, "sys", "class", "kfd", "kfd", "topology", "nodes")
    for nid, node in spec.items():
        nd = os.path.join(nodes_dir, nid)
        os.makedirs(nd, exist_ok=True)
        with open(os.path.join(nd, "properties"), "w") as f:
            f.write(f"cpu_cores_count {node['cpu_cores']}\n")
            f.write(f"gfx_target_version {node['gfx']}\n")
            f.write(f"drm_render_minor {node['render_minor']}\n")
        for k, (frm, to, typ) in enumerate(node["links"]):
            ld = os.path.join(nd, "io_links", str(k))
            os.makedirs(ld, exist_ok=True)
            with open(os.path.join(ld, "properties"), "w") as f:
                f.write(f"type {typ}\nnode_from {frm}\nnode_to {to}\n")
    paths = SysPaths(root=root)
    return paths


def _synth_devices(topo: KFDTopology, dev_count: int, parts_per_dev: int,
                   numa_count: int, start_node: int, end_node: int):
    """The reference's synthetic device builder (device_test.go:43-67)."""
    devs = []
    node_id = start_node
    numa_div = dev_count // numa_count
    for i in range(dev_count):
        for j in range(parts_per_dev):
            if node_id > end_node:
                break
            dev_id = f"test{i + 1}" if j == 0 else f"amdgpu_xcp_{i * 8 + j}"
            node = topo.nodes.get(node_id)
            devs.append(
                GPUDevice(
                    id=dev_id,
                    dev_id=str(i),
                    numa_node=i // numa_div,
                    node_id=node_id,
                    render_d=node.render_minor if node else 0,
                )
            )
            node_id += 1
    return devs


def _make_policy(tmp_path_factory, spec_file, dev_count, parts, numa, start, end):
    root = str(tmp_path_factory.mktemp(spec_file.split(".")[0]))
    paths = _materialize(spec_file, root)
    topo = KFDTopology.load(paths)
    devices = _synth_devices(topo, dev_count, parts, numa, start, end)
    policy = BestEffortPolicy()
    policy.init(devices, topology=topo)
    return policy, devices, topo


@pytest.fixture(scope="module")
def mi210(tmp_path_factory):
    return _make_policy(
        tmp_path_factory, "topo_mi210_xgmi_pcie.json", 8, 1, 2, 2, 9
    )


@pytest.fixture(scope="module")
def mi300cpx(tmp_path_factory):
    return _make_policy(
        tmp_path_factory, "topo_mi300_cpx.json", 8, 8, 2, 2, 64
    )


@pytest.fixture(scope="module")
def mi308(tmp_path_factory):
    return _make_policy(
        tmp_path_factory, "topo_mi308.json", 4, 8, 2, 2, 33
    )


# ---- topology 2: MI210, 8 whole GPUs, xGMI islands + PCIe ----
# expected sets: besteffort_policy_test.go:77-103

MI210_CASES = [
    ("allocate 1", None, [], 1, ["test1"]),
    ("allocate 3", None, [], 3, ["test1", "test2", "test3"]),
    ("allocate 5", None, [], 5, ["test1", "test2", "test3", "test4", "test5"]),
    (
        "allocate 3 same numa",
        ["test3", "test4", "test5", "test6", "test7", "test8"],
        [],
        3,
        ["test5", "test6", "test7"],
    ),
]


@pytest.mark.parametrize("desc,available,required,size,expected",
                         MI210_CASES, ids=[c[0] for c in MI210_CASES])
def test_mi210_oracle(mi210, desc, available, required, size, expected):
    policy, devices, _ = mi210
    av = available if available else sorted(
        (d.id for d in devices), key=lambda s: next(
            d.node_id for d in devices if d.id == s)
    )
    out = policy.allocate(av, required, size)
    assert sorted(out) == sorted(expected), f"{desc}: got {sorted(out)}"


# ---- topology 3: MI300X CPX, 8 GPUs x 8 partitions (63 kfd nodes) ----
# expected sets: besteffort_policy_test.go:105-160

MI300_CASES = [
    ("allocate 1", None, [], [], 1, ["test8"]),
    ("allocate 3", None, [], [], 3, ["test8", "amdgpu_xcp_57", "amdgpu_xcp_58"]),
    (
        "allocate 5", None, [], [], 5,
        ["test8", "amdgpu_xcp_57", "amdgpu_xcp_58", "amdgpu_xcp_59",
         "amdgpu_xcp_60"],
    ),
    (
        "allocate 3 same numa",
        ["test3", "test4", "test5", "test6", "test7", "test8"], [], [],
        3,
        ["test5", "test6", "test7"],
    ),
    (
        "allocate 3 required same numa",
        ["test3", "test4", "test5", "test6", "test7", "test8"], [],
        ["test5"],
        3,
        ["test5", "test6", "test7"],
    ),
    ("allocate 30 (size only)", None, [], [], 30, None),
    (
        "allocate 8 -> one whole GPU",
        None, [], [], 8,
        ["test1", "amdgpu_xcp_1", "amdgpu_xcp_2", "amdgpu_xcp_3",
         "amdgpu_xcp_4", "amdgpu_xcp_5", "amdgpu_xcp_6", "amdgpu_xcp_7"],
    ),
    (
        "allocate 7 -> the truncated GPU8 group",
        None, [], [], 7,
        ["test8", "amdgpu_xcp_57", "amdgpu_xcp_58", "amdgpu_xcp_59",
         "amdgpu_xcp_60", "amdgpu_xcp_61", "amdgpu_xcp_62"],
    ),
    (
        "allocate 4 with GPU8 partly taken",
        None, ["test8", "amdgpu_xcp_57", "amdgpu_xcp_58"], [], 4,
        ["amdgpu_xcp_59", "amdgpu_xcp_60", "amdgpu_xcp_61", "amdgpu_xcp_62"],
    ),
    (
        "allocate 10 across fragmented groups",
        None,
        ["test1", "test2", "test3", "test4", "test8", "amdgpu_xcp_57"],
        [], 10,
        ["test5", "amdgpu_xcp_33", "amdgpu_xcp_34", "amdgpu_xcp_35",
         "amdgpu_xcp_36", "amdgpu_xcp_37", "amdgpu_xcp_38", "amdgpu_xcp_39",
         "amdgpu_xcp_58", "amdgpu_xcp_59"],
    ),
]


@pytest.mark.parametrize("desc,available,filtered,required,size,expected",
                         MI300_CASES, ids=[c[0] for c in MI300_CASES])
def test_mi300cpx_oracle(mi300cpx, desc, available, filtered, required,
                         size, expected):
    policy, devices, _ = mi300cpx
    assert len(devices) == 63  # endNodeId truncation drops amdgpu_xcp_63
    if available:
        av = list(available)
    else:
        av = [d.id for d in devices]
    if filtered:
        av = [a for a in av if a not in set(filtered)]
    out = policy.allocate(av, required, size)
    assert len(out) == size
    if expected is not None:
        assert sorted(out) == sorted(expected), f"{desc}: got {sorted(out)}"


# ---- topology 1: MI308, structural cases (no frozen ID sets upstream) ----
# besteffort_policy_test.go:53-75 asserts only success + size

@pytest.mark.parametrize("size", [1, 3, 12])
def test_mi308_oracle_sizes(mi308, size):
    policy, devices, _ = mi308
    out = policy.allocate([d.id for d in devices], [], size)
    assert len(out) == size
    assert len(set(out)) == size


def test_mi308_pair_weight_count(mi308):
    """31 outer entries, as the reference pins (device_test.go:92-109:
    TestPairWeightsCalculation expects len(p2pWeights) == 31)."""
    _, devices, topo = mi308
    weights = compute_pair_weights(devices, topo)
    assert len(weights) == 31


def test_mi308_group_count(mi308):
    """4 parent groups (device_test.go:111-125)."""
    policy, _, _ = mi308
    assert len(policy._groups) == 4


def test_mi300cpx_group_shapes(mi300cpx):
    """8 groups; GPU1-7 have 8 partitions, the truncated GPU8 has 7."""
    policy, _, _ = mi300cpx
    sizes = sorted(len(g.node_ids) for g in policy._groups.values())
    assert sizes == [7, 8, 8, 8, 8, 8, 8, 8]


# ---- the same frozen oracles through the NATIVE server ----

@pytest.fixture(scope="module")
def native_oracle_server(mi300cpx, tmp_path_factory):
    """A bare _fastserver.Server loaded with the mi300-cpx oracle policy
    state — the reference's hardest frozen cases answered by the C++
    search (closed-form fast path on this uniform topology)."""
    from k8s_device_plugin_amd.native import load_fastserver

    mod = load_fastserver()
    if mod is None:
        pytest.skip("fastserver extension unavailable")
    policy, devices, _ = mi300cpx
    sock = str(tmp_path_factory.mktemp("native_oracle") / "s.sock")
    srv = mod.Server(sock)
    srv.set_list_response(b"")
    srv.set_allocator_state(*policy.export_state())
    srv.start()
    yield devices, sock
    srv.stop()


@pytest.mark.parametrize("desc,available,filtered,required,size,expected",
                         MI300_CASES, ids=["native:" + c[0] for c in MI300_CASES])
def test_mi300cpx_oracle_native(native_oracle_server, desc, available,
                                filtered, required, size, expected):
    import grpc

    from k8s_device_plugin_amd.protos import deviceplugin as dp

    devices, sock = native_oracle_server
    if available:
        av = list(available)
    else:
        av = [d.id for d in devices]
    if filtered:
        av = [a for a in av if a not in set(filtered)]
    ch = grpc.insecure_channel(f"unix://{sock}")
    stub = dp.DevicePluginStub(ch)
    req = dp.PreferredAllocationRequest()
    cr = req.container_requests.add()
    cr.available_deviceIDs.extend(av)
    cr.must_include_deviceIDs.extend(required)
    cr.allocation_size = size
    out = list(
        stub.GetPreferredAllocation(req, timeout=30)
        .container_responses[0].deviceIDs
    )
    ch.close()
    assert len(out) == size
    if expected is not None:
        assert sorted(out) == sorted(expected), f"{desc}: got {sorted(out)}"


@pytest.mark.parametrize("desc,available,required,size,expected",
                         MI210_CASES, ids=["native:" + c[0] for c in MI210_CASES])
def test_mi210_oracle_native(mi210, tmp_path, desc, available, required,
                             size, expected):
    """The MI210 whole-GPU oracles through the C++ search as well."""
    import grpc

    from k8s_device_plugin_amd.native import load_fastserver
    from k8s_device_plugin_amd.protos import deviceplugin as dp

    mod = load_fastserver()
    if mod is None:
        pytest.skip("fastserver extension unavailable")
    policy, devices, _ = mi210
    sock = str(tmp_path / "s.sock")
    srv = mod.Server(sock)
    srv.set_list_response(b"")
    srv.set_allocator_state(*policy.export_state())
    srv.start()
    try:
        av = available if available else sorted(
            (d.id for d in devices), key=lambda s: next(
                d.node_id for d in devices if d.id == s)
        )
        ch = grpc.insecure_channel(f"unix://{sock}")
        stub = dp.DevicePluginStub(ch)
        req = dp.PreferredAllocationRequest()
        cr = req.container_requests.add()
        cr.available_deviceIDs.extend(av)
        cr.must_include_deviceIDs.extend(required)
        cr.allocation_size = size
        out = list(
            stub.GetPreferredAllocation(req, timeout=30)
            .container_responses[0].deviceIDs
        )
        ch.close()
        assert sorted(out) == sorted(expected), f"{desc}: got {sorted(out)}"
    finally:
        srv.stop()
