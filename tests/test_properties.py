"""Property-based robustness tests (hypothesis): the parser never raises on
arbitrary input; the allocator always returns a valid subset."""

import hypothesis.strategies as st
from hypothesis import given, settings

from k8s_device_plugin_amd.topology import parse_properties_text
from k8s_device_plugin_amd.topology.firmware import parse_debugfs_firmware_text


@given(st.text(max_size=2000))
@settings(max_examples=200, deadline=None)
def test_properties_parser_total(text):
    props = parse_properties_text(text)
    assert all(isinstance(v, int) for v in props.values())


@given(st.text(max_size=2000))
@settings(max_examples=100, deadline=None)
def test_debugfs_parser_total(text):
    feat, fw = parse_debugfs_firmware_text(text)
    assert set(feat) == set(fw)


@given(
    n_gpus=st.integers(min_value=2, max_value=8),
    parts=st.sampled_from([1, 2, 4, 8]),
    seed=st.integers(min_value=0, max_value=2**31),
)
@settings(max_examples=15, deadline=None)
def test_allocator_invariants(tmp_path_factory, n_gpus, parts, seed):
    import random

    from k8s_device_plugin_amd.allocator import BestEffortPolicy
    from k8s_device_plugin_amd.testing.fakesysfs import build_mi355x_node
    from k8s_device_plugin_amd.topology import KFDTopology, discover_gpus

    root = tmp_path_factory.mktemp("ht")
    fs = build_mi355x_node(
        str(root), n_gpus=n_gpus, partitions_per_gpu=parts,
        compute_partition="CPX" if parts > 1 else "SPX",
    )
    topo = KFDTopology.load(fs.paths)
    devices = discover_gpus(fs.paths, topology=topo)
    policy = BestEffortPolicy()
    policy.init(devices.values(), topology=topo)

    rng = random.Random(seed)
    ids = sorted(devices)
    available = rng.sample(ids, rng.randint(2, len(ids)))
    size = rng.randint(1, len(available))
    n_req = rng.randint(0, min(size, len(available)))
    required = rng.sample(available, n_req)

    out = policy.allocate(available, required, size)
    assert len(out) == size
    assert len(set(out)) == size
    assert set(out).issubset(set(available))
    assert set(required).issubset(set(out))
    # determinism
    assert policy.allocate(available, required, size) == out
