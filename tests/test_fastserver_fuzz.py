"""Differential fuzz: the native C++ preferred-allocation search must agree
with the Python policy on random topologies and random requests."""

import os
import random

import grpc
import hypothesis.strategies as st
from hypothesis import given, settings

# long local runs: AMDXDP_FUZZ_EXAMPLES=300 pytest tests/test_fastserver_fuzz.py
_EXAMPLES = int(os.environ.get("AMDXDP_FUZZ_EXAMPLES", "12"))

from k8s_device_plugin_amd.plugin import AMDGPUPlugin
from k8s_device_plugin_amd.plugin.native_server import NativePluginServer
from k8s_device_plugin_amd.protos import deviceplugin as dp


@given(
    n_gpus=st.integers(min_value=2, max_value=8),
    parts=st.sampled_from([1, 2, 4]),
    seed=st.integers(min_value=0, max_value=2**31),
)
@settings(max_examples=_EXAMPLES, deadline=None)
def test_native_matches_python_random(tmp_path_factory, n_gpus, parts, seed):
    from k8s_device_plugin_amd.testing.fakesysfs import build_mi355x_node

    root = tmp_path_factory.mktemp("fz")
    fs = build_mi355x_node(
        str(root), n_gpus=n_gpus, partitions_per_gpu=parts,
        compute_partition="CPX" if parts > 1 else "SPX",
    )
    plugin = AMDGPUPlugin(resource="gpu", paths=fs.paths)
    plugin.start()
    srv = NativePluginServer(plugin, str(root / "s.sock"))
    srv.start()
    try:
        ch = grpc.insecure_channel(f"unix://{root}/s.sock")
        stub = dp.DevicePluginStub(ch)
        rng = random.Random(seed)
        ids = sorted(plugin.devices)
        for _ in range(5):
            available = rng.sample(ids, rng.randint(2, len(ids)))
            size = rng.randint(1, len(available))
            required = rng.sample(available, rng.randint(0, min(2, size)))

            req = dp.PreferredAllocationRequest()
            cr = req.container_requests.add()
            cr.available_deviceIDs.extend(available)
            cr.must_include_deviceIDs.extend(required)
            cr.allocation_size = size
            native_out = list(
                stub.GetPreferredAllocation(req, timeout=30)
                .container_responses[0].deviceIDs
            )
            py_out = plugin.allocator.allocate(available, required, size)
            assert native_out == py_out, (available, required, size)
        ch.close()
    finally:
        srv.stop()
