"""Wire-format compatibility tests for the hand-authored descriptors.

Golden bytes are computed by hand from the protobuf wire spec (varint tags,
length-delimited submessages) against the field numbers in the upstream
kubelet api.proto — if these pass, a real kubelet parses our messages.
"""

from k8s_device_plugin_amd.protos import deviceplugin as dp
from k8s_device_plugin_amd.protos import metricssvc as ms


def test_device_golden_bytes():
    d = dp.Device(ID="0000:0c:00.0", health="Healthy")
    d.topology.nodes.add().ID = 1
    expect = (
        b"\x0a\x0c" + b"0000:0c:00.0"      # field 1 string, len 12
        + b"\x12\x07" + b"Healthy"          # field 2 string, len 7
        + b"\x1a\x04\x0a\x02\x08\x01"       # field 3: TopologyInfo{NUMANode{ID:1}}
    )
    assert d.SerializeToString() == expect


def test_register_request_golden_bytes():
    r = dp.RegisterRequest(
        version="v1beta1",
        endpoint="amd.com_gpu",
        resource_name="amd.com/gpu",
    )
    r.options.get_preferred_allocation_available = True
    expect = (
        b"\x0a\x07" + b"v1beta1"
        + b"\x12\x0b" + b"amd.com_gpu"
        + b"\x1a\x0b" + b"amd.com/gpu"
        + b"\x22\x02\x10\x01"               # field 4: options{field2 bool true}
    )
    assert r.SerializeToString() == expect


def test_allocate_response_roundtrip():
    resp = dp.AllocateResponse()
    car = resp.container_responses.add()
    spec = car.devices.add()
    spec.host_path = "/dev/kfd"
    spec.container_path = "/dev/kfd"
    spec.permissions = "rw"
    car.envs["ROCR_VISIBLE_DEVICES"] = "0"
    data = resp.SerializeToString()
    back = dp.AllocateResponse.FromString(data)
    assert back.container_responses[0].devices[0].host_path == "/dev/kfd"
    assert back.container_responses[0].envs["ROCR_VISIBLE_DEVICES"] == "0"


def test_preferred_allocation_fields():
    req = dp.PreferredAllocationRequest()
    cr = req.container_requests.add()
    cr.available_deviceIDs.extend(["a", "b"])
    cr.must_include_deviceIDs.append("a")
    cr.allocation_size = 2
    data = req.SerializeToString()
    # container_requests is field 1; inside: 0a "a", 0a "b", 12 "a", 18 02
    inner = b"\x0a\x01a\x0a\x01b\x12\x01a\x18\x02"
    assert data == b"\x0a" + bytes([len(inner)]) + inner


def test_metricssvc_gpustate_golden_bytes():
    st = ms.GPUState(ID="0", Health="healthy", Device="0000:0c:00.0")
    expect = (
        b"\x0a\x010"                         # ID field 1
        + b"\x1a\x07" + b"healthy"           # Health field 3
        + b"\x2a\x0c" + b"0000:0c:00.0"      # Device field 5
    )
    assert st.SerializeToString() == expect


def test_constants_match_kubelet_abi():
    assert dp.VERSION == "v1beta1"
    assert dp.KUBELET_SOCKET == "/var/lib/kubelet/device-plugins/kubelet.sock"
    assert dp.HEALTHY == "Healthy" and dp.UNHEALTHY == "Unhealthy"
    assert ms.EXPORTER_SOCKET.endswith("amdgpu_device_metrics_exporter_grpc.socket")


def test_podresources_golden_bytes():
    """kubelet PodResources v1 ABI pins (field numbers are the wire
    contract; see protos/podresources.py)."""
    from k8s_device_plugin_amd.protos import podresources as pr

    cd = pr.ContainerDevices(resource_name="amd.com/gpu",
                             device_ids=["x", "y"])
    expect = (
        b"\x0a\x0b" + b"amd.com/gpu"   # resource_name field 1
        + b"\x12\x01x" + b"\x12\x01y"  # device_ids field 2
    )
    assert cd.SerializeToString() == expect

    req = pr.GetPodResourcesRequest(pod_name="p", pod_namespace="ns")
    assert req.SerializeToString() == b"\x0a\x01p\x12\x02ns"

    pod = pr.PodResources(name="p", namespace="ns")
    c = pod.containers.add()
    c.name = "main"
    c.cpu_ids.extend([3, 4])
    inner = b"\x0a\x04main" + b"\x1a\x02\x03\x04"  # name f1, cpu_ids f3 packed
    assert pod.SerializeToString() == (
        b"\x0a\x01p\x12\x02ns" + b"\x1a" + bytes([len(inner)]) + inner
    )

    assert pr.POD_RESOURCES_SERVICE == "v1.PodResourcesLister"
    assert pr.PODRESOURCES_SOCKET == "/var/lib/kubelet/pod-resources/kubelet.sock"
