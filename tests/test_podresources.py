"""kubelet PodResources v1 client (beyond-reference observability)."""

import json

import grpc
import pytest

from k8s_device_plugin_amd.plugin.podresources import (
    allocatable_gpu_devices,
    gpu_allocation_summary,
    list_gpu_allocations,
)
from k8s_device_plugin_amd.protos import podresources as pr
from k8s_device_plugin_amd.testing.fake_podresources import FakePodResources


@pytest.fixture
def kubelet(tmp_path):
    srv = FakePodResources(str(tmp_path / "podres.sock"))
    srv.pods[("ml", "trainer-0")] = {
        "main": {"amd.com/gpu": ["0000:01:00.0", "0000:02:00.0"],
                 "cpu": []},
    }
    srv.pods[("ml", "trainer-1")] = {
        "main": {"amd.com/gpu": ["0000:03:00.0"]},
        "sidecar": {},
    }
    srv.pods[("infra", "dns")] = {"coredns": {}}
    srv.allocatable = {
        "amd.com/gpu": [f"0000:0{i}:00.0" for i in range(1, 9)],
        "example.com/other": ["x"],
    }
    srv.start()
    yield srv
    srv.stop()


def test_list_gpu_allocations(kubelet):
    allocs = list_gpu_allocations(kubelet.socket_path)
    assert len(allocs) == 2
    by_pod = {(a.namespace, a.pod): a for a in allocs}
    a0 = by_pod[("ml", "trainer-0")]
    assert a0.container == "main"
    assert a0.resource == "amd.com/gpu"
    assert a0.device_ids == ["0000:01:00.0", "0000:02:00.0"]
    assert by_pod[("ml", "trainer-1")].device_ids == ["0000:03:00.0"]


def test_allocatable_filters_amd(kubelet):
    alloc = allocatable_gpu_devices(kubelet.socket_path)
    assert list(alloc) == ["amd.com/gpu"]
    assert len(alloc["amd.com/gpu"]) == 8


def test_get_single_pod(kubelet):
    ch = grpc.insecure_channel(f"unix://{kubelet.socket_path}")
    stub = pr.PodResourcesListerStub(ch)
    resp = stub.Get(
        pr.GetPodResourcesRequest(pod_name="trainer-0", pod_namespace="ml"),
        timeout=5,
    )
    assert resp.pod_resources.name == "trainer-0"
    with pytest.raises(grpc.RpcError) as exc:
        stub.Get(pr.GetPodResourcesRequest(pod_name="nope",
                                           pod_namespace="ml"), timeout=5)
    assert exc.value.code() == grpc.StatusCode.NOT_FOUND
    ch.close()


def test_summary_with_advertised_delta(kubelet):
    s = gpu_allocation_summary(
        kubelet.socket_path,
        advertised={"amd.com/gpu": [f"0000:0{i}:00.0" for i in range(1, 8)]},
    )
    assert s["allocated_device_ids"] == [
        "0000:01:00.0", "0000:02:00.0", "0000:03:00.0"
    ]
    # kubelet still offers 08 which we no longer advertise
    assert s["kubelet_only"] == ["0000:08:00.0"]
    assert s["plugin_only"] == []


def test_cli_dump_podresources(kubelet, capsys):
    from k8s_device_plugin_amd.cli import device_plugin_main

    rc = device_plugin_main([
        "--dump-podresources", "--podresources-socket", kubelet.socket_path,
    ])
    assert rc == 0
    out = json.loads(capsys.readouterr().out)
    assert len(out["allocations"]) == 2
    assert out["allocatable"]["amd.com/gpu"]


def test_cli_dump_podresources_no_kubelet(tmp_path, capsys):
    from k8s_device_plugin_amd.cli import device_plugin_main

    rc = device_plugin_main([
        "--dump-podresources",
        "--podresources-socket", str(tmp_path / "absent.sock"),
    ])
    assert rc == 1


def test_wire_unknown_fields_tolerated(kubelet):
    """A DRA-aware kubelet appends dynamic_resources (field 5) to
    ContainerResources; our transcription omits it, and parsing must
    skip it."""
    raw = pr.ContainerResources(name="c").SerializeToString()
    # field 5 (dynamic_resources), wire type 2, 3 payload bytes
    raw += b"\x2a\x03abc"
    msg = pr.ContainerResources.FromString(raw)
    assert msg.name == "c"
