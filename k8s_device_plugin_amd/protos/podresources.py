"""kubelet PodResources v1 API: messages + gRPC stubs.

Transcription of the stable k8s.io/kubelet/pkg/apis/podresources/v1 schema
(field numbers are the kubelet ABI; the reference does not use this API —
it is a beyond-reference observability feature).  The
device plugin acts as a CLIENT on the kubelet's pod-resources socket to
observe which pods actually hold amd.com/* devices — closing the
advertised-vs-allocated loop from the node side.

DynamicResource (DRA) fields are intentionally omitted from the message
set; protobuf parsing skips unknown fields, so responses from a DRA-aware
kubelet still decode.
"""

from __future__ import annotations

from ._build import FileBuilder, make_pool_with_wellknown

# The kubelet serves this socket when the PodResources feature is on
# (GA since 1.28 for List/GetAllocatableResources).
PODRESOURCES_SOCKET = "/var/lib/kubelet/pod-resources/kubelet.sock"

_pool = make_pool_with_wellknown()
_f = FileBuilder("kubelet/podresources/v1/api.proto", "v1")
_f.message("AllocatableResourcesRequest", [])
_f.message("AllocatableResourcesResponse", [
    ("devices", 1, "msg:v1.ContainerDevices", "repeated"),
    ("cpu_ids", 2, "int64", "repeated"),
    ("memory", 3, "msg:v1.ContainerMemory", "repeated"),
])
_f.message("ListPodResourcesRequest", [])
_f.message("ListPodResourcesResponse", [
    ("pod_resources", 1, "msg:v1.PodResources", "repeated"),
])
_f.message("PodResources", [
    ("name", 1, "string"),
    ("namespace", 2, "string"),
    ("containers", 3, "msg:v1.ContainerResources", "repeated"),
])
_f.message("ContainerResources", [
    ("name", 1, "string"),
    ("devices", 2, "msg:v1.ContainerDevices", "repeated"),
    ("cpu_ids", 3, "int64", "repeated"),
    ("memory", 4, "msg:v1.ContainerMemory", "repeated"),
])
_f.message("ContainerMemory", [
    ("memory_type", 1, "string"),
    ("size", 2, "uint64"),
    ("topology", 3, "msg:v1.TopologyInfo"),
])
_f.message("ContainerDevices", [
    ("resource_name", 1, "string"),
    ("device_ids", 2, "string", "repeated"),
    ("topology", 3, "msg:v1.TopologyInfo"),
])
_f.message("TopologyInfo", [
    ("nodes", 1, "msg:v1.NUMANode", "repeated"),
])
_f.message("NUMANode", [
    ("ID", 1, "int64"),
])
_f.message("GetPodResourcesRequest", [
    ("pod_name", 1, "string"),
    ("pod_namespace", 2, "string"),
])
_f.message("GetPodResourcesResponse", [
    ("pod_resources", 1, "msg:v1.PodResources"),
])
_classes = _f.build(_pool)

AllocatableResourcesRequest = _classes["AllocatableResourcesRequest"]
AllocatableResourcesResponse = _classes["AllocatableResourcesResponse"]
ListPodResourcesRequest = _classes["ListPodResourcesRequest"]
ListPodResourcesResponse = _classes["ListPodResourcesResponse"]
PodResources = _classes["PodResources"]
ContainerResources = _classes["ContainerResources"]
ContainerMemory = _classes["ContainerMemory"]
ContainerDevices = _classes["ContainerDevices"]
TopologyInfo = _classes["TopologyInfo"]
NUMANode = _classes["NUMANode"]
GetPodResourcesRequest = _classes["GetPodResourcesRequest"]
GetPodResourcesResponse = _classes["GetPodResourcesResponse"]

POD_RESOURCES_SERVICE = "v1.PodResourcesLister"


class PodResourcesListerStub:
    def __init__(self, channel):
        self.List = channel.unary_unary(
            f"/{POD_RESOURCES_SERVICE}/List",
            request_serializer=ListPodResourcesRequest.SerializeToString,
            response_deserializer=ListPodResourcesResponse.FromString,
        )
        self.GetAllocatableResources = channel.unary_unary(
            f"/{POD_RESOURCES_SERVICE}/GetAllocatableResources",
            request_serializer=AllocatableResourcesRequest.SerializeToString,
            response_deserializer=AllocatableResourcesResponse.FromString,
        )
        self.Get = channel.unary_unary(
            f"/{POD_RESOURCES_SERVICE}/Get",
            request_serializer=GetPodResourcesRequest.SerializeToString,
            response_deserializer=GetPodResourcesResponse.FromString,
        )


def add_pod_resources_servicer(server, servicer) -> None:
    """servicer implements List / GetAllocatableResources / Get (the
    kubelet side; used by the fake kubelet in tests)."""
    import grpc

    handlers = {
        "List": grpc.unary_unary_rpc_method_handler(
            servicer.List,
            request_deserializer=ListPodResourcesRequest.FromString,
            response_serializer=ListPodResourcesResponse.SerializeToString,
        ),
        "GetAllocatableResources": grpc.unary_unary_rpc_method_handler(
            servicer.GetAllocatableResources,
            request_deserializer=AllocatableResourcesRequest.FromString,
            response_serializer=AllocatableResourcesResponse.SerializeToString,
        ),
        "Get": grpc.unary_unary_rpc_method_handler(
            servicer.Get,
            request_deserializer=GetPodResourcesRequest.FromString,
            response_serializer=GetPodResourcesResponse.SerializeToString,
        ),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(POD_RESOURCES_SERVICE, handlers),)
    )
