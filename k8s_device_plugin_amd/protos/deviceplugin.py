"""Kubelet DevicePlugin v1beta1 API: messages, constants, gRPC stubs.

Wire-identical to the kubelet ABI.  Message/field numbers transcribed from
the upstream schema the reference vendors (reference:
vendor/k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1/api.proto; constants:
.../v1beta1/constants.go:19-45).  This module is the single place the rest of
the package imports the API from.
"""

from __future__ import annotations

from google.protobuf import descriptor_pool

from ._build import FileBuilder

# --- constants (reference: v1beta1/constants.go) ---
HEALTHY = "Healthy"
UNHEALTHY = "Unhealthy"
VERSION = "v1beta1"
DEVICE_PLUGIN_PATH = "/var/lib/kubelet/device-plugins/"
KUBELET_SOCKET = DEVICE_PLUGIN_PATH + "kubelet.sock"
RESOURCE_NAMESPACE = "amd.com"

_pool = descriptor_pool.DescriptorPool()
_f = FileBuilder("k8s.io/kubelet/deviceplugin/v1beta1/api.proto", "v1beta1")

_f.message("DevicePluginOptions", [
    ("pre_start_required", 1, "bool"),
    ("get_preferred_allocation_available", 2, "bool"),
])
_f.message("RegisterRequest", [
    ("version", 1, "string"),
    ("endpoint", 2, "string"),
    ("resource_name", 3, "string"),
    ("options", 4, "msg:v1beta1.DevicePluginOptions"),
])
_f.message("Empty", [])
_f.message("ListAndWatchResponse", [
    ("devices", 1, "msg:v1beta1.Device", "repeated"),
])
_f.message("TopologyInfo", [
    ("nodes", 1, "msg:v1beta1.NUMANode", "repeated"),
])
_f.message("NUMANode", [
    ("ID", 1, "int64"),
])
_f.message("Device", [
    ("ID", 1, "string"),
    ("health", 2, "string"),
    ("topology", 3, "msg:v1beta1.TopologyInfo"),
])
_f.message("PreStartContainerRequest", [
    ("devices_ids", 1, "string", "repeated"),
])
_f.message("PreStartContainerResponse", [])
_f.message("PreferredAllocationRequest", [
    ("container_requests", 1, "msg:v1beta1.ContainerPreferredAllocationRequest", "repeated"),
])
_f.message("ContainerPreferredAllocationRequest", [
    ("available_deviceIDs", 1, "string", "repeated"),
    ("must_include_deviceIDs", 2, "string", "repeated"),
    ("allocation_size", 3, "int32"),
])
_f.message("PreferredAllocationResponse", [
    ("container_responses", 1, "msg:v1beta1.ContainerPreferredAllocationResponse", "repeated"),
])
_f.message("ContainerPreferredAllocationResponse", [
    ("deviceIDs", 1, "string", "repeated"),
])
_f.message("AllocateRequest", [
    ("container_requests", 1, "msg:v1beta1.ContainerAllocateRequest", "repeated"),
])
_f.message("ContainerAllocateRequest", [
    ("devices_ids", 1, "string", "repeated"),
])
_f.message("CDIDevice", [
    ("name", 1, "string"),
])
_f.message("AllocateResponse", [
    ("container_responses", 1, "msg:v1beta1.ContainerAllocateResponse", "repeated"),
])
_f.message("ContainerAllocateResponse", [
    ("envs", 1, "map<string,string>"),
    ("mounts", 2, "msg:v1beta1.Mount", "repeated"),
    ("devices", 3, "msg:v1beta1.DeviceSpec", "repeated"),
    ("annotations", 4, "map<string,string>"),
    ("cdi_devices", 5, "msg:v1beta1.CDIDevice", "repeated"),
])
_f.message("Mount", [
    ("container_path", 1, "string"),
    ("host_path", 2, "string"),
    ("read_only", 3, "bool"),
])
_f.message("DeviceSpec", [
    ("container_path", 1, "string"),
    ("host_path", 2, "string"),
    ("permissions", 3, "string"),
])

_classes = _f.build(_pool)

DevicePluginOptions = _classes["DevicePluginOptions"]
RegisterRequest = _classes["RegisterRequest"]
Empty = _classes["Empty"]
ListAndWatchResponse = _classes["ListAndWatchResponse"]
TopologyInfo = _classes["TopologyInfo"]
NUMANode = _classes["NUMANode"]
Device = _classes["Device"]
PreStartContainerRequest = _classes["PreStartContainerRequest"]
PreStartContainerResponse = _classes["PreStartContainerResponse"]
PreferredAllocationRequest = _classes["PreferredAllocationRequest"]
ContainerPreferredAllocationRequest = _classes["ContainerPreferredAllocationRequest"]
PreferredAllocationResponse = _classes["PreferredAllocationResponse"]
ContainerPreferredAllocationResponse = _classes["ContainerPreferredAllocationResponse"]
AllocateRequest = _classes["AllocateRequest"]
ContainerAllocateRequest = _classes["ContainerAllocateRequest"]
CDIDevice = _classes["CDIDevice"]
AllocateResponse = _classes["AllocateResponse"]
ContainerAllocateResponse = _classes["ContainerAllocateResponse"]
Mount = _classes["Mount"]
DeviceSpec = _classes["DeviceSpec"]

# --- gRPC plumbing (method paths are part of the kubelet ABI) ---

REGISTRATION_SERVICE = "v1beta1.Registration"
DEVICE_PLUGIN_SERVICE = "v1beta1.DevicePlugin"


def add_device_plugin_servicer(server, servicer) -> None:
    """servicer implements GetDevicePluginOptions / ListAndWatch /
    GetPreferredAllocation / Allocate / PreStartContainer."""
    import grpc

    handlers = {
        "GetDevicePluginOptions": grpc.unary_unary_rpc_method_handler(
            servicer.GetDevicePluginOptions,
            request_deserializer=Empty.FromString,
            response_serializer=DevicePluginOptions.SerializeToString,
        ),
        "ListAndWatch": grpc.unary_stream_rpc_method_handler(
            servicer.ListAndWatch,
            request_deserializer=Empty.FromString,
            response_serializer=ListAndWatchResponse.SerializeToString,
        ),
        "GetPreferredAllocation": grpc.unary_unary_rpc_method_handler(
            servicer.GetPreferredAllocation,
            request_deserializer=PreferredAllocationRequest.FromString,
            response_serializer=PreferredAllocationResponse.SerializeToString,
        ),
        "Allocate": grpc.unary_unary_rpc_method_handler(
            servicer.Allocate,
            request_deserializer=AllocateRequest.FromString,
            response_serializer=AllocateResponse.SerializeToString,
        ),
        "PreStartContainer": grpc.unary_unary_rpc_method_handler(
            servicer.PreStartContainer,
            request_deserializer=PreStartContainerRequest.FromString,
            response_serializer=PreStartContainerResponse.SerializeToString,
        ),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(DEVICE_PLUGIN_SERVICE, handlers),)
    )


def add_registration_servicer(server, servicer) -> None:
    """servicer implements Register (the kubelet side; used by the stub
    kubelet in tests/bench)."""
    import grpc

    handlers = {
        "Register": grpc.unary_unary_rpc_method_handler(
            servicer.Register,
            request_deserializer=RegisterRequest.FromString,
            response_serializer=Empty.SerializeToString,
        ),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(REGISTRATION_SERVICE, handlers),)
    )


class RegistrationStub:
    """Client for the kubelet Registration service."""

    def __init__(self, channel):
        self.Register = channel.unary_unary(
            f"/{REGISTRATION_SERVICE}/Register",
            request_serializer=RegisterRequest.SerializeToString,
            response_deserializer=Empty.FromString,
        )


class DevicePluginStub:
    """Client for a DevicePlugin server (used by the stub kubelet)."""

    def __init__(self, channel):
        p = f"/{DEVICE_PLUGIN_SERVICE}"
        self.GetDevicePluginOptions = channel.unary_unary(
            f"{p}/GetDevicePluginOptions",
            request_serializer=Empty.SerializeToString,
            response_deserializer=DevicePluginOptions.FromString,
        )
        self.ListAndWatch = channel.unary_stream(
            f"{p}/ListAndWatch",
            request_serializer=Empty.SerializeToString,
            response_deserializer=ListAndWatchResponse.FromString,
        )
        self.GetPreferredAllocation = channel.unary_unary(
            f"{p}/GetPreferredAllocation",
            request_serializer=PreferredAllocationRequest.SerializeToString,
            response_deserializer=PreferredAllocationResponse.FromString,
        )
        self.Allocate = channel.unary_unary(
            f"{p}/Allocate",
            request_serializer=AllocateRequest.SerializeToString,
            response_deserializer=AllocateResponse.FromString,
        )
        self.PreStartContainer = channel.unary_unary(
            f"{p}/PreStartContainer",
            request_serializer=PreStartContainerRequest.SerializeToString,
            response_deserializer=PreStartContainerResponse.FromString,
        )
