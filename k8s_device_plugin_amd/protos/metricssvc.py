"""amd-metrics-exporter MetricsService: messages + gRPC stubs.

Wire-identical to the exporter's schema (reference:
internal/pkg/exporter/metricssvc/metricssvc.pb.go:95-110,179,227,284 and
metricssvc_grpc.pb.go:45-56,170).  List takes google.protobuf.Empty.
"""

from __future__ import annotations

from google.protobuf import empty_pb2

from ._build import FileBuilder, make_pool_with_wellknown

# GPUHealth enum string values (reference: metricssvc.pb.go:46-63); the
# exporter populates GPUState.Health with the lowercased form.
HEALTH_UNKNOWN = "UNKNOWN"
HEALTH_HEALTHY = "HEALTHY"
HEALTH_UNHEALTHY = "UNHEALTHY"

# Socket the exporter serves on (reference: internal/pkg/exporter/health.go:36)
EXPORTER_SOCKET = (
    "/var/lib/amd-metrics-exporter/amdgpu_device_metrics_exporter_grpc.socket"
)
QUERY_TIMEOUT_S = 5.0

_pool = make_pool_with_wellknown()
_f = FileBuilder(
    "amdgpu/metricssvc/metricssvc.proto",
    "metricssvc",
    dependencies=["google/protobuf/empty.proto"],
)
_f.message("GPUState", [
    ("ID", 1, "string"),
    ("UUID", 2, "string"),
    ("Health", 3, "string"),
    ("AssociatedWorkload", 4, "string", "repeated"),
    ("Device", 5, "string"),
])
_f.message("GPUGetRequest", [
    ("ID", 1, "string", "repeated"),
])
_f.message("GPUUpdateRequest", [
    ("ID", 1, "string", "repeated"),
    ("Health", 2, "string", "repeated"),
])
_f.message("GPUStateResponse", [
    ("GPUState", 1, "msg:metricssvc.GPUState", "repeated"),
])
_classes = _f.build(_pool)

GPUState = _classes["GPUState"]
GPUGetRequest = _classes["GPUGetRequest"]
GPUUpdateRequest = _classes["GPUUpdateRequest"]
GPUStateResponse = _classes["GPUStateResponse"]
Empty = empty_pb2.Empty

METRICS_SERVICE = "metricssvc.MetricsService"


class MetricsServiceStub:
    def __init__(self, channel):
        p = f"/{METRICS_SERVICE}"
        self.GetGPUState = channel.unary_unary(
            f"{p}/GetGPUState",
            request_serializer=GPUGetRequest.SerializeToString,
            response_deserializer=GPUStateResponse.FromString,
        )
        self.List = channel.unary_unary(
            f"{p}/List",
            request_serializer=Empty.SerializeToString,
            response_deserializer=GPUStateResponse.FromString,
        )


def add_metrics_servicer(server, servicer) -> None:
    """servicer implements GetGPUState / List (the fake exporter in tests)."""
    import grpc

    handlers = {
        "GetGPUState": grpc.unary_unary_rpc_method_handler(
            servicer.GetGPUState,
            request_deserializer=GPUGetRequest.FromString,
            response_serializer=GPUStateResponse.SerializeToString,
        ),
        "List": grpc.unary_unary_rpc_method_handler(
            servicer.List,
            request_deserializer=Empty.FromString,
            response_serializer=GPUStateResponse.SerializeToString,
        ),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(METRICS_SERVICE, handlers),)
    )
