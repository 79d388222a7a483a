"""kubelet PodResources client: who actually holds our GPUs.

Beyond-reference observability: the plugin (or an
operator via `amd-device-plugin --dump-podresources`) queries the
kubelet's pod-resources socket and reduces the answer to the AMD view —
per-pod amd.com/* device assignments and the allocatable set — closing
the advertised-vs-allocated loop from the node side.  Flag-gated and
optional like the other extras; the kubelet must run with the
PodResources API (GA for List since v1.28).
"""

from __future__ import annotations

import logging
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..protos import podresources as pr

log = logging.getLogger(__name__)

AMD_RESOURCE_PREFIX = "amd.com/"
QUERY_TIMEOUT_S = 5.0


@dataclass
class GPUAllocation:
    namespace: str
    pod: str
    container: str
    resource: str  # e.g. "amd.com/gpu"
    device_ids: List[str] = field(default_factory=list)


def _channel(socket_path: str):
    import grpc

    return grpc.insecure_channel(f"unix://{socket_path}")


def list_gpu_allocations(
    socket_path: str = pr.PODRESOURCES_SOCKET,
    timeout: float = QUERY_TIMEOUT_S,
) -> List[GPUAllocation]:
    """Every amd.com/* device assignment on the node, per container."""
    out: List[GPUAllocation] = []
    with _channel(socket_path) as ch:
        resp = pr.PodResourcesListerStub(ch).List(
            pr.ListPodResourcesRequest(), timeout=timeout
        )
    for pod in resp.pod_resources:
        for c in pod.containers:
            for devs in c.devices:
                if not devs.resource_name.startswith(AMD_RESOURCE_PREFIX):
                    continue
                out.append(
                    GPUAllocation(
                        namespace=pod.namespace,
                        pod=pod.name,
                        container=c.name,
                        resource=devs.resource_name,
                        device_ids=list(devs.device_ids),
                    )
                )
    return out


def allocatable_gpu_devices(
    socket_path: str = pr.PODRESOURCES_SOCKET,
    timeout: float = QUERY_TIMEOUT_S,
) -> Dict[str, List[str]]:
    """{resource_name: [device_ids]} the kubelet considers allocatable."""
    with _channel(socket_path) as ch:
        resp = pr.PodResourcesListerStub(ch).GetAllocatableResources(
            pr.AllocatableResourcesRequest(), timeout=timeout
        )
    out: Dict[str, List[str]] = {}
    for devs in resp.devices:
        if devs.resource_name.startswith(AMD_RESOURCE_PREFIX):
            out.setdefault(devs.resource_name, []).extend(devs.device_ids)
    return out


def gpu_allocation_summary(
    socket_path: str = pr.PODRESOURCES_SOCKET,
    advertised: Optional[Dict[str, List[str]]] = None,
    timeout: float = QUERY_TIMEOUT_S,
) -> dict:
    """One JSON-able view: allocations, allocatable, and (when the
    caller passes what it advertises) the delta — devices the kubelet
    considers allocatable that we no longer advertise, and vice versa."""
    allocations = list_gpu_allocations(socket_path, timeout)
    try:
        allocatable = allocatable_gpu_devices(socket_path, timeout)
    except Exception as e:  # older kubelet: List GA'd before GetAllocatable
        log.warning("GetAllocatableResources unavailable: %s", e)
        allocatable = {}
    summary = {
        "allocations": [a.__dict__ for a in allocations],
        "allocatable": allocatable,
        "allocated_device_ids": sorted(
            {d for a in allocations for d in a.device_ids}
        ),
    }
    if advertised is not None:
        adv = {d for ids in advertised.values() for d in ids}
        alloc = {d for ids in allocatable.values() for d in ids}
        summary["kubelet_only"] = sorted(alloc - adv)
        summary["plugin_only"] = sorted(adv - alloc)
    return summary
