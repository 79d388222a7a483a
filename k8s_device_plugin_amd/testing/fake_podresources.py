"""Fake kubelet PodResources server for tests.

Serves the v1 PodResourcesLister API on a unix socket from an in-memory
pod table, the way the kubelet does from its device-manager checkpoint.
"""

from __future__ import annotations

from concurrent.futures import ThreadPoolExecutor
from typing import Dict, List, Tuple

from ..protos import podresources as pr


class FakePodResources:
    """pods: {(namespace, pod): {container: {resource: [device_ids]}}}"""

    def __init__(self, socket_path: str):
        self.socket_path = socket_path
        self.pods: Dict[Tuple[str, str], Dict[str, Dict[str, List[str]]]] = {}
        self.allocatable: Dict[str, List[str]] = {}
        self._server = None

    # ---- servicer ----

    def _pod_msg(self, key):
        ns, name = key
        pod = pr.PodResources(name=name, namespace=ns)
        for cname, resources in self.pods[key].items():
            c = pod.containers.add()
            c.name = cname
            for resource, ids in resources.items():
                d = c.devices.add()
                d.resource_name = resource
                d.device_ids.extend(ids)
        return pod

    def List(self, request, context):
        resp = pr.ListPodResourcesResponse()
        for key in sorted(self.pods):
            resp.pod_resources.append(self._pod_msg(key))
        return resp

    def GetAllocatableResources(self, request, context):
        resp = pr.AllocatableResourcesResponse()
        for resource, ids in sorted(self.allocatable.items()):
            d = resp.devices.add()
            d.resource_name = resource
            d.device_ids.extend(ids)
        return resp

    def Get(self, request, context):
        key = (request.pod_namespace, request.pod_name)
        resp = pr.GetPodResourcesResponse()
        if key in self.pods:
            resp.pod_resources.CopyFrom(self._pod_msg(key))
        else:
            import grpc

            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"pod {key} not known to this kubelet")
        return resp

    # ---- lifecycle ----

    def start(self) -> "FakePodResources":
        import grpc

        self._server = grpc.server(ThreadPoolExecutor(max_workers=4))
        pr.add_pod_resources_servicer(self._server, self)
        self._server.add_insecure_port(f"unix://{self.socket_path}")
        self._server.start()
        return self

    def stop(self) -> None:
        if self._server is not None:
            self._server.stop(grace=0.5).wait()
