"""kubelet PodResources API client — pod → device correlation.

Reference: pkg/hypervisor/backend/kubernetes/pod_resources_proxy.go
(a gRPC proxy over kubelet's PodResources socket so the hypervisor can
see which pods kubelet gave which devices — the authoritative mapping
when a foreign device plugin is also in play).

The v1 podresources protobuf (k8s.io/kubelet/pkg/apis/podresources/v1)
is built at runtime like dp_proto.py; List() returns every pod's
per-container device assignments, which the hypervisor joins against
its own worker set and the checkpoint detector's foreign-device view.
"""
from __future__ import annotations

from typing import Dict, List

import grpc
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_F = descriptor_pb2.FieldDescriptorProto

PKG = "v1"
SERVICE = "/v1.PodResourcesLister"
M_LIST = f"{SERVICE}/List"
M_ALLOCATABLE = f"{SERVICE}/GetAllocatableResources"

KUBELET_PODRESOURCES_SOCKET = "/var/lib/kubelet/pod-resources/kubelet.sock"


def _field(name, num, ftype, label=_F.LABEL_OPTIONAL, type_name=""):
    f = _F(name=name, number=num, type=ftype, label=label)
    if type_name:
        f.type_name = f".{PKG}.{type_name}"
    return f


def _msg(name, *fields):
    m = descriptor_pb2.DescriptorProto(name=name)
    m.field.extend(fields)
    return m


def _build():
    fd = descriptor_pb2.FileDescriptorProto(
        name="tensorfusion/podresources_v1.proto", package=PKG,
        syntax="proto3")
    fd.message_type.append(_msg("ListPodResourcesRequest"))
    fd.message_type.append(_msg("AllocatableResourcesRequest"))
    fd.message_type.append(_msg(
        "NUMANode", _field("ID", 1, _F.TYPE_INT64)))
    fd.message_type.append(_msg(
        "TopologyInfo",
        _field("nodes", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
               type_name="NUMANode")))
    fd.message_type.append(_msg(
        "ContainerDevices",
        _field("resource_name", 1, _F.TYPE_STRING),
        _field("device_ids", 2, _F.TYPE_STRING, _F.LABEL_REPEATED),
        _field("topology", 3, _F.TYPE_MESSAGE, type_name="TopologyInfo")))
    fd.message_type.append(_msg(
        "ContainerResources",
        _field("name", 1, _F.TYPE_STRING),
        _field("devices", 2, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
               type_name="ContainerDevices")))
    fd.message_type.append(_msg(
        "PodResources",
        _field("name", 1, _F.TYPE_STRING),
        _field("namespace", 2, _F.TYPE_STRING),
        _field("containers", 3, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
               type_name="ContainerResources")))
    fd.message_type.append(_msg(
        "ListPodResourcesResponse",
        _field("pod_resources", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
               type_name="PodResources")))
    fd.message_type.append(_msg(
        "AllocatableResourcesResponse",
        _field("devices", 2, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
               type_name="ContainerDevices")))
    pool = descriptor_pool.DescriptorPool()
    pool.Add(fd)
    return {m.name: message_factory.GetMessageClass(
        pool.FindMessageTypeByName(f"{PKG}.{m.name}"))
        for m in fd.message_type}


MSG = _build()


class PodResourcesClient:
    """Typed client over kubelet's pod-resources unix socket."""

    def __init__(self, socket_path: str = KUBELET_PODRESOURCES_SOCKET):
        self.socket_path = socket_path

    def _channel(self):
        return grpc.insecure_channel(f"unix://{self.socket_path}")

    def list(self, timeout: float = 5.0) -> List[dict]:
        """[{namespace, pod, containers: [{name, devices:
        {resource_name: [ids]}}]}]"""

        ch = self._channel()
        try:
            stub = ch.unary_unary(
                M_LIST,
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=MSG[
                    "ListPodResourcesResponse"].FromString)
            resp = stub(MSG["ListPodResourcesRequest"](), timeout=timeout)
        finally:
            ch.close()
        out = []
        for pr in resp.pod_resources:
            containers = []
            for c in pr.containers:
                devs: Dict[str, List[str]] = {}
                for d in c.devices:
                    devs.setdefault(d.resource_name, []).extend(
                        d.device_ids)
                containers.append({"name": c.name, "devices": devs})
            out.append({"namespace": pr.namespace, "pod": pr.name,
                        "containers": containers})
        return out

    def device_map(self, resource_prefix: str = "tensor-fusion.ai/",
                   timeout: float = 5.0) -> Dict[str, List[str]]:
        """pod key → device ids for resources under `resource_prefix`
        (the hypervisor's pod→index correlation, reference
        pod_resources_proxy.go)."""

        out: Dict[str, List[str]] = {}
        for pr in self.list(timeout=timeout):
            ids: List[str] = []
            for c in pr["containers"]:
                for res, dids in c["devices"].items():
                    if res.startswith(resource_prefix):
                        ids.extend(dids)
            if ids:
                out[f"{pr['namespace']}/{pr['pod']}"] = ids
        return out

    def foreign_gpu_devices(
            self, prefixes=("amd.com/gpu", "nvidia.com/gpu"),
            timeout: float = 5.0) -> Dict[str, List[str]]:
        """pod key → GPU ids held by FOREIGN device plugins — the live
        complement to the checkpoint-file detector."""

        out: Dict[str, List[str]] = {}
        for pr in self.list(timeout=timeout):
            ids: List[str] = []
            for c in pr["containers"]:
                for res, dids in c["devices"].items():
                    if any(res.startswith(p) for p in prefixes):
                        ids.extend(dids)
            if ids:
                out[f"{pr['namespace']}/{pr['pod']}"] = ids
        return out
