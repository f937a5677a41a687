"""Dataclass model ⇄ Kubernetes wire JSON.

The internal model (api/types.py) maps 1:1 onto CRDs in group
`tensor-fusion.ai/v1` (reference api/v1/*.go + config/crd/bases/). This
module is the single conversion point: field names are camelCased on the
wire (k8s convention), data-bearing dict keys (labels, annotations, env,
topology uuid→tier) pass through untouched because conversion walks the
dataclass *structure*, never raw dicts.

Core objects (Pod, Node) convert to/from corev1 shapes.
"""
from __future__ import annotations

import dataclasses
import time
import typing
from typing import Any, Dict, Optional, Type

from ..api import types as T

GROUP = "tensor-fusion.ai"
VERSION = "v1"
API_VERSION = f"{GROUP}/{VERSION}"

# CRD kinds (everything in ALL_KINDS except core Pod/Node).
CRD_KINDS = [k for k in T.ALL_KINDS if k not in ("Pod", "Node")]

# kinds that are cluster-scoped (no namespace), mirroring the reference
# (GPU, GPUNode, TensorFusionCluster, GPUPool etc. are cluster-scoped;
# workload-facing kinds are namespaced — api/v1/*_types.go +kubebuilder
# scope markers).
CLUSTER_SCOPED = {
    "GPU", "GPUNode", "GPUPool", "TensorFusionCluster", "GPUNodeClass",
    "GPUNodeClaim", "SchedulingConfigTemplate", "ProviderConfig",
}

PLURALS = {
    "TensorFusionCluster": "tensorfusionclusters",
    "GPUPool": "gpupools",
    "GPUNode": "gpunodes",
    "GPU": "gpus",
    "TensorFusionWorkload": "tensorfusionworkloads",
    "WorkloadProfile": "workloadprofiles",
    "TensorFusionConnection": "tensorfusionconnections",
    "GPUResourceQuota": "gpuresourcequotas",
    "GPUNodeClass": "gpunodeclasses",
    "GPUNodeClaim": "gpunodeclaims",
    "SchedulingConfigTemplate": "schedulingconfigtemplates",
    "ProviderConfig": "providerconfigs",
    "Pod": "pods",
    "Node": "nodes",
}


def snake_to_camel(s: str) -> str:
    parts = s.split("_")
    return parts[0] + "".join(p.title() for p in parts[1:])


def camel_to_snake(s: str) -> str:
    out = []
    for ch in s:
        if ch.isupper():
            out.append("_")
            out.append(ch.lower())
        else:
            out.append(ch)
    return "".join(out)


# --------------------------------------------------- dataclass ⇄ wire dict


def _value_to_wire(v: Any) -> Any:
    if dataclasses.is_dataclass(v):
        return _dc_to_wire(v)
    if isinstance(v, list):
        return [_value_to_wire(x) for x in v]
    if isinstance(v, dict):
        return dict(v)  # data dict: keys untouched
    return v


def _dc_to_wire(obj: Any) -> Dict[str, Any]:
    out = {}
    for f in dataclasses.fields(obj):
        v = getattr(obj, f.name)
        if v is None:
            continue
        out[snake_to_camel(f.name.rstrip("_"))] = _value_to_wire(v)
    return out


def _wire_to_dc(cls: Type, d: Any) -> Any:
    if not (dataclasses.is_dataclass(cls) and isinstance(d, dict)):
        return d
    hints = typing.get_type_hints(cls)
    kwargs = {}
    for f in dataclasses.fields(cls):
        wire = snake_to_camel(f.name.rstrip("_"))
        if wire not in d:
            continue
        v = d[wire]
        t = hints.get(f.name)
        origin = typing.get_origin(t)
        if dataclasses.is_dataclass(t) and isinstance(v, dict):
            v = _wire_to_dc(t, v)
        elif origin is list and isinstance(v, list) and v:
            (et,) = typing.get_args(t)
            if dataclasses.is_dataclass(et):
                v = [_wire_to_dc(et, x) for x in v]
        elif origin is typing.Union and isinstance(v, dict):
            args = [a for a in typing.get_args(t) if a is not type(None)]
            if len(args) == 1 and dataclasses.is_dataclass(args[0]):
                v = _wire_to_dc(args[0], v)
        kwargs[f.name] = v
    return cls(**kwargs)


# ------------------------------------------------------------- metadata


def _ts_to_k8s(ts: Optional[float]) -> Optional[str]:
    if not ts:
        return None
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime(ts))


def _ts_from_k8s(s: Optional[str]) -> float:
    if not s:
        return 0.0
    try:
        return time.mktime(time.strptime(s[:19], "%Y-%m-%dT%H:%M:%S")) \
            - time.timezone
    except ValueError:
        return 0.0


def meta_to_k8s(meta: T.ObjectMeta) -> Dict[str, Any]:
    m: Dict[str, Any] = {"name": meta.name}
    if meta.namespace:
        m["namespace"] = meta.namespace
    if meta.uid:
        m["uid"] = meta.uid
    if meta.labels:
        m["labels"] = dict(meta.labels)
    if meta.annotations:
        m["annotations"] = dict(meta.annotations)
    if meta.resource_version:
        m["resourceVersion"] = str(meta.resource_version)
    ts = _ts_to_k8s(meta.creation_ts)
    if ts:
        m["creationTimestamp"] = ts
    if meta.deletion_ts:
        m["deletionTimestamp"] = _ts_to_k8s(meta.deletion_ts)
    if meta.finalizers:
        m["finalizers"] = list(meta.finalizers)
    if meta.owner:
        # internal owner ref "<Kind>/<namespace>/<name>"
        kind, _, rest = meta.owner.partition("/")
        ns, _, name = rest.partition("/")
        m["ownerReferences"] = [{
            "apiVersion": API_VERSION if kind in CRD_KINDS else "v1",
            "kind": kind, "name": name, "uid": "",
            "controller": True, "blockOwnerDeletion": True,
        }]
    return m


def meta_from_k8s(m: Dict[str, Any]) -> T.ObjectMeta:
    meta = T.ObjectMeta(
        name=m.get("name", "") or m.get("generateName", ""),
        namespace=m.get("namespace", ""),
        labels=dict(m.get("labels") or {}),
        annotations=dict(m.get("annotations") or {}),
        finalizers=list(m.get("finalizers") or []),
    )
    if m.get("uid"):
        meta.uid = m["uid"]
    rv = m.get("resourceVersion")
    if rv:
        try:
            meta.resource_version = int(rv)
        except ValueError:
            meta.resource_version = 0
    meta.creation_ts = _ts_from_k8s(m.get("creationTimestamp")) or meta.creation_ts
    if m.get("deletionTimestamp"):
        meta.deletion_ts = _ts_from_k8s(m["deletionTimestamp"])
    for ref in m.get("ownerReferences") or []:
        if ref.get("controller"):
            ns = meta.namespace if ref.get("kind") in (
                "TensorFusionWorkload", "WorkloadProfile",
                "TensorFusionConnection", "GPUResourceQuota", "Pod") else ""
            meta.owner = f"{ref.get('kind')}/{ns}/{ref.get('name')}"
            break
    return meta


# --------------------------------------------------------------- TF CRDs


def to_k8s(obj: T.TFObject) -> Dict[str, Any]:
    """Internal object → full k8s wire object."""

    if obj.kind == "Pod":
        return pod_to_k8s(obj)
    if obj.kind == "Node":
        return node_to_k8s(obj)
    spec = {}
    status = None
    for f in dataclasses.fields(obj):
        if f.name in ("meta", "kind"):
            continue
        v = getattr(obj, f.name)
        if f.name == "status":
            status = _value_to_wire(v)
            continue
        if v is None:
            continue
        spec[snake_to_camel(f.name.rstrip("_"))] = _value_to_wire(v)
    out = {
        "apiVersion": API_VERSION,
        "kind": obj.kind,
        "metadata": meta_to_k8s(obj.meta),
        "spec": spec,
    }
    if status is not None:
        out["status"] = status
    return out


def from_k8s(d: Dict[str, Any]) -> T.TFObject:
    """k8s wire object → internal object."""

    kind = d.get("kind", "")
    if kind == "Pod":
        return pod_from_k8s(d)
    if kind == "Node":
        return node_from_k8s(d)
    cls = getattr(T, kind, None)
    if cls is None or not dataclasses.is_dataclass(cls):
        raise ValueError(f"unknown kind {kind!r}")
    body = dict(d.get("spec") or {})
    obj = _wire_to_dc(cls, body)
    obj.kind = kind
    obj.meta = meta_from_k8s(d.get("metadata") or {})
    if "status" in d:
        hints = typing.get_type_hints(cls)
        st = hints.get("status")
        if st is not None:
            origin = typing.get_origin(st)
            if origin is typing.Union:
                st = [a for a in typing.get_args(st)
                      if a is not type(None)][0]
            obj.status = _wire_to_dc(st, d["status"])
    return obj


# ------------------------------------------------------------ corev1 Pod


def pod_to_k8s(p: T.Pod) -> Dict[str, Any]:
    containers = []
    for c in p.containers:
        cc: Dict[str, Any] = {"name": c.name}
        if c.image:
            cc["image"] = c.image
        if c.command:
            cc["command"] = list(c.command)
        if c.env:
            cc["env"] = [{"name": k, "value": v} for k, v in c.env.items()]
        if c.resources:
            cc["resources"] = {"limits": dict(c.resources),
                               "requests": dict(c.resources)}
        if c.volume_mounts:
            cc["volumeMounts"] = [dict(v) for v in c.volume_mounts]
        containers.append(cc)
    spec: Dict[str, Any] = {"containers": containers}
    if p.scheduler_name and p.scheduler_name != "default":
        spec["schedulerName"] = p.scheduler_name
    if p.node_selector:
        spec["nodeSelector"] = dict(p.node_selector)
    if p.status.node:
        spec["nodeName"] = p.status.node
    status: Dict[str, Any] = {"phase": _POD_PHASE_OUT.get(
        p.status.phase, p.status.phase)}
    if p.status.pod_ip:
        status["podIP"] = p.status.pod_ip
    if p.status.host_ip:
        status["hostIP"] = p.status.host_ip
    return {"apiVersion": "v1", "kind": "Pod",
            "metadata": meta_to_k8s(p.meta), "spec": spec, "status": status}


# internal phases ⇄ corev1 phases ("Scheduled" is internal-only — on the
# wire it is Pending + spec.nodeName set, which is how kube models it)
_POD_PHASE_OUT = {"Scheduled": "Pending"}


def pod_from_k8s(d: Dict[str, Any]) -> T.Pod:
    p = T.Pod()
    p.meta = meta_from_k8s(d.get("metadata") or {})
    if not p.meta.namespace:
        p.meta.namespace = "default"
    spec = d.get("spec") or {}
    p.scheduler_name = spec.get("schedulerName", "default")
    p.node_selector = dict(spec.get("nodeSelector") or {})
    for c in spec.get("containers") or []:
        cc = T.Container(name=c.get("name", "main"),
                         image=c.get("image", ""),
                         command=list(c.get("command") or []))
        for e in c.get("env") or []:
            if "value" in e:
                cc.env[e["name"]] = e["value"]
        res = c.get("resources") or {}
        for kind in ("limits", "requests"):
            for k, v in (res.get(kind) or {}).items():
                cc.resources[k] = str(v)
        cc.volume_mounts = [dict(v) for v in c.get("volumeMounts") or []]
        p.containers.append(cc)
    st = d.get("status") or {}
    p.status.phase = st.get("phase", "Pending")
    p.status.pod_ip = st.get("podIP", "")
    p.status.host_ip = st.get("hostIP", "")
    node = spec.get("nodeName", "")
    if node:
        p.status.node = node
        if p.status.phase == "Pending":
            p.status.phase = "Scheduled"
    return p


# ----------------------------------------------------------- corev1 Node


def node_to_k8s(n: T.Node) -> Dict[str, Any]:
    meta = meta_to_k8s(n.meta)
    labels = dict(n.labels_)
    if labels:
        meta.setdefault("labels", {}).update(labels)
    ready = n.status_phase == "Ready"
    spec: Dict[str, Any] = {}
    if n.taints:
        spec["taints"] = [dict(t) for t in n.taints]
    return {
        "apiVersion": "v1", "kind": "Node", "metadata": meta,
        "spec": spec,
        "status": {
            "capacity": {k: str(v) for k, v in n.capacity.items()},
            "addresses": [{"type": "InternalIP", "address": n.address}],
            "conditions": [{"type": "Ready",
                            "status": "True" if ready else "False"}],
        },
    }


def node_from_k8s(d: Dict[str, Any]) -> T.Node:
    n = T.Node()
    n.meta = meta_from_k8s(d.get("metadata") or {})
    n.labels_ = dict(n.meta.labels)
    st = d.get("status") or {}
    for k, v in (st.get("capacity") or {}).items():
        try:
            n.capacity[k] = float(v)
        except (TypeError, ValueError):
            pass
    ready = any(c.get("type") == "Ready" and c.get("status") == "True"
                for c in st.get("conditions") or [])
    n.status_phase = "Ready" if ready else "NotReady"
    for a in st.get("addresses") or []:
        if a.get("type") == "InternalIP":
            n.address = a.get("address", n.address)
    n.taints = [dict(t) for t in
                (d.get("spec") or {}).get("taints") or []]
    return n
