"""GPU filter chain — chain-of-responsibility over candidate devices.

Reference: internal/gpuallocator/filter/*.go (phase, resource, model, vendor,
index, isolation-mode, node-affinity, same-node, partition-template,
shared-whole-gpu). Same semantics, fresh implementation; the MI355X twist is
in the isolation/partition filters (CU-mask and XCD-slab aware).
"""
from __future__ import annotations

from typing import Callable, Dict, List, Optional, Tuple

from .. import constants as C
from ..api.types import GPU, AllocRequest

FilterFn = Callable[[AllocRequest, List[GPU]], Tuple[List[GPU], str]]


class FilterRegistry:
    """Ordered filter chain; each filter narrows the candidate list and
    reports why devices fell out (for scheduler failure messages)."""

    def __init__(self, filters: Optional[List[FilterFn]] = None):
        self.filters = filters or []

    def with_filters(self, *fns: FilterFn) -> "FilterRegistry":
        return FilterRegistry(self.filters + list(fns))

    def apply(self, req: AllocRequest, gpus: List[GPU]) -> Tuple[List[GPU], Dict[str, str]]:
        reasons: Dict[str, str] = {}
        cur = gpus
        for f in self.filters:
            if not cur:
                break
            cur, reason = f(req, cur)
            if reason:
                reasons[f.__name__] = reason
        return cur, reasons


# ------------------------------------------------------------ the filters


def phase_filter(req: AllocRequest, gpus: List[GPU]) -> Tuple[List[GPU], str]:
    out = [g for g in gpus if g.status.phase == "Ready"
           and g.status.used_by == "tensor-fusion"]
    return out, "" if out else "no GPU in Ready phase"


def resource_filter(req: AllocRequest, gpus: List[GPU]) -> Tuple[List[GPU], str]:
    out = [g for g in gpus if req.request.fits_in(g.status.available)]
    return out, "" if out else (
        f"insufficient resources (need {req.request.tflops:.0f} tflops / "
        f"{req.request.vram >> 30} GiB)")


def model_filter(req: AllocRequest, gpus: List[GPU]) -> Tuple[List[GPU], str]:
    if not req.gpu_model:
        return gpus, ""
    out = [g for g in gpus if g.status.model == req.gpu_model]
    return out, "" if out else f"no GPU of model {req.gpu_model}"


def vendor_filter(req: AllocRequest, gpus: List[GPU]) -> Tuple[List[GPU], str]:
    if not req.gpu_vendor:
        return gpus, ""
    out = [g for g in gpus if g.status.vendor.lower() == req.gpu_vendor.lower()]
    return out, "" if out else f"no GPU of vendor {req.gpu_vendor}"


def index_filter(req: AllocRequest, gpus: List[GPU]) -> Tuple[List[GPU], str]:
    if not req.gpu_indices:
        return gpus, ""
    want = set(req.gpu_indices)
    out = [g for g in gpus if g.status.index in want]
    return out, "" if out else f"no GPU at indices {sorted(want)}"


def isolation_filter(req: AllocRequest, gpus: List[GPU]) -> Tuple[List[GPU], str]:
    """A device already serving workloads in one isolation mode cannot mix
    modes (reference filter/isolation_filter semantics). Empty devices accept
    any mode; partitioned mode needs a device with free partition slots."""

    out = []
    for g in gpus:
        if not g.status.running_apps and not g.status.allocated_partitions:
            out.append(g)
            continue
        if req.partitioned or req.isolation_mode == C.IsolationPartitioned:
            if g.status.allocated_partitions or not g.status.running_apps:
                out.append(g)
            continue
        if g.status.isolation_mode == req.isolation_mode and \
                not g.status.allocated_partitions:
            out.append(g)
    return out, "" if out else f"isolation mode {req.isolation_mode} conflicts"


def node_affinity_filter(req: AllocRequest, gpus: List[GPU]) -> Tuple[List[GPU], str]:
    node = req.node_affinity.get("node")
    if not node:
        return gpus, ""
    out = [g for g in gpus if g.status.node == node]
    return out, "" if out else f"no GPU on node {node}"


def shared_whole_gpu_filter(req: AllocRequest, gpus: List[GPU]) -> Tuple[List[GPU], str]:
    """isolation=shared asks for the whole device: only fully-free GPUs."""

    if req.isolation_mode != C.IsolationShared:
        return gpus, ""
    out = [g for g in gpus
           if not g.status.running_apps
           and g.status.available.vram == g.status.capacity.vram]
    return out, "" if out else "no fully-free GPU for shared (whole-GPU) mode"


def make_same_node_filter(min_count: int) -> FilterFn:
    """For gpu_count>1 keep only nodes that can host the whole set
    (reference filter/node_filter.go SameNodeFilter)."""

    def same_node_filter(req: AllocRequest, gpus: List[GPU]) -> Tuple[List[GPU], str]:
        by_node: Dict[str, List[GPU]] = {}
        for g in gpus:
            by_node.setdefault(g.status.node, []).append(g)
        out: List[GPU] = []
        for node, lst in by_node.items():
            if len(lst) >= min_count:
                out.extend(lst)
        return out, "" if out else f"no node with {min_count} eligible GPUs"

    return same_node_filter


def default_registry() -> FilterRegistry:
    return FilterRegistry([
        phase_filter,
        model_filter,
        vendor_filter,
        index_filter,
        isolation_filter,
        shared_whole_gpu_filter,
        resource_filter,
        node_affinity_filter,
    ])
