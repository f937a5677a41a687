"""GPUNodeProvider contract + mock implementation + static pricing.

Reference: internal/cloudprovider/types/type.go:23-33 (interface),
mock/ (test provider), pricing/pricing.go (static $/h tables used by the
NodeExpander to pick the cheapest instance satisfying a request).
"""
from __future__ import annotations

import threading
import uuid
from dataclasses import dataclass
from typing import Dict, List, Optional

from .. import constants as C


@dataclass
class InstanceType:
    name: str
    gpu_model: str
    gpu_count: int
    tflops_per_gpu: float
    vram_per_gpu: int
    cost_per_hour: float


# MI355X-era static pricing (stand-in for pricing/pricing.go data files).
PRICING_TABLE: List[InstanceType] = [
    InstanceType("mi355x.1g", C.MI355X_MODEL, 1, C.MI355X_BF16_TFLOPS,
                 C.MI355X_VRAM_BYTES, 4.0),
    InstanceType("mi355x.2g", C.MI355X_MODEL, 2, C.MI355X_BF16_TFLOPS,
                 C.MI355X_VRAM_BYTES, 7.8),
    InstanceType("mi355x.4g", C.MI355X_MODEL, 4, C.MI355X_BF16_TFLOPS,
                 C.MI355X_VRAM_BYTES, 15.2),
    InstanceType("mi355x.8g", C.MI355X_MODEL, 8, C.MI355X_BF16_TFLOPS,
                 C.MI355X_VRAM_BYTES, 29.6),
]


def cheapest_instance_for(gpu_count: int, tflops: float, vram: int,
                          table: Optional[List[InstanceType]] = None
                          ) -> Optional[InstanceType]:
    """Pick the cheapest instance that satisfies the aggregate request
    (reference expander picks instance types by fit + price)."""

    best = None
    for it in (table or PRICING_TABLE):
        if it.gpu_count < gpu_count:
            continue
        if it.tflops_per_gpu * it.gpu_count < tflops:
            continue
        if it.vram_per_gpu * it.gpu_count < vram:
            continue
        if best is None or it.cost_per_hour < best.cost_per_hour:
            best = it
    return best


class GPUNodeProvider:
    """Create/terminate cloud nodes (reference type.go:23-33)."""

    def create_node(self, claim) -> str:
        raise NotImplementedError

    def terminate_node(self, instance_id: str) -> None:
        raise NotImplementedError

    def node_status(self, instance_id: str) -> Optional[str]:
        """Returns the joined node name once the instance is up."""

        raise NotImplementedError


@dataclass
class _MockInstance:
    instance_id: str
    instance_type: str
    polls_until_ready: int = 1
    node_name: str = ""
    terminated: bool = False


class MockProvider(GPUNodeProvider):
    """In-memory provider: instances become Ready after N status polls
    (reference internal/cloudprovider/mock/)."""

    def __init__(self, polls_until_ready: int = 1, store=None,
                 gpus_per_node: int = 8):
        self._lock = threading.Lock()
        self.instances: Dict[str, _MockInstance] = {}
        self.polls_until_ready = polls_until_ready
        self.store = store  # when set, a Ready instance creates Node+GPU objs
        self.gpus_per_node = gpus_per_node

    def create_node(self, claim) -> str:
        iid = f"i-{uuid.uuid4().hex[:12]}"
        with self._lock:
            self.instances[iid] = _MockInstance(
                iid, getattr(claim, "instance_type", "mi355x.8g"),
                self.polls_until_ready)
        return iid

    def terminate_node(self, instance_id: str) -> None:
        with self._lock:
            inst = self.instances.get(instance_id)
            if inst:
                inst.terminated = True

    def node_status(self, instance_id: str) -> Optional[str]:
        with self._lock:
            inst = self.instances.get(instance_id)
            if inst is None or inst.terminated:
                return None
            if inst.node_name:
                return inst.node_name
            inst.polls_until_ready -= 1
            if inst.polls_until_ready > 0:
                return None
            inst.node_name = f"node-{instance_id}"
        if self.store is not None:
            self._materialize(inst)
        return inst.node_name

    def _materialize(self, inst: _MockInstance):
        from ..api.store import AlreadyExists
        from ..api.types import Node
        node = Node()
        node.meta.name = inst.node_name
        node.labels_ = {"tensor-fusion.ai/provisioned": "true"}
        try:
            self.store.create(node)
        except AlreadyExists:
            pass
