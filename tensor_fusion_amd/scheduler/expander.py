"""NodeExpander — auto-provision nodes for unschedulable GPU pods.

Reference: internal/scheduler/expander/handler.go:37-447 + unsched_queue.go:
collects unschedulable pods, simulates whether an empty node of a known
instance type would fit them, creates a GPUNodeClaim (Karpenter-style) and
tracks in-flight claims + pre-scheduled pods so one claim isn't created
per retry.
"""
from __future__ import annotations

import threading
import time
from typing import Dict, List, Optional

from .. import constants as C
from ..api.store import AlreadyExists, Store
from ..api.types import AllocRequest, GPUNodeClaim
from ..cloudprovider import cheapest_instance_for


class NodeExpander:
    def __init__(self, store: Store, cooldown_s: float = 5.0):
        self.store = store
        self.cooldown_s = cooldown_s
        self._lock = threading.Lock()
        self._pre_scheduled: Dict[str, str] = {}  # pod key -> claim name
        self._last_claim_ts = 0.0

    def in_flight_claims(self) -> List[GPUNodeClaim]:
        return [c for c in self.store.list("GPUNodeClaim")
                if c.status.phase in ("Pending", "Creating")]

    def handle_unschedulable(self, req: AllocRequest) -> Optional[str]:
        """Called by the scheduler when a pod can't fit anywhere. Returns
        the claim name backing this pod (new or pre-existing)."""

        with self._lock:
            prev = self._pre_scheduled.get(req.pod_key)
            if prev is not None:
                claim = self.store.try_get("GPUNodeClaim", prev)
                if claim is not None and claim.status.phase != "Failed":
                    return prev
                del self._pre_scheduled[req.pod_key]

            # an in-flight claim with room? ride it.
            for claim in self.in_flight_claims():
                riders = [k for k, v in self._pre_scheduled.items()
                          if v == claim.meta.name]
                inst = cheapest_instance_for(req.gpu_count, 0, 0)
                if inst and len(riders) < inst.gpu_count:
                    self._pre_scheduled[req.pod_key] = claim.meta.name
                    return claim.meta.name

            if time.time() - self._last_claim_ts < self.cooldown_s:
                return None
            inst = cheapest_instance_for(
                req.gpu_count,
                req.request.tflops * req.gpu_count,
                req.request.vram * req.gpu_count)
            if inst is None:
                return None
            name = f"claim-{req.namespace}-{req.pod_name}"[:60]
            claim = GPUNodeClaim()
            claim.meta.name = name
            claim.pool = req.pool
            claim.instance_type = inst.name
            try:
                self.store.create(claim)
            except AlreadyExists:
                pass
            self._last_claim_ts = time.time()
            self._pre_scheduled[req.pod_key] = name
            return name

    def forget_pod(self, pod_key: str):
        with self._lock:
            self._pre_scheduled.pop(pod_key, None)
