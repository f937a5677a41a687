"""GPUNetworkTopologyAware — interconnect-aware GPU-set selection.

Reference: internal/scheduler/gputopo/ — consumes GPUResourcesFit's
CycleState, picks the topology-best GPU combination per node (NVLink-clique
search + NUMA evaluators on NVIDIA).

MI355X redesign: every intra-node GPU pair has a dedicated
xGMI link (7 links/GPU, full mesh of 8) so all intra-node sets are
equal-cost for collectives — NVLink-style clique enumeration is pointless.
What still matters:
  1. NUMA locality of host staging buffers: prefer a GPU set inside one
     NUMA domain (tier 1 boundary), since RCCL host bounce buffers and the
     remoting worker's pinned arenas allocate NUMA-local.
  2. Fragmentation: prefer combinations that leave contiguous NUMA domains
     free for future gangs.
Modes soft (bonus score) / hard (filter out NUMA-crossing sets when a
one-domain set exists anywhere).
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

from ..allocator.gpuallocator import GpuAllocator
from ..api.types import Pod
from .framework import CycleState, Plugin, Status
from .gpuresources import S_REQ, S_SCORES, S_TOPO

NUMA_BONUS = 20.0


class GPUNetworkTopologyAware(Plugin):
    name = "GPUNetworkTopologyAware"

    def __init__(self, allocator: GpuAllocator, mode: str = "soft"):
        self.allocator = allocator
        self.mode = mode  # soft | hard

    def pre_filter(self, state: CycleState, pod: Pod
                   ) -> Tuple[Optional[List[str]], Status]:
        req = state.get(S_REQ)
        scores = state.get(S_SCORES)
        if req is None or not scores or req.gpu_count <= 1:
            return None, Status(code="Skip")
        combos: Dict[str, List[str]] = {}
        bonuses: Dict[str, float] = {}
        for node, ns in scores.items():
            names = list(ns.gpu_scores)
            if not names:  # allocator's fast path leaves gpu_scores lazy
                names = self.allocator.eligible_gpu_names(req, node)
            combo, same_numa = self._best_combo(node, names,
                                                ns.gpu_scores, req.gpu_count)
            if combo:
                combos[node] = combo
                bonuses[node] = NUMA_BONUS if same_numa else 0.0
        state[S_TOPO] = combos
        state[S_TOPO + "/bonus"] = bonuses
        if self.mode == "hard" and any(b > 0 for b in bonuses.values()):
            keep = [n for n, b in bonuses.items() if b > 0]
            return keep, Status.ok_()
        return None, Status.ok_()

    def _best_combo(self, node: str, names: List[str],
                    gpu_scores: Dict[str, float], count: int
                    ) -> Tuple[Optional[List[str]], bool]:
        gpus = [self.allocator.gpu(n) for n in names]
        gpus = [g for g in gpus if g is not None]
        if len(gpus) < count:
            return None, False
        by_numa: Dict[int, List] = {}
        for g in gpus:
            by_numa.setdefault(g.status.numa_node, []).append(g)
        # 1) a single NUMA domain that fits the whole set
        candidates = [lst for lst in by_numa.values() if len(lst) >= count]
        if candidates:
            # least-remaining-space domain first (anti-fragmentation)
            lst = min(candidates, key=len)
            lst = sorted(lst, key=lambda g: gpu_scores.get(g.meta.name, 0),
                         reverse=True)
            return [g.meta.name for g in lst[:count]], True
        # 2) spill across domains, fewest domains first, then score
        ordered = sorted(by_numa.values(), key=len, reverse=True)
        combo: List[str] = []
        for lst in ordered:
            lst = sorted(lst, key=lambda g: gpu_scores.get(g.meta.name, 0),
                         reverse=True)
            for g in lst:
                combo.append(g.meta.name)
                if len(combo) == count:
                    return combo, False
        return None, False

    def score(self, state: CycleState, pod: Pod, node: str) -> float:
        return state.get(S_TOPO + "/bonus", {}).get(node, 0.0)
