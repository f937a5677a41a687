"""Placement strategies: score candidate GPUs / nodes.

Reference: internal/gpuallocator/strategy_*.go — CompactFirst (binpack),
LowLoadFirst (spread), NodeCompactGPULowLoad (default: pack nodes, spread
GPUs inside a node). Scores are 0..100, higher = preferred.

MI355X note: all 8 intra-node GPUs are one xGMI full mesh, so
unlike the NVLink-clique scoring of the reference there is no intra-node
"topology distance" to optimise — strategies optimise packing and NUMA
locality only; the gputopo plugin handles multi-GPU set selection.
"""
from __future__ import annotations

from typing import Dict, List

from ..api.types import GPU, AllocRequest, Resource


def _usage_fraction(g: GPU) -> float:
    cap, avail = g.status.capacity, g.status.available
    fr_t = 1.0 - (avail.tflops / cap.tflops) if cap.tflops else 0.0
    fr_v = 1.0 - (avail.vram / cap.vram) if cap.vram else 0.0
    return 0.5 * fr_t + 0.5 * fr_v


class Strategy:
    name = "base"

    def __init__(self, vram_weight: float = 0.7, tflops_weight: float = 0.3):
        self.vram_weight = vram_weight
        self.tflops_weight = tflops_weight

    def _after_fraction(self, g: GPU, req: Resource) -> float:
        """Used fraction of the device if the request landed on it."""

        cap, avail = g.status.capacity, g.status.available
        ft = 1.0 - ((avail.tflops - req.tflops) / cap.tflops) if cap.tflops else 0.0
        fv = 1.0 - ((avail.vram - req.vram) / cap.vram) if cap.vram else 0.0
        return self.tflops_weight * ft + self.vram_weight * fv

    def score_gpu(self, g: GPU, req: AllocRequest) -> float:
        raise NotImplementedError

    def score_node(self, gpus: List[GPU], req: AllocRequest) -> float:
        """Node-level score = mean of its best gpu_count device scores."""

        scores = sorted((self.score_gpu(g, req) for g in gpus), reverse=True)
        take = scores[:max(1, req.gpu_count)]
        return sum(take) / len(take)


class CompactFirst(Strategy):
    """Binpack: prefer the device that ends up fullest (minimise fragments,
    free whole GPUs for future big requests)."""

    name = "CompactFirst"

    def score_gpu(self, g: GPU, req: AllocRequest) -> float:
        return 100.0 * min(1.0, max(0.0, self._after_fraction(g, req.request)))


class LowLoadFirst(Strategy):
    """Spread: prefer the emptiest device (minimise interference)."""

    name = "LowLoadFirst"

    def score_gpu(self, g: GPU, req: AllocRequest) -> float:
        return 100.0 * (1.0 - min(1.0, max(0.0, self._after_fraction(g, req.request))))


class NodeCompactGPULowLoad(Strategy):
    """Default: pack nodes (so whole nodes drain for defrag/provisioning),
    but spread across GPUs inside the chosen node."""

    name = "NodeCompactGPULowLoad"

    def score_gpu(self, g: GPU, req: AllocRequest) -> float:
        return 100.0 * (1.0 - min(1.0, max(0.0, self._after_fraction(g, req.request))))

    def score_node(self, gpus: List[GPU], req: AllocRequest) -> float:
        used = sum(_usage_fraction(g) for g in gpus) / max(1, len(gpus))
        return 100.0 * used


STRATEGIES: Dict[str, type] = {
    "CompactFirst": CompactFirst,
    "LowLoadFirst": LowLoadFirst,
    "NodeCompactGPULowLoad": NodeCompactGPULowLoad,
}


def make_strategy(name: str, vram_weight: float = 0.7,
                  tflops_weight: float = 0.3) -> Strategy:
    cls = STRATEGIES.get(name, NodeCompactGPULowLoad)
    return cls(vram_weight=vram_weight, tflops_weight=tflops_weight)
