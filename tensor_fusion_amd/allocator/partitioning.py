"""Partitioned isolation: MI355X compute-partition (XCD-slab) strategy.

Reference: internal/gpuallocator/partitioned_scheduling.go:45-279 +
partition_strategy.go:29-423 (NVIDIA MIG slot bitmaps, Ascend vGroups).
MI355X has no MIG; its native partitioning is the CPX/SPX compute-partition
modes — a GPU splits into 1/2/4/8 slices along its 8 XCDs, each slice
getting its XCDs' 32 CUs and a proportional HBM3E share (NPS1/NPS4 memory
interleave). We model a partition as an XCD slab: template `xcdN` covers N
contiguous XCDs starting at an allowed placement offset, tracked per device
with an 8-bit occupancy bitmap (1 bit per XCD).

Template matching follows the reference's waste-score: pick the smallest
template satisfying tflops+vram with score = 0.6*compute_waste+0.4*vram_waste
(partitioned_scheduling.go:30-31).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

from .. import constants as C
from ..api.types import GPU, AllocRequest, GPUPartition, PartitionTemplate, Resource

COMPUTE_WASTE_WEIGHT = 0.6
VRAM_WASTE_WEIGHT = 0.4


def match_partition_template(req: AllocRequest,
                             templates: List[PartitionTemplate]
                             ) -> Optional[PartitionTemplate]:
    """Smallest template that satisfies the request (min waste score)."""

    best: Optional[PartitionTemplate] = None
    best_score = float("inf")
    for t in templates:
        if t.tflops + 1e-9 < req.request.tflops or t.vram < req.request.vram:
            continue
        cw = (t.tflops - req.request.tflops) / t.tflops if t.tflops else 0.0
        vw = (t.vram - req.request.vram) / t.vram if t.vram else 0.0
        score = COMPUTE_WASTE_WEIGHT * cw + VRAM_WASTE_WEIGHT * vw
        if score < best_score:
            best, best_score = t, score
    return best


def occupancy_bitmap(g: GPU) -> int:
    bm = 0
    for p in g.status.allocated_partitions:
        for x in p.xcds:
            bm |= 1 << x
    return bm


def find_slot(g: GPU, t: PartitionTemplate) -> Optional[List[int]]:
    """First allowed placement whose XCDs are all free on this device."""

    bm = occupancy_bitmap(g)
    placements = t.placements or list(range(0, C.MI355X_XCDS, t.xcds))
    for start in placements:
        if start + t.xcds > C.MI355X_XCDS:
            continue
        xcds = list(range(start, start + t.xcds))
        if all(not (bm >> x) & 1 for x in xcds):
            return xcds
    return None


@dataclass
class PartitionPlacement:
    template: PartitionTemplate
    xcds: List[int]

    def to_partition(self, req: AllocRequest, partition_id: str) -> GPUPartition:
        return GPUPartition(
            partition_id=partition_id,
            template_id=self.template.id,
            workload=req.workload,
            pod=req.pod_key,
            resource=Resource(tflops=self.template.tflops,
                              vram=self.template.vram,
                              compute_percent=self.template.compute_percent),
            xcds=self.xcds,
        )


def place_partition(g: GPU, req: AllocRequest,
                    templates: List[PartitionTemplate]
                    ) -> Optional[PartitionPlacement]:
    t = match_partition_template(req, templates)
    if t is None:
        return None
    xcds = find_slot(g, t)
    if xcds is None:
        return None
    return PartitionPlacement(template=t, xcds=xcds)


def cu_mask_for_xcds(xcds: List[int]) -> str:
    """HSA_CU_MASK value confining a process to the given XCDs.

    MI355X: 256 CUs, XCD x owns CUs [32x, 32x+32). Format is the ROCr
    `queue:start-end,...` CU list; we emit the global mask form used by
    HSA_CU_MASK (ranges of CU ids)."""

    ranges = []
    for x in sorted(xcds):
        lo, hi = 32 * x, 32 * x + 31
        if ranges and ranges[-1][1] + 1 == lo:
            ranges[-1] = (ranges[-1][0], hi)
        else:
            ranges.append((lo, hi))
    return ",".join(f"{lo}-{hi}" for lo, hi in ranges)


def cu_mask_for_percent(percent: float) -> Tuple[str, int]:
    """Hard (non-partitioned) isolation: a CU range covering `percent` of the
    256 CUs, spread across XCDs in whole-XCD chunks first for cache locality.
    Returns (mask, cu_count). 1-CU granularity = 0.39%."""

    cus = max(1, min(C.MI355X_CUS, round(C.MI355X_CUS * percent / 100.0)))
    return f"0-{cus - 1}", cus
