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

# AMD compute-partition modes on an 8-XCD MI355X: mode -> (slices,
# xcds per slice, paired memory interleave). The mode is DEVICE-GLOBAL
# (amdsmi_set_gpu_compute_partition switches the whole GPU), so slabs of
# different modes can never coexist on one device — unlike NVIDIA MIG's
# per-slice profiles (reference partition_strategy.go:90-236).
PARTITION_MODES = {
    "SPX": (1, 8, "NPS1"),
    "DPX": (2, 4, "NPS2"),
    "QPX": (4, 2, "NPS4"),
    "CPX": (8, 1, "NPS4"),
}


def default_partition_templates() -> List[PartitionTemplate]:
    """One template per compute-partition mode, sized by its XCD share
    (tflops/vram proportional; NPS pairing per mode)."""

    out = []
    for mode, (slices, xcds, nps) in PARTITION_MODES.items():
        if slices == 1:
            continue  # SPX = whole GPU, not a partition template
        out.append(PartitionTemplate(
            id=f"{mode.lower()}-{xcds}xcd",
            name=f"{xcds}xcd.{(C.MI355X_VRAM_BYTES // slices) >> 30}gb",
            xcds=xcds,
            compute_percent=100.0 / slices,
            tflops=C.MI355X_BF16_TFLOPS / slices,
            vram=C.MI355X_VRAM_BYTES // slices,
            placements=list(range(0, C.MI355X_XCDS, xcds)),
            mode=mode,
            memory_mode=nps,
        ))
    return out


def device_partition_mode(g: GPU) -> str:
    """The compute-partition mode a device is committed to, derived from
    its bound partitions ("" = unpartitioned, any mode may claim it)."""

    for p in g.status.allocated_partitions:
        tid = p.template_id
        for mode, (_, xcds, _) in PARTITION_MODES.items():
            if tid.startswith(mode.lower()):
                return mode
        if p.xcds:  # legacy slab templates: infer from slab width
            for mode, (_, xcds, _) in PARTITION_MODES.items():
                if xcds == len(p.xcds):
                    return mode
    return ""


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
    """First allowed placement whose XCDs are all free on this device.
    Enforces mode exclusivity: a device committed to one compute-
    partition mode rejects templates of any other (the mode is a
    device-global amdsmi setting on MI355X)."""

    if t.mode:
        cur = device_partition_mode(g)
        if cur and cur != t.mode:
            return None
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
