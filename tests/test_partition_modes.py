"""AMD compute-partition modes (SPX/DPX/QPX/CPX x NPS) — CPU tests.

Reference analog: NVIDIA MIG slot placement (partition_strategy.go:90-236).
The MI355X difference under test: the compute-partition mode is
DEVICE-GLOBAL (one amdsmi setting for the whole GPU), so templates of
different modes must never coexist on one device.
"""
import pytest

import tensor_fusion_amd.constants as C
from tensor_fusion_amd.allocator import partitioning as P
from tensor_fusion_amd.api.types import (GPU, AllocRequest, Resource,
                                         default_mi355x_partition_templates)


def mk_gpu(name="g0"):
    g = GPU()
    g.meta.name = name
    g.status.capacity = Resource(C.MI355X_BF16_TFLOPS, C.MI355X_VRAM_BYTES,
                                 100.0)
    g.status.available = Resource(C.MI355X_BF16_TFLOPS, C.MI355X_VRAM_BYTES,
                                  100.0)
    return g


def req(tflops, vram, name="p"):
    return AllocRequest(pod_name=name, workload="w", partitioned=True,
                        request=Resource(tflops, vram, 0.0),
                        limit=Resource(tflops, vram, 0.0))


TEMPLATES = default_mi355x_partition_templates()


class TestModeModel:
    def test_default_templates_cover_modes(self):
        modes = {t.mode for t in TEMPLATES}
        assert modes == {"CPX", "QPX", "DPX", "SPX"}
        cpx = next(t for t in TEMPLATES if t.mode == "CPX")
        assert cpx.xcds == 1 and cpx.memory_mode == "NPS4"
        assert len(cpx.placements) == 8
        dpx = next(t for t in TEMPLATES if t.mode == "DPX")
        assert dpx.xcds == 4 and dpx.vram == C.MI355X_VRAM_BYTES // 2

    def test_matching_picks_smallest_fitting_mode(self):
        t = P.match_partition_template(
            req(C.MI355X_BF16_TFLOPS / 8, 30 << 30), TEMPLATES)
        assert t.mode == "CPX"
        t = P.match_partition_template(
            req(C.MI355X_BF16_TFLOPS / 4, 60 << 30), TEMPLATES)
        assert t.mode == "QPX"


class TestModeExclusivity:
    def test_device_committed_to_cpx_rejects_qpx(self):
        g = mk_gpu()
        cpx = next(t for t in TEMPLATES if t.mode == "CPX")
        qpx = next(t for t in TEMPLATES if t.mode == "QPX")
        pl = P.place_partition(g, req(cpx.tflops, cpx.vram), [cpx])
        g.status.allocated_partitions.append(
            pl.to_partition(req(cpx.tflops, cpx.vram), "p0"))
        assert P.device_partition_mode(g) == "CPX"
        # a QPX template must not land on a CPX-committed device even
        # though 7 XCDs are free
        assert P.find_slot(g, qpx) is None
        # another CPX slice is fine
        assert P.find_slot(g, cpx) is not None

    def test_mode_released_when_partitions_gone(self):
        g = mk_gpu()
        cpx = next(t for t in TEMPLATES if t.mode == "CPX")
        qpx = next(t for t in TEMPLATES if t.mode == "QPX")
        pl = P.place_partition(g, req(cpx.tflops, cpx.vram), [cpx])
        part = pl.to_partition(req(cpx.tflops, cpx.vram), "p0")
        g.status.allocated_partitions.append(part)
        assert P.find_slot(g, qpx) is None
        g.status.allocated_partitions.remove(part)
        assert P.device_partition_mode(g) == ""
        assert P.find_slot(g, qpx) is not None


class TestSlotExhaustion:
    def test_cpx_eight_slots_then_full(self):
        g = mk_gpu()
        cpx = next(t for t in TEMPLATES if t.mode == "CPX")
        r = req(cpx.tflops, cpx.vram)
        seen = []
        for i in range(8):
            pl = P.place_partition(g, r, [cpx])
            assert pl is not None, f"slot {i} should fit"
            g.status.allocated_partitions.append(
                pl.to_partition(r, f"p{i}"))
            seen.extend(pl.xcds)
        assert sorted(seen) == list(range(8))  # every XCD used once
        assert P.place_partition(g, r, [cpx]) is None  # exhausted

    def test_qpx_four_slots_aligned(self):
        g = mk_gpu()
        qpx = next(t for t in TEMPLATES if t.mode == "QPX")
        r = req(qpx.tflops, qpx.vram)
        starts = []
        for i in range(4):
            pl = P.place_partition(g, r, [qpx])
            assert pl is not None
            starts.append(pl.xcds[0])
            g.status.allocated_partitions.append(
                pl.to_partition(r, f"p{i}"))
        assert starts == [0, 2, 4, 6]  # aligned placements only
        assert P.place_partition(g, r, [qpx]) is None

    def test_cu_mask_for_cpx_slot(self):
        assert P.cu_mask_for_xcds([3]) == "96-127"
        assert P.cu_mask_for_xcds([0, 1]) == "0-63"
        assert P.cu_mask_for_xcds([6, 7]) == "192-255"


class TestAcceleratorModeApi:
    def test_mock_get_set_roundtrip(self):
        from tensor_fusion_amd.hypervisor.device import Accelerator
        a = Accelerator(mock_devices=2)
        assert a.compute_partition(0) == "SPX"
        assert a.memory_partition(0) == "NPS1"
        assert a.set_compute_partition(0, "CPX")
        assert a.set_memory_partition(0, "NPS4")
        assert a.compute_partition(0) == "CPX"
        assert a.memory_partition(0) == "NPS4"
        # device 1 untouched
        assert a.compute_partition(1) == "SPX"
        # invalid mode rejected
        with pytest.raises(RuntimeError):
            a.set_compute_partition(0, "BOGUS")
        a.set_compute_partition(0, "SPX")
        a.set_memory_partition(0, "NPS1")
