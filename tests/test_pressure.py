"""VRAM pressure controller (hypervisor/pressure.py) — CPU tests over
real shm pages with an injected free-memory function.

Reference behavior being reproduced: Oversubscription budgets
(gpupool_types.go:64-86) + /trap low-QoS victim selection
(legacy.go:124-150).
"""
import pytest

import tensor_fusion_amd.constants as C
from tensor_fusion_amd.hypervisor import shm as S
from tensor_fusion_amd.hypervisor.pressure import PressureController

GB = 1 << 30
TOTAL = 288 * GB


def mk_page(tmp_path, name, used=0, vmm=0):
    page = S.WorkerShm.create(str(tmp_path / name / "shm"))
    page.set_device(0, f"uuid-{name}", up_limit_percent=100,
                    mem_limit_bytes=96 * GB, total_cus=256,
                    refill_rate=0.0, capacity=0.0)
    if used:
        page.write_u64(page._dev_off(0) + S.E_MEM_USED, used)
    if vmm:
        page.write_u64(page._dev_off(0) + S.E_VMM_BYTES, vmm)
    return page


class TestBudgets:
    def test_qos_weighted_waterfill(self, tmp_path):
        free = [200 * GB]
        pc = PressureController(lambda: (free[0], TOTAL),
                                reserve_bytes=4 * GB)
        pages = {}
        for name, qos in (("low", C.QosLow), ("med", C.QosMedium),
                          ("high", C.QosHigh)):
            p = mk_page(tmp_path, name)
            pages[name] = p
            pc.attach(p, qos=qos, provisioned_bytes=96 * GB)
        d = pc.tick()
        assert d.oversubscribed  # 3 x 96 GB > 284 GB usable
        budgets = {n: d.budgets[pages[n].path] for n in pages}
        # high QoS gets its full provisioned cap; the rest share what's
        # left, weighted; nothing exceeds its cap; the sum fits usable
        assert budgets["high"] == 96 * GB
        assert budgets["med"] >= budgets["low"]
        assert budgets["low"] > 0
        assert sum(budgets.values()) <= TOTAL - 4 * GB + 1
        for n in pages:
            assert budgets[n] <= 96 * GB
        # budgets land in each worker's shm mem_limit
        for n, p in pages.items():
            assert p.device(0).mem_limit_bytes == budgets[n]

    def test_undersubscribed_gets_full_caps(self, tmp_path):
        pc = PressureController(lambda: (250 * GB, TOTAL),
                                reserve_bytes=4 * GB)
        p1 = mk_page(tmp_path, "a")
        p2 = mk_page(tmp_path, "b")
        pc.attach(p1, qos=C.QosMedium, provisioned_bytes=96 * GB)
        pc.attach(p2, qos=C.QosMedium, provisioned_bytes=96 * GB)
        d = pc.tick()
        assert not d.oversubscribed
        assert d.budgets[p1.path] == 96 * GB
        assert d.budgets[p2.path] == 96 * GB


class TestPressure:
    def test_low_qos_victim_first_with_hysteresis(self, tmp_path):
        free = [2 * GB]  # below the 4 GB reserve
        pc = PressureController(lambda: (free[0], TOTAL),
                                reserve_bytes=4 * GB)
        plow = mk_page(tmp_path, "low", used=80 * GB)
        phigh = mk_page(tmp_path, "high", used=80 * GB)
        pc.attach(plow, qos=C.QosLow, provisioned_bytes=96 * GB)
        pc.attach(phigh, qos=C.QosCritical, provisioned_bytes=96 * GB)
        d = pc.tick()
        assert d.pressured == [plow.path]
        assert plow.flags() & S.FLAG_VRAM_PRESSURE
        assert not (phigh.flags() & S.FLAG_VRAM_PRESSURE)
        # free recovers a little (< 2x reserve): flag must NOT flap off
        free[0] = 6 * GB
        pc.tick()
        assert plow.flags() & S.FLAG_VRAM_PRESSURE
        # full recovery clears it
        free[0] = 10 * GB
        pc.tick()
        assert not (plow.flags() & S.FLAG_VRAM_PRESSURE)

    def test_vmm_bytes_counted_as_residency(self, tmp_path):
        """The remoting worker reports heap usage via vmm_bytes; the
        controller must see it (total_used) when picking victims."""

        free = [1 * GB]
        pc = PressureController(lambda: (free[0], TOTAL),
                                reserve_bytes=4 * GB)
        p = mk_page(tmp_path, "w", used=0, vmm=90 * GB)
        pc.attach(p, qos=C.QosLow, provisioned_bytes=96 * GB)
        d = pc.tick()
        assert d.pressured == [p.path]
        snap = p.device(0)
        assert snap.total_used == 90 * GB

    def test_detach_stops_budgeting(self, tmp_path):
        pc = PressureController(lambda: (100 * GB, TOTAL))
        p = mk_page(tmp_path, "w")
        pc.attach(p, provisioned_bytes=96 * GB)
        assert pc.tick().budgets
        pc.detach(p.path)
        assert not pc.tick().budgets


class TestBudgetInvariants:
    def test_waterfill_invariants_randomized(self, tmp_path):
        """Random tenant mixes: budgets never exceed caps, never exceed
        usable HBM in total, respect QoS ordering at equal caps, and
        spare capacity from capped tenants is redistributed."""

        import random
        rng = random.Random(9)
        for trial in range(25):
            n = rng.randint(1, 6)
            free = rng.randint(1, 280) * GB
            pc = PressureController(lambda: (free, TOTAL),
                                    reserve_bytes=4 * GB)
            caps, qoses, pages = [], [], []
            for i in range(n):
                cap = rng.choice([24, 48, 96, 192]) * GB
                qos = rng.choice([C.QosLow, C.QosMedium, C.QosHigh,
                                  C.QosCritical])
                p = mk_page(tmp_path, f"t{trial}-{i}")
                pc.attach(p, qos=qos, provisioned_bytes=cap)
                caps.append(cap)
                qoses.append(qos)
                pages.append(p)
            d = pc.tick()
            budgets = [d.budgets[p.path] for p in pages]
            usable = TOTAL - 4 * GB
            assert sum(budgets) <= usable + n  # rounding slack
            for b, cap in zip(budgets, caps):
                assert 0 <= b <= cap
            # QoS ordering among equal-cap, uncapped tenants
            from tensor_fusion_amd.hypervisor.pressure import QOS_WEIGHT
            for i in range(n):
                for j in range(n):
                    if caps[i] == caps[j] and budgets[i] < caps[i] \
                            and budgets[j] < caps[j]:
                        if QOS_WEIGHT[qoses[i]] > QOS_WEIGHT[qoses[j]]:
                            assert budgets[i] >= budgets[j] - 1
            # fully-provisionable fleets get their full caps
            if sum(caps) <= usable:
                for b, cap in zip(budgets, caps):
                    assert b == cap
