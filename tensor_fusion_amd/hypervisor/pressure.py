"""VRAM pressure controller — the node-level oversubscription brain.

BASELINE config 4: N vGPUs whose provisioned caps sum past physical HBM
(e.g. 4 × 96 GB on one 288 GB MI355X). Each tenant's limiter spills
over-budget allocations into the host-DRAM managed tier and migrates by
LRU (native/limiter/hip_limiter.cpp); THIS controller closes the loop
across tenants:

  - dynamic device budgets: each worker's shm mem_limit_bytes is set to
    min(provisioned cap, QoS-weighted fair share of usable HBM). Under
    density a 96 GB-provisioned tenant may get a 60 GB HBM budget — the
    overflow runs from the host tier and is promoted back as neighbours
    leave (reference surface: Oversubscription gpupool_types.go:64-86).
  - pressure signal: when measured device-free drops below the reserve,
    the controller raises FLAG_VRAM_PRESSURE on the lowest-QoS workers
    (the /trap victim policy, reference legacy.go:124-150) until free
    recovers; their tier threads demote cold ranges.

Runs alongside the ERL loop at the same 500 ms-class cadence.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple

from .. import constants as C
from . import shm as S

QOS_WEIGHT = {C.QosLow: 1.0, C.QosMedium: 2.0, C.QosHigh: 4.0,
              C.QosCritical: 8.0}


@dataclass
class PressureWorker:
    page: S.WorkerShm
    qos: str = C.QosMedium
    provisioned_bytes: int = 0  # the vGPU's contract cap (annotation)
    device_index: int = 0
    pressured: bool = False


@dataclass
class PressureDecision:
    budgets: Dict[str, int] = field(default_factory=dict)  # path -> bytes
    pressured: List[str] = field(default_factory=list)
    free_bytes: int = 0
    oversubscribed: bool = False


class PressureController:
    """free_fn() returns (free_bytes, total_bytes) for the device —
    normally the accelerator lib's memory info; injectable for tests."""

    def __init__(self, free_fn: Callable[[], Tuple[int, int]],
                 reserve_bytes: int = 4 << 30,
                 interval_s: float = 0.5):
        self.free_fn = free_fn
        self.reserve_bytes = reserve_bytes
        self.interval_s = interval_s
        self.workers: Dict[str, PressureWorker] = {}
        self._mu = threading.Lock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.last: PressureDecision = PressureDecision()

    # ------------------------------------------------------------ admin

    def attach(self, page: S.WorkerShm, qos: str = C.QosMedium,
               provisioned_bytes: int = 0, device_index: int = 0):
        with self._mu:
            self.workers[page.path] = PressureWorker(
                page, qos, provisioned_bytes, device_index)

    def detach(self, path: str):
        with self._mu:
            self.workers.pop(path, None)

    # ------------------------------------------------------------- tick

    def tick(self) -> PressureDecision:
        with self._mu:
            workers = dict(self.workers)
        d = PressureDecision()
        if not workers:
            return d
        try:
            free, total = self.free_fn()
        except Exception:
            return d
        d.free_bytes = free
        usable = max(total - self.reserve_bytes, 0)
        provisioned_sum = sum(w.provisioned_bytes or total
                              for w in workers.values())
        d.oversubscribed = provisioned_sum > usable

        # ---- dynamic HBM budgets: QoS-weighted shares of usable HBM,
        # capped at each vGPU's provisioned bytes; spare capacity from
        # under-provisioned tenants is redistributed (water-filling)
        pool = usable
        remaining = dict(workers)
        budgets: Dict[str, int] = {}
        while remaining and pool > 0:
            wsum = sum(QOS_WEIGHT.get(w.qos, 2.0)
                       for w in remaining.values())
            if wsum <= 0:
                break
            per_weight = pool / wsum
            done = []
            for path, w in remaining.items():
                share = int(per_weight * QOS_WEIGHT.get(w.qos, 2.0))
                cap = w.provisioned_bytes or usable
                if cap <= share:
                    budgets[path] = cap
                    pool -= cap
                    done.append(path)
            if not done:
                for path, w in remaining.items():
                    budgets[path] = int(
                        per_weight * QOS_WEIGHT.get(w.qos, 2.0))
                pool = 0
                remaining = {}
                break
            for p in done:
                remaining.pop(p)
        for path, w in workers.items():
            b = budgets.get(path, 0)
            d.budgets[path] = b
            try:
                w.page.set_limits(0, mem_limit_bytes=b)
            except (OSError, ValueError):
                continue

        # ---- pressure: free below reserve → demote lowest-QoS first
        # (hysteresis: release at 2x reserve so flags don't flap)
        order = sorted(workers.items(),
                       key=lambda kv: QOS_WEIGHT.get(kv[1].qos, 2.0))
        if free < self.reserve_bytes:
            for path, w in order:
                try:
                    dev = w.page.device(0)
                    resident = dev.total_used
                except (OSError, ValueError):
                    continue
                if resident > d.budgets.get(path, 0) or not w.pressured:
                    w.page.set_flag(S.FLAG_VRAM_PRESSURE, True)
                    w.pressured = True
                    d.pressured.append(path)
                    break  # one victim per tick (reference /trap picks
                           # low-QoS victims incrementally)
        elif free > 2 * self.reserve_bytes:
            for path, w in workers.items():
                if w.pressured:
                    w.page.set_flag(S.FLAG_VRAM_PRESSURE, False)
                    w.pressured = False
        self.last = d
        return d

    # ------------------------------------------------------------- loop

    def start(self) -> "PressureController":
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="vram-pressure")
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()

    def _run(self):
        while not self._stop.wait(self.interval_s):
            try:
                self.tick()
            except Exception:
                pass
