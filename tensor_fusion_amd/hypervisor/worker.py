"""Worker controller — per-worker shm lifecycle, limits, orphan sweeps.

Reference: pkg/hypervisor/worker/controller.go:28-720 (ensureWorkerShared
Memory, shm sync loop writing mem usage + heartbeat checks, orphan shm
cleanup, per-worker metrics join, computeUpLimit).
"""
from __future__ import annotations

import os
import shutil
import threading
import time
from dataclasses import dataclass
from typing import Dict, List, Optional

from .. import constants as C
from .allocation import AllocationController, WorkerAllocation, WorkerSpec
from .device import DeviceController
from .erl import ErlQuotaController
from . import shm as S


@dataclass
class WorkerStatus:
    key: str
    allocation: WorkerAllocation
    shm: S.WorkerShm
    started: float
    last_heartbeat_ns: int = 0
    pids: Optional[List[int]] = None


class WorkerController:
    def __init__(self, devices: DeviceController,
                 allocator: AllocationController,
                 erl: Optional[ErlQuotaController] = None,
                 shm_root: str = C.ShmRoot,
                 pressure=None):
        self.devices = devices
        self.allocator = allocator
        self.erl = erl
        self.pressure = pressure  # hypervisor.pressure.PressureController
        self.shm_root = shm_root
        self.workers: Dict[str, WorkerStatus] = {}
        self._mu = threading.RLock()

    # ---------------------------------------------------------- lifecycle

    def add_worker(self, spec: WorkerSpec) -> WorkerStatus:
        with self._mu:
            if spec.key in self.workers:
                return self.workers[spec.key]
            alloc = self.allocator.allocate(spec)
            page = self._ensure_shm(alloc)
            st = WorkerStatus(key=spec.key, allocation=alloc, shm=page,
                              started=time.time())
            self.workers[spec.key] = st
            if self.erl:
                self.erl.attach(page)
            if self.pressure:
                self.pressure.attach(
                    page, qos=spec.qos,
                    provisioned_bytes=spec.vram_limit)
            return st

    def remove_worker(self, key: str, delete_shm: bool = True):
        with self._mu:
            st = self.workers.pop(key, None)
            if st is None:
                return
            if self.erl:
                self.erl.detach(st.shm.path)
            if self.pressure:
                self.pressure.detach(st.shm.path)
            self.allocator.deallocate(key)
            st.shm.close()
            if delete_shm:
                shutil.rmtree(os.path.dirname(st.shm.path), ignore_errors=True)

    def get(self, key: str) -> Optional[WorkerStatus]:
        with self._mu:
            return self.workers.get(key)

    def allocation_of(self, key: str) -> Optional[WorkerAllocation]:
        """Composed env/devices for a worker (device-plugin Allocate)."""

        with self._mu:
            st = self.workers.get(key)
            return st.allocation if st else None

    def list(self) -> List[WorkerStatus]:
        with self._mu:
            return list(self.workers.values())

    def _ensure_shm(self, alloc: WorkerAllocation) -> S.WorkerShm:
        """Create (open-not-truncate) the worker's shm page and program the
        per-device limit entries."""

        page = S.WorkerShm.create(alloc.shm_path)
        for i, dev in enumerate(alloc.devices[:S.MAX_DEVICES]):
            existing = page.device(i)
            rate = existing.erl_refill_rate if existing.active else 1000.0
            cap = existing.erl_capacity if existing.active else 200.0
            page.set_device(
                i, dev.uuid, up_limit_percent=alloc.up_limit_percent,
                mem_limit_bytes=alloc.spec.vram_limit,
                total_cus=dev.compute_units, refill_rate=rate, capacity=cap)
        return page

    # -------------------------------------------------------------- sync

    def register_pid(self, key: str, pid: int) -> bool:
        st = self.get(key)
        if st is None:
            return False
        st.shm.add_pid(pid)
        return True

    def sync_once(self):
        """Periodic pass: heartbeats, dead-pid sweep, orphan shm dirs
        (reference shm sync loop :402-529 + orphan cleanup :438-521)."""

        with self._mu:
            known = {os.path.dirname(st.shm.path) for st in self.workers.values()}
            workers = list(self.workers.values())
        for st in workers:
            st.shm.sweep_dead_pids()
            st.last_heartbeat_ns = st.shm.heartbeat()
            st.pids = st.shm.pids()
            st.shm.touch_hypervisor_heartbeat()
        # orphan sweep: shm dirs with no live worker and no live pids
        if os.path.isdir(self.shm_root):
            for ns in os.listdir(self.shm_root):
                nsdir = os.path.join(self.shm_root, ns)
                if not os.path.isdir(nsdir):
                    continue
                for pod in os.listdir(nsdir):
                    d = os.path.join(nsdir, pod)
                    if d in known:
                        continue
                    try:
                        page = S.WorkerShm.open(os.path.join(d, "shm"))
                        page.sweep_dead_pids()
                        stale = not page.pids()
                        page.close()
                    except (FileNotFoundError, ValueError, OSError):
                        stale = True
                    if stale:
                        shutil.rmtree(d, ignore_errors=True)

    # ----------------------------------------------------------- metrics

    def worker_metrics(self) -> List[dict]:
        """Join process info × worker pid sets → per-worker per-device usage
        (reference :212-290)."""

        out = []
        proc_by_dev = {}
        for d in self.devices.devices():
            proc_by_dev[d.index] = self.devices.processes(d.index)
        for st in self.list():
            pids = set(st.shm.pids())
            for i, dev in enumerate(st.allocation.devices):
                entry = st.shm.device(i) if i < S.MAX_DEVICES else None
                vram = sum(p.vram_bytes for p in proc_by_dev.get(dev.index, [])
                           if p.pid in pids)
                out.append({
                    "worker": st.key,
                    "workload": st.allocation.spec.workload,
                    "device": dev.uuid,
                    "vram_used": vram or (entry.pod_memory_used if entry else 0),
                    "launches": entry.launch_count if entry else 0,
                    "block_ns": entry.block_ns_total if entry else 0,
                    "up_limit": st.allocation.up_limit_percent,
                    "qos": st.allocation.spec.qos,
                })
        return out
