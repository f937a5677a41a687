"""Per-namespace GPU resource quota with assumed-usage overlay.

Reference: internal/quota/quota_store.go:34-750 — check/allocate/deallocate/
assume/forget with an in-memory overlay for scheduler-assumed (not yet
committed) allocations, plus dirty sync of usage back into the
GPUResourceQuota status objects.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Dict, Optional

from ..api.store import NotFound, Store
from ..api.types import AllocRequest, GPUResourceQuota, Resource


class QuotaExceeded(Exception):
    pass


@dataclass
class _NsUsage:
    committed: Resource = field(default_factory=Resource)
    assumed: Resource = field(default_factory=Resource)
    workers_committed: int = 0
    workers_assumed: int = 0


class QuotaStore:
    def __init__(self, store: Optional[Store] = None):
        self._lock = threading.RLock()
        self._usage: Dict[str, _NsUsage] = {}
        self._store = store
        self._dirty: set = set()
        if store:
            store.on_change("GPUResourceQuota", self._on_quota_event)

    # ------------------------------------------------------------ internal

    def _quota(self, ns: str) -> Optional[GPUResourceQuota]:
        if not self._store:
            return None
        q = self._store.try_get("GPUResourceQuota", ns, ns) or \
            self._store.try_get("GPUResourceQuota", "default", ns)
        if q is not None:
            return q
        # the reference keys quotas by NAMESPACE (informer, any name):
        # fall back to the first quota CR living in the namespace so an
        # unconventionally-named object still applies
        lst = self._store.list("GPUResourceQuota", namespace=ns)
        return lst[0] if lst else None

    def _on_quota_event(self, event: str, obj):
        # spec changes need no local recompute: usage is tracked locally,
        # limits are read on every check.
        pass

    def _u(self, ns: str) -> _NsUsage:
        return self._usage.setdefault(ns, _NsUsage())

    # ------------------------------------------------------------- checks

    def check(self, req: AllocRequest) -> None:
        """Raise QuotaExceeded if admitting `req` would break the namespace
        quota (total across committed+assumed, single-workload max, count)."""

        q = self._quota(req.namespace)
        if q is None:
            return
        per_gpu = req.request
        total_req = Resource(per_gpu.tflops * req.gpu_count,
                             per_gpu.vram * req.gpu_count,
                             per_gpu.compute_percent * req.gpu_count)
        sm = q.spec.single_max
        if (sm.tflops or sm.vram) and not per_gpu.fits_in(sm):
            raise QuotaExceeded(
                f"ns {req.namespace}: request exceeds single-workload max")
        with self._lock:
            u = self._u(req.namespace)
            used = u.committed.add(u.assumed).add(total_req)
            tot = q.spec.total
            if (tot.tflops or tot.vram) and not used.fits_in(tot):
                raise QuotaExceeded(
                    f"ns {req.namespace}: total quota exceeded "
                    f"(used {used.tflops:.0f}T/{used.vram >> 30}G of "
                    f"{tot.tflops:.0f}T/{tot.vram >> 30}G)")
            if q.spec.max_workers and \
                    u.workers_committed + u.workers_assumed + 1 > q.spec.max_workers:
                raise QuotaExceeded(f"ns {req.namespace}: max workers reached")

    # ----------------------------------------------------- two-phase hooks

    def _total(self, req: AllocRequest) -> Resource:
        r = req.request
        return Resource(r.tflops * req.gpu_count, r.vram * req.gpu_count,
                        r.compute_percent * req.gpu_count)

    def assume(self, req: AllocRequest) -> None:
        with self._lock:
            u = self._u(req.namespace)
            u.assumed = u.assumed.add(self._total(req))
            u.workers_assumed += 1

    def forget(self, req: AllocRequest) -> None:
        with self._lock:
            u = self._u(req.namespace)
            u.assumed = u.assumed.sub(self._total(req))
            u.workers_assumed -= 1
            if u.assumed.any_negative():
                u.assumed = Resource()
            u.workers_assumed = max(0, u.workers_assumed)

    def commit(self, req: AllocRequest) -> None:
        with self._lock:
            u = self._u(req.namespace)
            u.assumed = u.assumed.sub(self._total(req))
            u.workers_assumed = max(0, u.workers_assumed - 1)
            if u.assumed.any_negative():
                u.assumed = Resource()
            u.committed = u.committed.add(self._total(req))
            u.workers_committed += 1
            self._dirty.add(req.namespace)

    def release(self, req: AllocRequest) -> None:
        with self._lock:
            u = self._u(req.namespace)
            u.committed = u.committed.sub(self._total(req))
            u.workers_committed = max(0, u.workers_committed - 1)
            if u.committed.any_negative():
                u.committed = Resource()
            self._dirty.add(req.namespace)

    # -------------------------------------------------------------- status

    def usage(self, ns: str) -> Resource:
        with self._lock:
            u = self._u(ns)
            return u.committed.add(u.assumed)

    def sync_dirty(self) -> int:
        """Flush usage into GPUResourceQuota .status (dirty-queue pattern,
        reference quota_store.go:600). Returns number synced."""

        if not self._store:
            return 0
        with self._lock:
            dirty, self._dirty = self._dirty, set()
        n = 0
        for ns in dirty:
            with self._lock:
                u = self._u(ns)
                committed = Resource(u.committed.tflops, u.committed.vram,
                                     u.committed.compute_percent)
                workers = u.workers_committed
            for name in (ns, "default"):
                try:
                    def _p(obj, committed=committed, workers=workers):
                        obj.status.used = committed
                        obj.status.worker_count = workers
                    self._store.patch("GPUResourceQuota", name, ns, _p)
                    n += 1
                    break
                except NotFound:
                    continue
        return n
