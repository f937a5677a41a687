"""GpuAllocator — the in-memory source of truth for device capacity.

Reference: internal/gpuallocator/gpuallocator.go (3.5 kLoC): gpuStore /
nodeGpuStore / poolGpuStore indexes, the scheduler-cache two-phase lifecycle
  CheckQuotaAndFilter → Assume → Commit → NotifyBound
                          ↘ Rollback/Forget
TTL sweep of stale assumed allocations (gang-aware), Dealloc,
AdjustAllocation (vertical scaling), preemption simulation, partition bind,
dirty-queue sync of GPU status back to the object store, and full state
reconcile from existing worker records after a restart.

Fresh implementation for MI355X: partitions are XCD slabs (partitioning.py),
topology scoring treats the intra-node xGMI mesh as flat (strategy.py).
"""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Set, Tuple

import numpy as np

from .. import constants as C
from ..api.store import NotFound, Store
from ..api.types import (GPU, AllocRequest, GPUPartition, PartitionTemplate,
                         Resource)
from ..quota.quota_store import QuotaExceeded, QuotaStore
from . import partitioning
from .filters import default_registry, make_same_node_filter
from .strategy import Strategy, make_strategy


class AllocationError(Exception):
    pass


@dataclass
class Allocation:
    """One pod's device binding (assumed, then committed)."""

    req: AllocRequest
    gpu_names: List[str]
    partition: Optional[GPUPartition] = None
    assumed_at: float = field(default_factory=time.time)
    committed: bool = False
    bound: bool = False


@dataclass(slots=True)
class NodeScore:  # slots: ~1000 of these are built per PreFilter call
    node: str
    score: float
    gpu_scores: Dict[str, float]  # gpu name -> score


class LazyNodeScores(dict):
    """node -> NodeScore, materialized on first read.

    The vectorized fast path scores EVERY feasible node, but a cycle
    only reads NodeScore objects for the ~numFeasibleNodes it samples —
    so store raw floats and wrap lazily; at 1k-node scale this skips
    ~90% of the dataclass constructions (the fast path's top cost)."""

    __slots__ = ()

    def __getitem__(self, k):
        v = dict.__getitem__(self, k)
        if type(v) is float:
            v = NodeScore(node=k, score=v, gpu_scores={})
            dict.__setitem__(self, k, v)
        return v

    def get(self, k, default=None):
        try:
            return self[k]
        except KeyError:
            return default

    def values(self):
        return [self[k] for k in dict.keys(self)]

    def items(self):
        return [(k, self[k]) for k in dict.keys(self)]


class GpuAllocator:
    ASSUME_TTL_S = 60.0

    def __init__(self, store: Optional[Store] = None,
                 quota: Optional[QuotaStore] = None,
                 strategy: Optional[Strategy] = None,
                 partition_templates: Optional[List[PartitionTemplate]] = None):
        self._mu = threading.RLock()
        self._store = store
        self.quota = quota or QuotaStore(store)
        self.strategy = strategy or make_strategy("NodeCompactGPULowLoad")
        self.partition_templates = partition_templates or []
        self._gpus: Dict[str, GPU] = {}          # name -> working copy
        self._node_gpus: Dict[str, Set[str]] = {}
        self._pool_gpus: Dict[str, Set[str]] = {}
        self._allocations: Dict[str, Allocation] = {}  # pod key -> alloc
        self._dirty: Set[str] = set()
        self._filters = default_registry()
        self._soa: Optional[dict] = None  # numpy mirror (lazy, see _soa_get)
        self._soa_rows: Dict[str, int] = {}
        if store:
            store.on_change("GPU", self._on_gpu_event)
            for g in store.list("GPU"):
                self._ingest(g)

    # ------------------------------------------------------------ ingest

    def _on_gpu_event(self, event: str, obj: GPU):
        with self._mu:
            if event == "DELETED":
                self._evict(obj.meta.name)
            else:
                # Never clobber local available/running_apps with stale CR
                # status: capacity/topology/phase come from the CR, usage is
                # owned locally (the reference's informer does the same).
                cur = self._gpus.get(obj.meta.name)
                if cur is None:
                    self._ingest(obj)
                else:
                    cur.status.phase = obj.status.phase
                    cur.status.used_by = obj.status.used_by
                    cur.status.capacity = obj.status.capacity
                    cur.status.topology = obj.status.topology
                    cur.status.isolation_mode = obj.status.isolation_mode
                    self._soa_sync(obj.meta.name)

    def _ingest(self, g: GPU):
        g = g.deepcopy()
        self._gpus[g.meta.name] = g
        self._node_gpus.setdefault(g.status.node, set()).add(g.meta.name)
        self._pool_gpus.setdefault(g.status.pool, set()).add(g.meta.name)
        self._soa_invalidate()

    def _evict(self, name: str):
        g = self._gpus.pop(name, None)
        if g:
            self._node_gpus.get(g.status.node, set()).discard(name)
            self._pool_gpus.get(g.status.pool, set()).discard(name)
        self._soa_invalidate()

    def upsert_gpu_for_testing(self, g: GPU):
        """Direct injection (reference UpsertGPUForTesting :421)."""

        with self._mu:
            self._ingest(g)

    # ------------------------------------------------------------ queries

    def gpu_uuid(self, name: str) -> str:
        """UUID without the deepcopy gpu() pays (PreBind hot path)."""

        with self._mu:
            g = self._gpus.get(name)
            return g.status.uuid if g else ""

    def gpu_capacity_tflops(self, name: str) -> float:
        with self._mu:
            g = self._gpus.get(name)
            return g.status.capacity.tflops if g else 0.0

    def gpu(self, name: str) -> Optional[GPU]:
        with self._mu:
            g = self._gpus.get(name)
            return g.deepcopy() if g else None

    def gpus(self, pool: Optional[str] = None, node: Optional[str] = None) -> List[GPU]:
        with self._mu:
            if node is not None:
                names = self._node_gpus.get(node, set())
            elif pool is not None:
                names = self._pool_gpus.get(pool, set())
            else:
                names = self._gpus.keys()
            return [self._gpus[n].deepcopy() for n in names if n in self._gpus]

    def allocation(self, pod_key: str) -> Optional[Allocation]:
        with self._mu:
            return self._allocations.get(pod_key)

    # ------------------------------------------------------ numpy mirror
    # Structure-of-arrays mirror of the device fleet: the per-pod filter +
    # score pass touches every GPU, and pure-Python loops capped the
    # scheduler at ~12 pods/s on 8k GPUs. Rebuilt on topology changes,
    # updated in place on assume/release.

    _ISO_CODES = {C.IsolationShared: 0, C.IsolationSoft: 1,
                  C.IsolationHard: 2, C.IsolationPartitioned: 3}

    def _soa_invalidate(self):
        self._soa = None

    def _soa_get(self) -> dict:
        if self._soa is not None:
            return self._soa
        names = list(self._gpus.keys())
        n = len(names)
        node_names: List[str] = []
        node_idx_of: Dict[str, int] = {}
        soa = {
            "names": names,
            "av_t": np.zeros(n), "av_v": np.zeros(n), "av_c": np.zeros(n),
            "cap_t": np.ones(n), "cap_v": np.ones(n),
            "ready": np.zeros(n, dtype=bool),
            "n_apps": np.zeros(n, dtype=np.int32),
            "n_parts": np.zeros(n, dtype=np.int32),
            "iso": np.zeros(n, dtype=np.int8),
            "node_id": np.zeros(n, dtype=np.int32),
            "node_names": node_names,
        }
        self._soa_rows = {}
        for i, name in enumerate(names):
            self._soa_rows[name] = i
            g = self._gpus[name]
            s = g.status
            nid = node_idx_of.setdefault(s.node, len(node_names))
            if nid == len(node_names):
                node_names.append(s.node)
            soa["node_id"][i] = nid
            self._soa_fill_row(soa, i, g)
        self._soa = soa
        return soa

    def _soa_fill_row(self, soa: dict, i: int, g: GPU):
        s = g.status
        soa["av_t"][i] = s.available.tflops
        soa["av_v"][i] = s.available.vram
        soa["av_c"][i] = s.available.compute_percent
        soa["cap_t"][i] = s.capacity.tflops or 1.0
        soa["cap_v"][i] = s.capacity.vram or 1.0
        soa["ready"][i] = (s.phase == "Ready"
                           and s.used_by == "tensor-fusion")
        soa["n_apps"][i] = len(s.running_apps)
        soa["n_parts"][i] = len(s.allocated_partitions)
        soa["iso"][i] = self._ISO_CODES.get(s.isolation_mode, 1)

    def _soa_sync(self, name: str):
        if self._soa is None:
            return
        i = self._soa_rows.get(name)
        g = self._gpus.get(name)
        if i is None or g is None:
            self._soa = None
            return
        self._soa_fill_row(self._soa, i, g)

    def _fast_filter_score(self, req: AllocRequest
                           ) -> Optional[Tuple[Dict[str, NodeScore],
                                               Dict[str, str]]]:
        """Vectorized filter+score for the common request shape. Returns
        None when the request needs the full chain."""

        if (req.partitioned or req.gpu_model or req.gpu_vendor
                or req.gpu_indices or req.node_affinity or req.pool
                or type(self.strategy).__name__ not in (
                    "NodeCompactGPULowLoad", "CompactFirst", "LowLoadFirst")):
            return None
        soa = self._soa_get()
        if not soa["names"]:
            return {}, {"resource": "no GPUs"}
        r = req.request
        fits = ((soa["av_t"] >= r.tflops - 1e-9)
                & (soa["av_v"] >= r.vram)
                & (soa["av_c"] >= r.compute_percent - 1e-9))
        empty = (soa["n_apps"] == 0) & (soa["n_parts"] == 0)
        if req.isolation_mode == C.IsolationShared:
            # shared = whole-GPU: mirror shared_whole_gpu_filter exactly —
            # only fully-free devices (no apps, all VRAM available) qualify,
            # else the fast path ranks infeasible nodes that Reserve then
            # rejects (retry churn / unschedulable shared pods)
            iso_ok = empty & (soa["av_v"] >= soa["cap_v"] - 0.5)
        else:
            iso_ok = empty | ((soa["iso"] == self._ISO_CODES.get(
                req.isolation_mode, 1)) & (soa["n_parts"] == 0))
        ok = soa["ready"] & fits & iso_ok
        if not ok.any():
            reasons = {}
            if not soa["ready"].any():
                reasons["phase"] = "no GPU in Ready phase"
            elif not (soa["ready"] & fits).any():
                reasons["resource"] = (
                    f"insufficient resources (need {r.tflops:.0f} tflops / "
                    f"{r.vram >> 30} GiB)")
            else:
                reasons["isolation"] = \
                    f"isolation mode {req.isolation_mode} conflicts"
            return {}, reasons
        idx = np.nonzero(ok)[0]
        nid = soa["node_id"][idx]
        # per-node eligible count (same-node constraint for gpu_count>1)
        counts = np.bincount(nid, minlength=len(soa["node_names"]))
        good_nodes = np.nonzero(counts >= req.gpu_count)[0]
        if len(good_nodes) == 0:
            return {}, {"same_node":
                        f"no node with {req.gpu_count} eligible GPUs"}
        good_mask = np.isin(nid, good_nodes)
        idx = idx[good_mask]
        nid = nid[good_mask]
        # scores
        sname = type(self.strategy).__name__
        nn = len(soa["node_names"])
        if sname == "NodeCompactGPULowLoad":
            # node score is pure utilization — skip the per-GPU affinity
            # score entirely (it is only consumed by the other branches)
            usage = (0.5 * (1.0 - soa["av_t"][idx] / soa["cap_t"][idx])
                     + 0.5 * (1.0 - soa["av_v"][idx] / soa["cap_v"][idx]))
            sums = np.bincount(nid, weights=usage, minlength=nn)
            cnts = np.bincount(nid, minlength=nn)
            with np.errstate(invalid="ignore"):
                node_score = np.where(cnts > 0, 100.0 * sums /
                                      np.maximum(cnts, 1), 0.0)
        else:
            tw = self.strategy.tflops_weight
            vw = self.strategy.vram_weight
            af = (tw * (1.0 - (soa["av_t"][idx] - r.tflops)
                        / soa["cap_t"][idx])
                  + vw * (1.0 - (soa["av_v"][idx] - r.vram)
                          / soa["cap_v"][idx]))
            af = np.clip(af, 0.0, 1.0)
            gpu_score = 100.0 * af if sname == "CompactFirst" \
                else 100.0 * (1.0 - af)
            # mean of top gpu_count scores per node; k==1 → per-node max
            if req.gpu_count == 1:
                node_score = np.full(nn, -1.0)
                np.maximum.at(node_score, nid, gpu_score)
            else:
                order = np.lexsort((-gpu_score, nid))
                node_score = np.zeros(nn)
                taken = np.zeros(nn, dtype=np.int32)
                for j in order:
                    b = nid[j]
                    if taken[b] < req.gpu_count:
                        node_score[b] += gpu_score[j]
                        taken[b] += 1
                node_score = np.where(taken > 0,
                                      node_score / np.maximum(taken, 1), 0.0)
        nnames = soa["node_names"]
        vals = node_score[good_nodes].tolist()
        keys = [nnames[b] for b in good_nodes.tolist()]
        return LazyNodeScores(zip(keys, vals)), {}

    def eligible_gpu_names(self, req: AllocRequest, node: str) -> List[str]:
        """Names of devices on `node` passing the filter chain for `req`
        (gputopo's combo source when PreFilter left gpu_scores lazy)."""

        with self._mu:
            names = [n for n in self._node_gpus.get(node, set())
                     if n in self._gpus]
            eligible, _ = self._filters.apply(
                req, [self._gpus[n] for n in names])
            return [g.meta.name for g in eligible]

    # --------------------------------------------- CheckQuotaAndFilter

    def check_quota_and_filter(self, req: AllocRequest
                               ) -> Tuple[Dict[str, NodeScore], Dict[str, str]]:
        """Quota check + filter chain + per-node scoring.

        Returns ({node: NodeScore}, failure_reasons). Empty dict = pod is
        unschedulable with the reasons explaining why (reference
        CheckQuotaAndFilter :1653 feeding PreFilter's CycleState).
        """

        self.quota.check(req)  # raises QuotaExceeded
        with self._mu:
            fast = self._fast_filter_score(req)
            if fast is not None:
                return fast
            pool_names = self._pool_gpus.get(req.pool) if req.pool else None
            cands = [self._gpus[n] for n in (pool_names if pool_names is not None
                                             else self._gpus.keys())
                     if n in self._gpus]
            filters = self._filters
            if req.gpu_count > 1:
                filters = filters.with_filters(make_same_node_filter(req.gpu_count))
            eligible, reasons = filters.apply(req, cands)
            if req.partitioned:
                eligible = [g for g in eligible
                            if partitioning.place_partition(g, req, self.partition_templates)]
                if not eligible:
                    reasons["partition"] = "no device with a free partition slot"
            by_node: Dict[str, List[GPU]] = {}
            for g in eligible:
                by_node.setdefault(g.status.node, []).append(g)
            out: Dict[str, NodeScore] = {}
            for node, lst in by_node.items():
                if len(lst) < req.gpu_count:
                    continue
                # gpu_scores stay lazy: pick_gpus re-scores only the chosen
                # node at Reserve (eager per-GPU dicts dominated PreFilter)
                out[node] = NodeScore(
                    node=node,
                    score=self.strategy.score_node(lst, req),
                    gpu_scores={},
                )
            return out, reasons

    # ------------------------------------------------------------ Assume

    def pick_gpus(self, req: AllocRequest, node: str,
                  gpu_scores: Optional[Dict[str, float]] = None) -> List[str]:
        """Top-N device pick on the chosen node (Reserve-time)."""

        with self._mu:
            names = [n for n in self._node_gpus.get(node, set()) if n in self._gpus]
            eligible, _ = self._filters.apply(req, [self._gpus[n] for n in names])
            if len(eligible) < req.gpu_count:
                raise AllocationError(
                    f"node {node}: only {len(eligible)} eligible GPUs for "
                    f"count {req.gpu_count}")
            scored = sorted(
                eligible,
                key=lambda g: (gpu_scores or {}).get(
                    g.meta.name, self.strategy.score_gpu(g, req)),
                reverse=True)
            return [g.meta.name for g in scored[:req.gpu_count]]

    def assume(self, req: AllocRequest, gpu_names: List[str]) -> Allocation:
        """Atomically apply the allocation to the in-memory device copies and
        the quota overlay (reference Assume :1191)."""

        with self._mu:
            key = req.pod_key
            if key in self._allocations:
                raise AllocationError(f"{key} already assumed")
            gpus = []
            for n in gpu_names:
                g = self._gpus.get(n)
                if g is None:
                    raise AllocationError(f"unknown GPU {n}")
                if not req.request.fits_in(g.status.available):
                    raise AllocationError(f"GPU {n} no longer fits request")
                gpus.append(g)
            partition = None
            if req.partitioned:
                pl = partitioning.place_partition(gpus[0], req,
                                                  self.partition_templates)
                if pl is None:
                    raise AllocationError("no partition slot")
                partition = pl.to_partition(
                    req, partition_id=f"{gpus[0].meta.name}-{pl.template.id}-"
                    f"{pl.xcds[0]}")
            for g in gpus:
                av = g.status.available
                g.status.available = av.sub(req.request)
                if req.workload and req.workload not in g.status.running_apps:
                    g.status.running_apps.append(req.workload)
                if g.status.isolation_mode != req.isolation_mode and \
                        len(g.status.running_apps) <= 1:
                    g.status.isolation_mode = req.isolation_mode
                if partition is not None and g is gpus[0]:
                    g.status.allocated_partitions.append(partition)
                self._dirty.add(g.meta.name)
                self._soa_sync(g.meta.name)
            self.quota.assume(req)
            alloc = Allocation(req=req, gpu_names=list(gpu_names),
                               partition=partition)
            self._allocations[key] = alloc
            return alloc

    # ------------------------------------------------------- simulation

    def fork_for_simulation(self, exclude_nodes: Optional[List[str]] = None
                            ) -> "GpuAllocator":
        """Detached deep copy of the device state for what-if placement
        (defrag joint simulation, preemption dry-runs). No store, no quota
        backing — mutations never touch the live allocator."""

        import copy as _copy
        with self._mu:
            sim = GpuAllocator(store=None, quota=QuotaStore(None),
                               strategy=self.strategy,
                               partition_templates=self.partition_templates)
            excl = set(exclude_nodes or [])
            for name, g in self._gpus.items():
                if g.status.node in excl:
                    continue
                sim._ingest(g)
            sim._allocations = {k: _copy.deepcopy(a)
                                for k, a in self._allocations.items()
                                if not any(self._gpus[n].status.node in excl
                                           for n in a.gpu_names
                                           if n in self._gpus)}
            return sim

    def allocations_on(self, nodes: List[str]
                       ) -> List[Tuple[str, "Allocation"]]:
        """Committed allocations whose devices live on the given nodes."""

        with self._mu:
            out = []
            nodeset = set(nodes)
            for key, alloc in self._allocations.items():
                if any(self._gpus.get(n) is not None
                       and self._gpus[n].status.node in nodeset
                       for n in alloc.gpu_names):
                    out.append((key, alloc))
            return out

    # --------------------------------------------------- Commit / Rollback

    def commit(self, pod_key: str) -> Allocation:
        with self._mu:
            alloc = self._allocations.get(pod_key)
            if alloc is None:
                raise AllocationError(f"{pod_key}: nothing assumed")
            if not alloc.committed:
                alloc.committed = True
                self.quota.commit(alloc.req)
            return alloc

    def rollback(self, pod_key: str) -> None:
        """Undo an assumed (or failed-to-bind committed) allocation."""

        with self._mu:
            alloc = self._allocations.pop(pod_key, None)
            if alloc is None:
                return
            self._release_devices(alloc)
            if alloc.committed:
                self.quota.release(alloc.req)
            else:
                self.quota.forget(alloc.req)

    forget = rollback  # scheduler Unreserve naming

    def notify_bound(self, pod_key: str) -> None:
        with self._mu:
            alloc = self._allocations.get(pod_key)
            if alloc:
                alloc.bound = True

    # ------------------------------------------------------------ Dealloc

    def _release_devices(self, alloc: Allocation):
        for n in alloc.gpu_names:
            g = self._gpus.get(n)
            if g is None:
                continue
            g.status.available = g.status.available.add(alloc.req.request)
            cap = g.status.capacity
            g.status.available = Resource(
                min(g.status.available.tflops, cap.tflops),
                min(g.status.available.vram, cap.vram),
                min(g.status.available.compute_percent, cap.compute_percent))
            other = any(a is not alloc and n in a.gpu_names and
                        a.req.workload == alloc.req.workload
                        for a in self._allocations.values())
            if alloc.req.workload in g.status.running_apps and not other:
                g.status.running_apps.remove(alloc.req.workload)
            if alloc.partition is not None:
                g.status.allocated_partitions = [
                    p for p in g.status.allocated_partitions
                    if p.partition_id != alloc.partition.partition_id]
            self._dirty.add(n)
            self._soa_sync(n)

    def dealloc(self, pod_key: str) -> None:
        """Release a committed allocation (pod deleted/failed;
        reference Dealloc :1730)."""

        with self._mu:
            alloc = self._allocations.pop(pod_key, None)
            if alloc is None:
                return
            self._release_devices(alloc)
            if alloc.committed:
                self.quota.release(alloc.req)
            else:
                self.quota.forget(alloc.req)

    # -------------------------------------------------- vertical scaling

    def adjust_allocation(self, pod_key: str, new_request: Resource,
                          new_limit: Optional[Resource] = None) -> None:
        """In-place resize (autoscaler recommendation; reference
        AdjustAllocation :1864)."""

        with self._mu:
            alloc = self._allocations.get(pod_key)
            if alloc is None:
                raise AllocationError(f"{pod_key}: not allocated")
            delta = new_request.sub(alloc.req.request)
            for n in alloc.gpu_names:
                g = self._gpus.get(n)
                if g is None:
                    continue
                if not delta.fits_in(g.status.available):
                    raise AllocationError(f"GPU {n}: cannot grow by delta")
            old_req = alloc.req
            if alloc.committed:
                self.quota.release(old_req)
            import copy as _c
            alloc.req = _c.deepcopy(old_req)
            alloc.req.request = new_request
            if new_limit is not None:
                alloc.req.limit = new_limit
            if alloc.committed:
                self.quota.commit(alloc.req)
            for n in alloc.gpu_names:
                g = self._gpus.get(n)
                if g is None:
                    continue
                g.status.available = g.status.available.sub(delta)
                self._dirty.add(n)
                self._soa_sync(n)

    # ---------------------------------------------------- preemption sim

    def filter_with_preempt(self, req: AllocRequest
                            ) -> Optional[Tuple[str, List[str]]]:
        """Can `req` fit on some node if lower-QoS pods were evicted?
        Returns (node, victim pod keys) or None (reference
        FilterWithPreempt :743-995 — simulation only; eviction is the
        scheduler's PostFilter's job)."""

        order = {q: i for i, q in enumerate(C.QosLevels)}
        rq = order.get(req.qos, 1)
        with self._mu:
            for node, names in self._node_gpus.items():
                victims: List[str] = []
                freed: Dict[str, Resource] = {n: Resource() for n in names}
                # lowest QoS first, newest first
                cand = sorted(
                    (a for a in self._allocations.values()
                     if order.get(a.req.qos, 1) < rq
                     and any(n in names for n in a.gpu_names)),
                    key=lambda a: (order.get(a.req.qos, 1), -a.assumed_at))
                for a in cand:
                    victims.append(a.req.pod_key)
                    for n in a.gpu_names:
                        if n in freed:
                            freed[n] = freed[n].add(a.req.request)
                    fit = sum(
                        1 for n in names if n in self._gpus and req.request.fits_in(
                            self._gpus[n].status.available.add(freed[n])))
                    if fit >= req.gpu_count:
                        return node, victims
        return None

    # ------------------------------------------------------- maintenance

    def sweep_stale_assumed(self, gang_active: Optional[Set[str]] = None) -> List[str]:
        """Drop assumed-but-never-committed allocations past TTL, unless the
        pod's gang group is still actively waiting (reference :1563 +
        gang-aware probe :417)."""

        now = time.time()
        dropped = []
        with self._mu:
            for key, a in list(self._allocations.items()):
                if a.committed:
                    continue
                if now - a.assumed_at < self.ASSUME_TTL_S:
                    continue
                if gang_active and a.req.gang_group in gang_active:
                    continue
                self._allocations.pop(key)
                self._release_devices(a)
                self.quota.forget(a.req)
                dropped.append(key)
        return dropped

    def sync_dirty(self) -> int:
        """Flush local device state into the object store (dirty-queue,
        reference :2157-2621)."""

        if not self._store:
            return 0
        with self._mu:
            dirty, self._dirty = self._dirty, set()
            snaps = {n: self._gpus[n].deepcopy() for n in dirty if n in self._gpus}
        n_synced = 0
        for name, snap in snaps.items():
            try:
                def _p(obj, snap=snap):
                    obj.status.available = snap.status.available
                    obj.status.running_apps = snap.status.running_apps
                    obj.status.allocated_partitions = snap.status.allocated_partitions
                    obj.status.isolation_mode = snap.status.isolation_mode
                self._store.patch("GPU", name, "", _p)
                n_synced += 1
            except NotFound:
                continue
        self.quota.sync_dirty()
        return n_synced

    def reconcile_from_allocations(self, records: List[Tuple[AllocRequest, List[str]]]):
        """Rebuild committed state after an operator restart from worker pod
        annotations (reference reconcileAllocationState :2906)."""

        with self._mu:
            self._soa_invalidate()  # bulk reset below bypasses row sync
            # Rebuild from scratch: reset devices to capacity, then reapply
            # every record (the store's synced status already reflects the
            # old allocations — replaying on top would double-subtract).
            for g in self._gpus.values():
                cap = g.status.capacity
                g.status.available = Resource(cap.tflops, cap.vram,
                                              cap.compute_percent)
                g.status.running_apps = []
                g.status.allocated_partitions = []
                self._dirty.add(g.meta.name)
            for req, gpu_names in records:
                if req.pod_key in self._allocations:
                    continue
                try:
                    self.assume(req, gpu_names)
                    self.commit(req.pod_key)
                    self.notify_bound(req.pod_key)
                except (AllocationError, QuotaExceeded):
                    continue
