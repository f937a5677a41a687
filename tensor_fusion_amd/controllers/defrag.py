"""Pool defragmentation / compaction — bin-pack consolidation.

Reference: internal/controller/gpupool_compaction_controller.go (17 kLoC)
+ gpupool_defrag.go (80 kLoC, the repo's largest file): campaign
scheduling (:275), candidate node selection (:547), joint placement
simulation of every evictee onto the remaining nodes (:1095), eviction
markers with TTL (:686-897). Compaction frees whole nodes (scale-in);
defrag migrates fragments so large requests fit.

Here: one DefragController drives both — pick the least-utilized
candidate nodes, simulate placing all their allocations on the remaining
GPUs with a scratch allocator, and when the whole set fits, mark the
victim pods with an eviction annotation + TTL. Eviction = pod delete;
the workload controller recreates replicas and the scheduler packs them
onto the remaining nodes (live-migration does the same dance through
snapshot/resume for running state).
"""
from __future__ import annotations

import copy
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from .. import constants as C
from ..api.store import Store
from ..api.types import AllocRequest, Resource

AnnoEvictionMark = f"{C.Domain}/evict-at"  # unix ts after which evictable
AnnoEvictionReason = f"{C.Domain}/evict-reason"


@dataclass
class DefragPlan:
    """One campaign: evict these pods; their allocations fit on the rest."""

    candidate_nodes: List[str]
    evict_pods: List[str]  # pod keys
    placements: Dict[str, List[str]] = field(default_factory=dict)
    freed: Resource = field(default_factory=Resource)


class DefragController:
    def __init__(self, store: Store, allocator,
                 utilization_threshold: float = 0.3,
                 eviction_ttl_s: float = 60.0,
                 campaign_cooldown_s: float = 300.0,
                 migrate_fn=None):
        self.store = store
        self.allocator = allocator
        self.utilization_threshold = utilization_threshold
        self.eviction_ttl_s = eviction_ttl_s
        self.campaign_cooldown_s = campaign_cooldown_s
        # migrate_fn(pod_key, target_gpus) → bool: live migration via the
        # vGPU worker snapshot machinery (hypervisor/vgpu_manager). When
        # set, defrag MOVES workers instead of evict+reschedule — the
        # workload never restarts.
        self.migrate_fn = migrate_fn
        self._last_campaign = 0.0
        self.migrated: List[str] = []

    # ------------------------------------------------------ candidates

    def node_utilization(self) -> Dict[str, float]:
        """Max of tflops/vram utilization per node over its GPUs."""

        util: Dict[str, Tuple[Resource, Resource]] = {}
        for g in self.allocator.gpus():
            s = g.status
            cap, avail = util.get(s.node, (Resource(), Resource()))
            util[s.node] = (cap.add(s.capacity), avail.add(s.available))
        out = {}
        for node, (cap, avail) in util.items():
            t = 1.0 - (avail.tflops / cap.tflops) if cap.tflops else 0.0
            v = 1.0 - (avail.vram / cap.vram) if cap.vram else 0.0
            out[node] = max(t, v)
        return out

    def pick_candidates(self) -> List[str]:
        """Least-utilized non-empty nodes under the threshold (reference
        :547 candidate selection)."""

        util = self.node_utilization()
        cands = [n for n, u in util.items() if 0.0 < u < self.utilization_threshold]
        return sorted(cands, key=lambda n: util[n])

    # ------------------------------------------------------ simulation

    def simulate(self, candidates: List[str]) -> Optional[DefragPlan]:
        """Joint placement simulation (reference :1095): can every
        allocation currently on `candidates` be placed on the other nodes
        simultaneously? Uses a scratch copy of the allocator state."""

        scratch = self.allocator.fork_for_simulation(exclude_nodes=candidates)
        moves: List[Tuple[str, AllocRequest]] = []
        for pod_key, alloc in self.allocator.allocations_on(candidates):
            moves.append((pod_key, alloc.req))
        if not moves:
            return None
        # place big ones first (standard bin-pack heuristic)
        moves.sort(key=lambda m: (m[1].request.vram, m[1].request.tflops),
                   reverse=True)
        plan = DefragPlan(candidate_nodes=list(candidates), evict_pods=[])
        for pod_key, req in moves:
            req2 = copy.deepcopy(req)
            req2.pod_name = req.pod_name + "-defrag-sim"
            try:
                node_scores, _reasons = scratch.check_quota_and_filter(req2)
            except Exception:
                return None
            placed = False
            ranked = sorted(node_scores.values(), key=lambda s: s.score,
                            reverse=True)
            for ns_ in ranked:
                node = ns_.node
                try:
                    gpus = scratch.pick_gpus(req2, node)
                    scratch.assume(req2, gpus)
                    scratch.commit(req2.pod_key)
                    plan.placements[pod_key] = gpus
                    placed = True
                    break
                except Exception:
                    continue
            if not placed:
                return None
            plan.evict_pods.append(pod_key)
            plan.freed = plan.freed.add(req.request)
        return plan

    # --------------------------------------------------------- campaign

    def run_campaign(self, now: Optional[float] = None) -> Optional[DefragPlan]:
        """One defrag campaign: candidates → simulate → mark evictions.
        Returns the executed plan (or None)."""

        now = now if now is not None else time.time()
        if now - self._last_campaign < self.campaign_cooldown_s:
            return None
        cands = self.pick_candidates()
        # consider prefixes: free as many nodes as jointly fit
        for k in range(len(cands), 0, -1):
            plan = self.simulate(cands[:k])
            if plan is not None:
                if self.migrate_fn is not None:
                    self._execute_migrations(plan)
                else:
                    self._mark_evictions(plan, now)
                self._last_campaign = now
                return plan
        return None

    def _execute_migrations(self, plan: DefragPlan):
        """Live-migrate each planned worker to its simulated placement;
        fall back to eviction marking for any that fail."""

        import time as _time
        failed = []
        for pod_key in plan.evict_pods:
            target = plan.placements.get(pod_key, [])
            try:
                ok = self.migrate_fn(pod_key, target)
            except Exception:
                ok = False
            if ok:
                self.migrated.append(pod_key)
                # move the allocation record onto the new devices
                alloc = self.allocator.allocation(pod_key)
                if alloc is not None:
                    self.allocator.dealloc(pod_key)
                    try:
                        self.allocator.assume(alloc.req, target)
                        self.allocator.commit(pod_key)
                        self.allocator.notify_bound(pod_key)
                    except Exception:
                        pass
            else:
                failed.append(pod_key)
        if failed:
            sub = DefragPlan(candidate_nodes=plan.candidate_nodes,
                             evict_pods=failed,
                             placements=plan.placements)
            self._mark_evictions(sub, _time.time())

    def _mark_evictions(self, plan: DefragPlan, now: float):
        """Eviction markers with TTL (reference :686-897): pods are only
        deleted after the TTL passes AND the mark is still present."""

        for pod_key in plan.evict_pods:
            ns, name = pod_key.split("/", 1)

            def _p(obj):
                obj.meta.annotations[AnnoEvictionMark] = str(
                    now + self.eviction_ttl_s)
                obj.meta.annotations[AnnoEvictionReason] = "defrag"
            try:
                self.store.patch("Pod", name, ns, _p)
            except Exception:
                continue

    def execute_due_evictions(self, now: Optional[float] = None) -> List[str]:
        """Delete pods whose eviction TTL expired (the grace window lets
        operators cancel by removing the annotation).

        Before deleting, the plan is RE-VALIDATED: cluster state may have
        moved during the TTL window (new pods landed on the target nodes)
        and an eviction whose allocation no longer fits elsewhere would
        strand the workload — the reference re-checks placements when the
        campaign executes (gpupool_defrag.go eviction path). A pod that
        fails revalidation keeps running and its mark is cleared."""

        now = now if now is not None else time.time()
        evicted = []
        for pod in self.store.list("Pod"):
            mark = pod.meta.annotations.get(AnnoEvictionMark)
            if not mark:
                continue
            if float(mark) > now:
                continue
            pod_key = pod.meta.key
            alloc = self.allocator.allocation(pod_key)
            if alloc is not None and not self._still_placeable(pod_key,
                                                               alloc):
                self._clear_mark(pod)
                continue
            try:
                self.store.delete("Pod", pod.meta.name, pod.meta.namespace)
                evicted.append(pod_key)
            except Exception:
                continue
        return evicted

    def _still_placeable(self, pod_key: str, alloc) -> bool:
        """Would this allocation still fit somewhere off its current
        node if evicted right now?"""

        node = ""
        if alloc.gpu_names:
            g = self.allocator.gpu(alloc.gpu_names[0])
            if g is not None:
                node = g.status.node
        scratch = self.allocator.fork_for_simulation(
            exclude_nodes=[node] if node else [])
        req2 = copy.deepcopy(alloc.req)
        req2.pod_name = alloc.req.pod_name + "-reval"
        try:
            scores, _ = scratch.check_quota_and_filter(req2)
        except Exception:
            return False
        for ns_ in sorted(scores.values(), key=lambda s: s.score,
                          reverse=True):
            try:
                gpus = scratch.pick_gpus(req2, ns_.node)
                scratch.assume(req2, gpus)
                return True
            except Exception:
                continue
        return False

    def _clear_mark(self, pod):
        def _p(obj):
            obj.meta.annotations.pop(AnnoEvictionMark, None)
            obj.meta.annotations.pop(AnnoEvictionReason, None)
        try:
            self.store.patch("Pod", pod.meta.name, pod.meta.namespace, _p)
        except Exception:
            pass

    def abort_campaign(self) -> int:
        """Operator escape hatch: clear every pending eviction mark
        (reference campaigns are cancellable until execution)."""

        n = 0
        for pod in self.store.list("Pod"):
            if AnnoEvictionMark in pod.meta.annotations:
                self._clear_mark(pod)
                n += 1
        return n
