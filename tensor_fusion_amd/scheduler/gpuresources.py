"""GPUResourcesFit — the main scheduler plugin.

Reference: internal/scheduler/gpuresources/gpuresources.go — PreEnqueue
(gang gate), PreFilter (AllocRequest → CheckQuotaAndFilter → per-node scores
into CycleState), Filter, Score, Reserve (top-N GPU pick + Assume),
PostFilter (preemption), Permit (gang wait), PreBind (Commit + pod
annotation patch with rollback), PostBind, Unreserve.
"""
from __future__ import annotations

import threading
from typing import List, Optional, Tuple

from .. import constants as C
from ..allocator.gpuallocator import AllocationError, GpuAllocator
from ..allocator.partitioning import cu_mask_for_percent
from ..api.store import NotFound, Store
from ..api.types import Pod, WorkloadProfile
from ..gang.manager import GangManager
from ..quota.quota_store import QuotaExceeded
from ..utils.resource import compose_allocation_request, profile_from_annotations
from .framework import Code, CycleState, Plugin, Scheduler, Status

S_REQ = "gpufit/alloc_request"
S_SCORES = "gpufit/node_scores"
S_PICKED = "gpufit/picked_gpus"
S_TOPO = "gputopo/combos"


class GPUResourcesFit(Plugin):
    name = "GPUResourcesFit"

    def __init__(self, store: Store, allocator: GpuAllocator,
                 gang: Optional[GangManager] = None,
                 index_allocator=None, port_allocator=None,
                 expander=None, metrics=None):
        self.store = store
        self.allocator = allocator
        self.gang = gang
        self.index_allocator = index_allocator
        self.port_allocator = port_allocator
        self.expander = expander  # NodeExpander: auto-provision on no-fit
        self.metrics = metrics
        self._mu = threading.Lock()

    # -------------------------------------------------------------- hooks

    def pre_enqueue(self, pod: Pod) -> Status:
        if self.gang:
            reason = self.gang.pre_enqueue(pod)
            if reason:
                return Status.unschedulable(reason)
        return Status.ok_()

    def _compose(self, pod: Pod):
        base: Optional[WorkloadProfile] = None
        pname = pod.meta.annotations.get(C.AnnoWorkloadProfile)
        if pname:
            got = self.store.try_get("WorkloadProfile", pname,
                                     pod.meta.namespace)
            if got:
                base = got
        profile = profile_from_annotations(pod, base)
        pool = pod.meta.labels.get(C.LabelPool, profile.pool)
        return compose_allocation_request(pod, profile, pool)

    def pre_filter(self, state: CycleState, pod: Pod
                   ) -> Tuple[Optional[List[str]], Status]:
        try:
            req = self._compose(pod)
        except (ValueError, KeyError) as e:
            return None, Status.error(f"bad annotations: {e}")
        state[S_REQ] = req
        try:
            scores, reasons = self.allocator.check_quota_and_filter(req)
        except QuotaExceeded as e:
            return None, Status.unschedulable(str(e))
        state[S_SCORES] = scores
        if not scores:
            if self.metrics:
                self.metrics.record_unschedulable()
            if self.expander is not None:
                claim = self.expander.handle_unschedulable(req)
                if claim:
                    reasons = dict(reasons)
                    reasons["expander"] = f"node claim {claim} in flight"
            msg = "; ".join(f"{k}: {v}" for k, v in reasons.items()) or \
                "no eligible GPU"
            return None, Status.unschedulable(msg)
        return list(scores.keys()), Status.ok_()

    def filter(self, state: CycleState, pod: Pod, node: str) -> Status:
        scores = state.get(S_SCORES, {})
        if node not in scores:
            return Status.unschedulable("no eligible GPUs on node")
        return Status.ok_()

    def post_filter(self, state: CycleState, pod: Pod,
                    filtered_nodes: List[str]) -> Status:
        """Preemption: find victims, evict them (delete their pods), requeue
        (reference PostFilter :744)."""

        req = state.get(S_REQ)
        if req is None or req.qos in (C.QosLow,):
            return Status.unschedulable()
        got = self.allocator.filter_with_preempt(req)
        if got is None:
            return Status.unschedulable()
        node, victims = got
        for vk in victims:
            ns, name = vk.split("/", 1)
            try:
                self.store.patch("Pod", name, ns, _mark_evicted)
            except NotFound:
                pass
            self.allocator.dealloc(vk)
        if self.metrics:
            self.metrics.record_preempted()
        return Status(Code.Success,
                      [f"evicted {len(victims)} lower-QoS pods on {node}"])

    def score(self, state: CycleState, pod: Pod, node: str) -> float:
        scores = state.get(S_SCORES, {})
        ns = scores.get(node)
        return ns.score if ns else 0.0

    def reserve(self, state: CycleState, pod: Pod, node: str) -> Status:
        req = state[S_REQ]
        scores = state.get(S_SCORES, {})
        combo = state.get(S_TOPO, {}).get(node)
        try:
            if combo:
                picked = combo[:req.gpu_count]
            else:
                gpu_scores = scores[node].gpu_scores if node in scores else None
                picked = self.allocator.pick_gpus(req, node, gpu_scores)
            self.allocator.assume(req, picked)
        except (AllocationError, KeyError) as e:
            return Status.unschedulable(f"reserve: {e}")
        state[S_PICKED] = picked
        return Status.ok_()

    def unreserve(self, state: CycleState, pod: Pod, node: str) -> None:
        req = state.get(S_REQ)
        if req is not None:
            self.allocator.rollback(req.pod_key)

    def permit(self, state: CycleState, pod: Pod, node: str
               ) -> Tuple[Status, float]:
        if not self.gang:
            return Status.ok_(), 0.0
        timeout = self.gang.permit(pod)
        if timeout is None:
            return Status.ok_(), 0.0
        return Status.wait(f"gang {self.gang.group_key_of(pod)} quorum"), timeout

    # waiting-pool integration (called by the framework)
    def on_permit_allowed(self, sched: Scheduler, pod: Pod) -> None:
        if not self.gang:
            return
        key = self.gang.group_key_of(pod)
        if key and self.gang.quorum_met(key):
            self._release_group(sched, key)

    def on_pod_waiting(self, sched: Scheduler, wp) -> None:
        if not self.gang:
            return
        key = self.gang.group_key_of(wp.pod)
        if key and self.gang.quorum_met(key):
            self._release_group(sched, key)

    def on_pod_rejected(self, sched: Scheduler, pod: Pod) -> None:
        """A waiting gang member timed out: reject the whole group."""

        if not self.gang:
            return
        key = self.gang.group_key_of(pod)
        if not key:
            return
        self.gang.reject_group(key)
        for wp in sched.waiting_pods():
            if self.gang.group_key_of(wp.pod) == key:
                wp.reject()

    def _release_group(self, sched: Scheduler, key: str):
        for wp in sched.waiting_pods():
            if self.gang.group_key_of(wp.pod) == key:
                wp.allow()

    def pre_bind(self, state: CycleState, pod: Pod, node: str) -> Status:
        req = state[S_REQ]
        try:
            alloc = self.allocator.commit(req.pod_key)
        except AllocationError as e:
            return Status.error(str(e))
        gpu_uuids = [self.allocator.gpu_uuid(n) or n
                     for n in alloc.gpu_names]
        annos = {
            C.AnnoGpuIds: ",".join(gpu_uuids),
            C.AnnoContainerGpus: ",".join(alloc.gpu_names),
        }
        if self.index_allocator is not None:
            idx = self.index_allocator.occupy(node, pod.meta.key)
            annos[C.AnnoPodIndex] = str(idx)
        if req.isolation_mode == C.IsolationHard:
            cap_tf = self.allocator.gpu_capacity_tflops(
                alloc.gpu_names[0]) or C.MI355X_BF16_TFLOPS
            pct = req.limit.compute_percent or \
                (100.0 * req.limit.tflops / cap_tf if cap_tf else 100.0)
            mask, cus = cu_mask_for_percent(max(pct, 0.5))
            annos[C.AnnoEffectiveHardCuPercent] = f"{100.0 * cus / 256:.2f}"
        if alloc.partition is not None:
            annos[C.AnnoPartitionId] = alloc.partition.partition_id
        try:
            def _p(obj):
                obj.meta.annotations.update(annos)
            self.store.patch("Pod", pod.meta.name, pod.meta.namespace, _p)
        except Exception as e:
            # rollback path: the framework calls unreserve on failure
            return Status.error(f"annotation patch: {e}")
        return Status.ok_()

    def post_bind(self, state: CycleState, pod: Pod, node: str) -> None:
        req = state[S_REQ]
        self.allocator.notify_bound(req.pod_key)
        if self.expander is not None:
            self.expander.forget_pod(req.pod_key)
        if self.metrics:
            self.metrics.record_scheduled(0.0)
        if self.gang:
            self.gang.mark_scheduled(pod)


def _mark_evicted(obj):
    obj.meta.annotations[f"{C.Domain}/evicted"] = "true"
    obj.status.phase = "Failed"
    obj.status.node = ""
