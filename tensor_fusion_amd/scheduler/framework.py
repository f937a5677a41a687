"""Scheduler framework — the kube-scheduler plugin contract, embedded.

Reference: cmd/sched/setup.go embeds a patched kube-scheduler; plugins
implement PreEnqueue/PreFilter/Filter/Score/Reserve/Permit/PreBind/PostBind/
Unreserve/PostFilter. This module provides the same extension points over
the embedded object store, with a waiting-pod pool for gang Permit and a
nominated-pod watchdog for preemption.
"""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from ..api.store import Store
from ..api.types import Pod

# ------------------------------------------------------------------ status


class Code:
    Success = "Success"
    Unschedulable = "Unschedulable"
    Error = "Error"
    Wait = "Wait"
    Skip = "Skip"


@dataclass
class Status:
    code: str = Code.Success
    reasons: List[str] = field(default_factory=list)

    @property
    def ok(self) -> bool:
        return self.code in (Code.Success, Code.Skip)

    @classmethod
    def ok_(cls):
        return cls(Code.Success)

    @classmethod
    def unschedulable(cls, *reasons: str):
        return cls(Code.Unschedulable, list(reasons))

    @classmethod
    def error(cls, *reasons: str):
        return cls(Code.Error, list(reasons))

    @classmethod
    def wait(cls, *reasons: str):
        return cls(Code.Wait, list(reasons))


class CycleState(dict):
    """Per-scheduling-attempt scratch space shared across plugins."""


# ------------------------------------------------------------------ plugin


class Plugin:
    name = "Plugin"

    # Queue gate: may this pod enter the active queue at all?
    def pre_enqueue(self, pod: Pod) -> Status:
        return Status.ok_()

    # Compute pod-wide state; returns (allowed node names or None=all, status)
    def pre_filter(self, state: CycleState, pod: Pod
                   ) -> Tuple[Optional[List[str]], Status]:
        return None, Status.ok_()

    def filter(self, state: CycleState, pod: Pod, node: str) -> Status:
        return Status.ok_()

    # Called when filtering failed everywhere (preemption hook).
    def post_filter(self, state: CycleState, pod: Pod,
                    filtered_nodes: List[str]) -> Status:
        return Status.unschedulable()

    def score(self, state: CycleState, pod: Pod, node: str) -> float:
        return 0.0

    def reserve(self, state: CycleState, pod: Pod, node: str) -> Status:
        return Status.ok_()

    def unreserve(self, state: CycleState, pod: Pod, node: str) -> None:
        pass

    # Permit: Success, Wait (gang) or Unschedulable.
    def permit(self, state: CycleState, pod: Pod, node: str
               ) -> Tuple[Status, float]:
        return Status.ok_(), 0.0

    def pre_bind(self, state: CycleState, pod: Pod, node: str) -> Status:
        return Status.ok_()

    def post_bind(self, state: CycleState, pod: Pod, node: str) -> None:
        pass


# ----------------------------------------------------------------- waiting


@dataclass
class WaitingPod:
    pod: Pod
    node: str
    state: CycleState
    deadline: float
    allowed: Optional[bool] = None  # None = still waiting
    cv: threading.Condition = field(default_factory=threading.Condition)

    def allow(self):
        with self.cv:
            if self.allowed is None:
                self.allowed = True
                self.cv.notify_all()

    def reject(self):
        with self.cv:
            if self.allowed is None:
                self.allowed = False
                self.cv.notify_all()

    def wait(self) -> bool:
        with self.cv:
            while self.allowed is None:
                remain = self.deadline - time.time()
                if remain <= 0:
                    self.allowed = False
                    break
                self.cv.wait(timeout=min(remain, 0.1))
            return bool(self.allowed)


# --------------------------------------------------------------- scheduler


@dataclass
class ScheduleResult:
    pod_key: str
    node: str = ""
    status: str = Code.Success
    reasons: List[str] = field(default_factory=list)


class Scheduler:
    """The scheduling loop over the embedded store.

    Pods with spec.scheduler_name == tensor-fusion-scheduler and no node get
    one full framework cycle each; gang pods park in the waiting pool during
    Permit and bind together (binding threads block in Permit-wait like
    kube-scheduler's WaitOnPermit).
    """

    def __init__(self, store: Store, plugins: List[Plugin],
                 scheduler_name: str = "tensor-fusion-scheduler",
                 bind_fn=None):
        self.store = store
        self.plugins = plugins
        self.scheduler_name = scheduler_name
        self.waiting: Dict[str, WaitingPod] = {}
        self._waiting_mu = threading.Lock()
        self._bind_fn = bind_fn or self._default_bind
        self._unsched_backoff: Dict[str, float] = {}
        self.results: Dict[str, ScheduleResult] = {}
        self._next_start = 0  # rotating feasible-node scan offset
        # event-maintained ready-node cache: store.list deepcopies every
        # object, which dominated the scheduling hot path at 1k nodes
        self._node_mu = threading.Lock()
        self._ready_nodes: Dict[str, bool] = {
            n.meta.name: n.status_phase == "Ready"
            for n in store.list("Node")}
        self._ready_list: Optional[List[str]] = None  # cache of _nodes()
        store.on_change("Node", self._on_node_event)

    def _on_node_event(self, event: str, obj):
        with self._node_mu:
            if event == "DELETED":
                self._ready_nodes.pop(obj.meta.name, None)
            else:
                self._ready_nodes[obj.meta.name] = (
                    obj.status_phase == "Ready")
            self._ready_list = None

    # ------------------------------------------------------------- binding

    def _default_bind(self, pod: Pod, node: str):
        def _p(obj):
            obj.status.node = node
            obj.status.phase = "Scheduled"
        self.store.patch("Pod", pod.meta.name, pod.meta.namespace, _p)

    # ------------------------------------------------------------- helpers

    def waiting_pods(self) -> List[WaitingPod]:
        with self._waiting_mu:
            return list(self.waiting.values())

    def iterate_waiting(self, fn):
        for wp in self.waiting_pods():
            fn(wp)

    def _nodes(self) -> List[str]:
        with self._node_mu:
            if self._ready_list is None:
                self._ready_list = [n for n, ready in
                                    self._ready_nodes.items() if ready]
            return self._ready_list

    # -------------------------------------------------------------- cycle

    def schedule_pod(self, pod: Pod) -> ScheduleResult:
        key = pod.meta.key
        res = ScheduleResult(pod_key=key)
        state = CycleState()

        for p in self.plugins:
            st = p.pre_enqueue(pod)
            if not st.ok:
                res.status, res.reasons = st.code, st.reasons
                self.results[key] = res
                return res

        allowed: Optional[set] = None
        for p in self.plugins:
            nodes, st = p.pre_filter(state, pod)
            if st.code == Code.Skip:
                continue
            if not st.ok:
                if st.code == Code.Unschedulable:
                    # PostFilter (preemption) runs on any unschedulable
                    # outcome, PreFilter-rejects included (kube-scheduler
                    # semantics).
                    for q in self.plugins:
                        pst = q.post_filter(state, pod, [])
                        if pst.ok:
                            res.status = Code.Unschedulable
                            res.reasons = (["preemption initiated"] +
                                           pst.reasons + st.reasons)
                            self.results[key] = res
                            return res
                res.status, res.reasons = st.code, st.reasons
                self.results[key] = res
                return res
            if nodes is not None:
                allowed = set(nodes) if allowed is None else allowed & set(nodes)

        cand = [n for n in self._nodes() if allowed is None or n in allowed]
        # kube-scheduler's numFeasibleNodesToFind: at scale, stop after
        # enough feasible nodes instead of filtering/scoring all of them.
        # Rotate the scan start across cycles (kube-scheduler's
        # nextStartNodeIndex) so truncation doesn't always consider the
        # same prefix of the fleet — without this, nodes past `want` are
        # never scored and placement hotspots on the first nodes.
        if len(cand) > 1:
            start = self._next_start % len(cand)
            self._next_start += 1
            cand = cand[start:] + cand[:start]
        want = max(100, len(cand) * 8 // 100)
        feasible = []
        reasons: List[str] = []
        for n in cand:
            ok = True
            for p in self.plugins:
                st = p.filter(state, pod, n)
                if not st.ok:
                    ok = False
                    reasons.extend(f"{n}: {r}" for r in st.reasons)
                    break
            if ok:
                feasible.append(n)
                if len(feasible) >= want:
                    break

        if not feasible:
            # PostFilter (preemption) — first plugin that succeeds wins.
            for p in self.plugins:
                st = p.post_filter(state, pod, cand)
                if st.ok:
                    res.status = Code.Unschedulable
                    res.reasons = ["preemption initiated"] + st.reasons
                    self.results[key] = res
                    return res
            res.status = Code.Unschedulable
            res.reasons = reasons or ["no feasible node"]
            self.results[key] = res
            return res

        scored = sorted(
            feasible,
            key=lambda n: sum(p.score(state, pod, n) for p in self.plugins),
            reverse=True)
        node = scored[0]

        # Reserve
        for i, p in enumerate(self.plugins):
            st = p.reserve(state, pod, node)
            if not st.ok:
                for q in self.plugins[:i + 1]:
                    q.unreserve(state, pod, node)
                res.status, res.reasons = st.code, st.reasons
                self.results[key] = res
                return res

        # Permit
        max_timeout = 0.0
        wait_needed = False
        permit_failed: Optional[Status] = None
        for p in self.plugins:
            st, timeout = p.permit(state, pod, node)
            if st.code == Code.Wait:
                wait_needed = True
                max_timeout = max(max_timeout, timeout)
            elif not st.ok:
                permit_failed = st
                break
        if permit_failed is not None:
            for p in self.plugins:
                p.unreserve(state, pod, node)
            res.status, res.reasons = permit_failed.code, permit_failed.reasons
            self.results[key] = res
            return res

        if not wait_needed:
            # e.g. the quorum-completing gang member: let plugins release
            # their parked peers from the waiting pool.
            for p in self.plugins:
                hook = getattr(p, "on_permit_allowed", None)
                if hook:
                    hook(self, pod)

        if wait_needed:
            wp = WaitingPod(pod=pod, node=node, state=state,
                            deadline=time.time() + max_timeout)
            with self._waiting_mu:
                self.waiting[key] = wp
            try:
                for p in self.plugins:
                    on_wait = getattr(p, "on_pod_waiting", None)
                    if on_wait:
                        on_wait(self, wp)
                ok = wp.wait()
            finally:
                with self._waiting_mu:
                    self.waiting.pop(key, None)
            if not ok:
                for p in self.plugins:
                    p.unreserve(state, pod, node)
                for p in self.plugins:
                    on_rej = getattr(p, "on_pod_rejected", None)
                    if on_rej:
                        on_rej(self, pod)
                res.status = Code.Unschedulable
                res.reasons = ["permit wait timed out / gang rejected"]
                self.results[key] = res
                return res

        # PreBind (commit + annotation patch), with rollback on failure.
        for p in self.plugins:
            st = p.pre_bind(state, pod, node)
            if not st.ok:
                for q in self.plugins:
                    q.unreserve(state, pod, node)
                res.status, res.reasons = st.code, st.reasons
                self.results[key] = res
                return res

        try:
            self._bind_fn(pod, node)
        except Exception as e:  # bind failed: roll everything back
            for p in self.plugins:
                p.unreserve(state, pod, node)
            res.status, res.reasons = Code.Error, [f"bind: {e}"]
            self.results[key] = res
            return res

        for p in self.plugins:
            p.post_bind(state, pod, node)
        res.node = node
        self.results[key] = res
        return res

    # --------------------------------------------------------------- loop

    def pending_pods(self) -> List[Pod]:
        return [p for p in self.store.list("Pod")
                if p.scheduler_name == self.scheduler_name
                and not p.status.node and p.status.phase == "Pending"
                and p.meta.deletion_ts is None]

    def schedule_pending(self, parallel: bool = True) -> List[ScheduleResult]:
        """One pass over the pending queue. Gang members must be scheduled
        concurrently (they block in Permit), hence thread-per-pod."""

        pods = self.pending_pods()
        now = time.time()
        pods = [p for p in pods if self._unsched_backoff.get(p.meta.key, 0) <= now]
        if not pods:
            return []
        results: List[ScheduleResult] = []
        if parallel and len(pods) > 1:
            threads = []
            out: Dict[str, ScheduleResult] = {}

            def run(pod: Pod):
                out[pod.meta.key] = self.schedule_pod(pod)

            for p in pods:
                t = threading.Thread(target=run, args=(p,), daemon=True)
                t.start()
                threads.append(t)
            for t in threads:
                t.join(timeout=120)
            results = list(out.values())
        else:
            results = [self.schedule_pod(p) for p in pods]
        for r in results:
            if r.status != Code.Success:
                self._unsched_backoff[r.pod_key] = time.time() + 1.0
        return results
