"""Gang scheduling — all-or-nothing multi-pod admission.

Reference: internal/gang/manager.go:55-1179 — PodGroup registry keyed by
workload or explicit annotation, PreEnqueue quorum reachability, Permit
returning Wait until quorum, strict all-or-nothing rejection on timeout,
and workload Gang status flush.

On MI355X a gang is typically a TP/EP job whose RCCL collectives ride the
intra-node xGMI mesh — the gang guarantees all ranks co-start so RCCL's
bootstrap (rendezvous over the connection URLs) cannot deadlock.
"""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Dict, Optional, Set

from .. import constants as C
from ..api.store import Store
from ..api.types import Pod


@dataclass
class PodGroup:
    key: str
    min_members: int
    timeout_s: float = 60.0
    members: Set[str] = field(default_factory=set)       # known pod keys
    scheduled: Set[str] = field(default_factory=set)     # bound pod keys
    waiting: Set[str] = field(default_factory=set)       # parked in Permit
    waiting_since: Dict[str, float] = field(default_factory=dict)
    rejected_until: float = 0.0
    rejections: int = 0  # consecutive all-or-nothing rejections
    created: float = field(default_factory=time.time)

    @property
    def quorum_now(self) -> int:
        return len(self.scheduled | self.waiting)


class GangManager:
    BACKOFF_S = 3.0          # base backoff after a gang rejection
    BACKOFF_MAX_S = 120.0    # exponential cap (reference HandleTimeout
                             # re-arms with growing delay across cycles)

    def __init__(self, store: Optional[Store] = None):
        self._mu = threading.RLock()
        self.groups: Dict[str, PodGroup] = {}
        self._store = store

    # ------------------------------------------------------------ registry

    @staticmethod
    def group_key_of(pod: Pod) -> Optional[str]:
        a = pod.meta.annotations
        if a.get(C.AnnoGangEnabled, "").lower() != "true":
            return None
        return a.get(C.AnnoGangGroupKey) or \
            pod.meta.labels.get(C.LabelWorkload) or ""

    def register_pod(self, pod: Pod) -> Optional[PodGroup]:
        key = self.group_key_of(pod)
        if not key:
            return None
        a = pod.meta.annotations
        with self._mu:
            g = self.groups.get(key)
            if g is None:
                g = PodGroup(
                    key=key,
                    min_members=int(a.get(C.AnnoGangMinMembers, "0") or 0),
                    timeout_s=float(a.get(C.AnnoGangTimeout, "60") or 60))
                self.groups[key] = g
            if a.get(C.AnnoGangMinMembers):
                g.min_members = int(a[C.AnnoGangMinMembers])
            g.members.add(pod.meta.key)
            return g

    def forget_pod(self, pod_key: str):
        with self._mu:
            for g in self.groups.values():
                g.members.discard(pod_key)
                g.scheduled.discard(pod_key)
                g.waiting.discard(pod_key)
                g.waiting_since.pop(pod_key, None)

    def active_groups(self) -> Set[str]:
        """Groups still pursuing quorum (feeds the allocator's TTL sweep)."""

        with self._mu:
            return {k for k, g in self.groups.items()
                    if g.waiting and time.time() >= g.rejected_until}

    # ----------------------------------------------------- scheduler hooks

    def pre_enqueue(self, pod: Pod) -> Optional[str]:
        """None = admit; otherwise the unschedulable reason. Quorum must be
        *reachable*: enough member pods exist (reference PreEnqueue :509)."""

        g = self.register_pod(pod)
        if g is None:
            return None
        now = time.time()
        if now < g.rejected_until:
            return f"gang {g.key} in backoff after rejection"
        known = len(g.members)
        if self._store is not None:
            # Quorum reachability is judged against pods that EXIST (the
            # reference counts the informer cache, manager.go:509), not
            # against the subset this manager has already seen.
            existing = sum(1 for p in self._store.list("Pod")
                           if p.meta.deletion_ts is None
                           and self.group_key_of(p) == g.key)
            known = max(known, existing)
        if g.min_members and known < g.min_members:
            return (f"gang {g.key}: quorum unreachable "
                    f"({known}/{g.min_members} members exist)")
        return None

    def permit(self, pod: Pod) -> Optional[float]:
        """None = allow immediately (not ganged / quorum met); else the wait
        timeout in seconds (reference Permit :746)."""

        key = self.group_key_of(pod)
        if not key:
            return None
        with self._mu:
            g = self.groups[key]
            g.waiting.add(pod.meta.key)
            g.waiting_since.setdefault(pod.meta.key, time.time())
            if g.min_members and g.quorum_now < g.min_members:
                return g.timeout_s
            return None  # quorum met: allow, and release the others

    def quorum_met(self, key: str) -> bool:
        with self._mu:
            g = self.groups.get(key)
            return bool(g and (not g.min_members or g.quorum_now >= g.min_members))

    def mark_scheduled(self, pod: Pod):
        key = self.group_key_of(pod)
        if not key:
            return
        with self._mu:
            g = self.groups[key]
            g.waiting.discard(pod.meta.key)
            g.waiting_since.pop(pod.meta.key, None)
            g.scheduled.add(pod.meta.key)
            g.rejections = 0  # progress resets the backoff ladder
        self._flush_status(key)

    def reject_group(self, key: str):
        """Strict all-or-nothing: reject every waiting member and back the
        group off EXPONENTIALLY across cycles (reference :262/:1099 +
        HandleTimeout :977 — a gang that keeps missing quorum must not
        re-burn a full scheduling cycle every few seconds)."""

        with self._mu:
            g = self.groups.get(key)
            if g is None:
                return
            g.waiting.clear()
            g.waiting_since.clear()
            backoff = min(self.BACKOFF_S * (2 ** g.rejections),
                          self.BACKOFF_MAX_S)
            g.rejections += 1
            g.rejected_until = time.time() + backoff
        self._flush_status(key)

    def backoff_remaining(self, key: str,
                          now: Optional[float] = None) -> float:
        now = now if now is not None else time.time()
        with self._mu:
            g = self.groups.get(key)
            return max(0.0, g.rejected_until - now) if g else 0.0

    def sweep_timeouts(self, now: Optional[float] = None) -> Set[str]:
        """Reject every group whose oldest waiting member exceeded the
        gang timeout — the watchdog the scheduler loop calls so a
        half-assembled gang never parks reserved GPUs forever (the
        reference's timeout path frees Assume()d devices through
        Unreserve; here the framework's unreserve hook fires when the
        permit wait ends rejected)."""

        now = now if now is not None else time.time()
        expired: Set[str] = set()
        with self._mu:
            for key, g in self.groups.items():
                if not g.waiting:
                    continue
                oldest = min(g.waiting_since.get(p, now)
                             for p in g.waiting)
                if now - oldest > g.timeout_s:
                    expired.add(key)
        for key in expired:
            self.reject_group(key)
        return expired

    # -------------------------------------------------------------- status

    def _flush_status(self, key: str):
        """Write gang status into the owning TensorFusionWorkload."""

        if not self._store:
            return
        with self._mu:
            g = self.groups.get(key)
            if g is None:
                return
            phase = "Scheduled" if (g.min_members and
                                    len(g.scheduled) >= g.min_members) else "Pending"
            total, sched = len(g.members), len(g.scheduled)
        for wl in self._store.list("TensorFusionWorkload"):
            if wl.meta.name == key or wl.profile.gang.group_key == key:
                def _p(obj):
                    obj.status.gang.group = key
                    obj.status.gang.phase = phase
                    obj.status.gang.members_total = total
                    obj.status.gang.members_scheduled = sched
                try:
                    self._store.patch("TensorFusionWorkload", wl.meta.name,
                                      wl.meta.namespace, _p)
                except Exception:
                    pass
