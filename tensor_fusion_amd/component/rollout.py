"""Batch rolling-update engine for stack components.

Reference: internal/component/component.go ManageUpdate:31 +
hypervisor.go/worker.go/client.go: each component computes the hash of
its pool template, finds pods running an older hash, and recreates them
in batches of `batch_percent` with `interval_s` between batches.
"""
from __future__ import annotations

import hashlib
import json
import time
from typing import List, Optional

from .. import constants as C
from ..api.store import NotFound, Store

AnnoTemplateHash = f"{C.Domain}/template-hash"


def template_hash(template: dict) -> str:
    return hashlib.sha256(
        json.dumps(template, sort_keys=True).encode()).hexdigest()[:12]


class ComponentRollout:
    def __init__(self, store: Store, component: str,
                 batch_percent: int = 25, interval_s: float = 60.0):
        self.store = store
        self.component = component  # hypervisor | worker | client
        self.batch_percent = batch_percent
        self.interval_s = interval_s
        self._last_batch_ts = 0.0

    def _component_pods(self, pool: Optional[str] = None):
        return [p for p in self.store.list("Pod")
                if p.meta.labels.get(C.LabelComponent) == self.component
                and (pool is None or p.meta.labels.get(C.LabelPool) == pool)
                and p.meta.deletion_ts is None]

    def out_of_date(self, template: dict, pool: Optional[str] = None):
        want = template_hash(template)
        return [p for p in self._component_pods(pool)
                if p.meta.annotations.get(AnnoTemplateHash) != want]

    def stamp(self, pod, template: dict):
        pod.meta.annotations[AnnoTemplateHash] = template_hash(template)
        return pod

    def tick(self, template: dict, pool: Optional[str] = None,
             now: Optional[float] = None) -> List[str]:
        """One rollout step: delete up to batch_percent of out-of-date pods
        (their owners recreate them from the new template). Returns the
        deleted pod keys."""

        now = now if now is not None else time.time()
        if now - self._last_batch_ts < self.interval_s:
            return []
        stale = self.out_of_date(template, pool)
        if not stale:
            return []
        all_pods = self._component_pods(pool)
        batch = max(1, len(all_pods) * self.batch_percent // 100)
        victims = stale[:batch]
        deleted = []
        for p in victims:
            try:
                self.store.delete("Pod", p.meta.name, p.meta.namespace)
                deleted.append(p.meta.key)
            except NotFound:
                continue
        self._last_batch_ts = now
        return deleted
