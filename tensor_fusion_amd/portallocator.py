"""Host-port + pod-index allocators.

Reference: internal/portallocator/portallocator.go:36-307 (node-level
40000-42000 and cluster-level 42000-62000 bitmaps, release on pod delete)
and internal/indexallocator/indexallocator.go:29-221 (small integer 1-32
per worker pod, surfaced as the `tensor-fusion.ai/index-N` placeholder
resource so the kubelet device plugin can correlate container→pod).
"""
from __future__ import annotations

import threading
from typing import Dict, Optional

from . import constants as C
from .api.store import Store


class PortExhausted(Exception):
    pass


class PortAllocator:
    """Bitmap port allocator with per-node and cluster ranges."""

    def __init__(self, store: Optional[Store] = None):
        self._lock = threading.Lock()
        self._cluster: Dict[int, str] = {}  # port -> pod key
        self._node: Dict[str, Dict[int, str]] = {}  # node -> port -> pod key
        self._by_pod: Dict[str, tuple] = {}  # pod key -> (scope, port)
        if store is not None:
            store.on_change("Pod", self._on_pod_event)

    def _on_pod_event(self, event: str, obj):
        if event == "DELETED":
            self.release(obj.meta.key)

    def assign_cluster_port(self, pod_key: str) -> int:
        with self._lock:
            if pod_key in self._by_pod:
                return self._by_pod[pod_key][1]
            for p in range(C.ClusterPortRangeStart, C.ClusterPortRangeEnd):
                if p not in self._cluster:
                    self._cluster[p] = pod_key
                    self._by_pod[pod_key] = ("cluster", p)
                    return p
        raise PortExhausted("cluster port range exhausted")

    def assign_node_port(self, node: str, pod_key: str) -> int:
        with self._lock:
            if pod_key in self._by_pod:
                return self._by_pod[pod_key][1]
            used = self._node.setdefault(node, {})
            for p in range(C.NodePortRangeStart, C.NodePortRangeEnd):
                if p not in used:
                    used[p] = pod_key
                    self._by_pod[pod_key] = (node, p)
                    return p
        raise PortExhausted(f"node {node} port range exhausted")

    def release(self, pod_key: str) -> None:
        with self._lock:
            rec = self._by_pod.pop(pod_key, None)
            if rec is None:
                return
            scope, port = rec
            if scope == "cluster":
                self._cluster.pop(port, None)
            else:
                self._node.get(scope, {}).pop(port, None)

    def in_use(self) -> int:
        with self._lock:
            return len(self._by_pod)


class IndexExhausted(Exception):
    pass


class IndexAllocator:
    """Per-node index 1..MaxWorkersPerNode. Before scheduling the node is
    unknown, so assignment is cluster-unique-per-pod first and occupied
    per node at Reserve (reference indexallocator.go:181)."""

    def __init__(self, store: Optional[Store] = None):
        self._lock = threading.Lock()
        self._by_pod: Dict[str, int] = {}
        self._node_used: Dict[str, Dict[int, str]] = {}
        if store is not None:
            store.on_change("Pod", self._on_pod_event)

    def _on_pod_event(self, event: str, obj):
        if event == "DELETED":
            self.release(obj.meta.key)

    def assign(self, pod_key: str) -> int:
        """Webhook-time: tentative index (reused verbatim if free on the
        chosen node, else re-assigned at occupy)."""

        with self._lock:
            if pod_key in self._by_pod:
                return self._by_pod[pod_key]
            used = set(self._by_pod.values())
            for i in range(1, C.MaxWorkersPerNode + 1):
                if i not in used:
                    self._by_pod[pod_key] = i
                    return i
            # more pending pods than per-node slots: wrap (nodes disambiguate)
            i = (len(self._by_pod) % C.MaxWorkersPerNode) + 1
            self._by_pod[pod_key] = i
            return i

    def occupy(self, node: str, pod_key: str) -> int:
        """Scheduler Reserve: claim the index on the chosen node; returns
        the (possibly re-assigned) index."""

        with self._lock:
            used = self._node_used.setdefault(node, {})
            want = self._by_pod.get(pod_key)
            if want is not None and used.get(want) in (None, pod_key):
                used[want] = pod_key
                return want
            for i in range(1, C.MaxWorkersPerNode + 1):
                if i not in used:
                    used[i] = pod_key
                    self._by_pod[pod_key] = i
                    return i
        raise IndexExhausted(f"node {node}: all {C.MaxWorkersPerNode} slots used")

    def release(self, pod_key: str) -> None:
        with self._lock:
            self._by_pod.pop(pod_key, None)
            for used in self._node_used.values():
                for i, k in list(used.items()):
                    if k == pod_key:
                        del used[i]

    def index_of(self, pod_key: str) -> Optional[int]:
        with self._lock:
            return self._by_pod.get(pod_key)
