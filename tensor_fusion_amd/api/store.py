"""Embedded object store — the stack's API server.

The reference keeps all state in a kube-apiserver (CRDs) and consumes it via
informers (list+watch). This store provides the same contract in-process:
thread-safe CRUD with resource versions, label selection, watch queues and
informer-style event handlers, plus optional JSON-lines persistence so a
hypervisor restart can rebuild state (reference §3.5 recovery paths).
"""
from __future__ import annotations

import copy
import json
import os
import threading
import time
from dataclasses import asdict, is_dataclass
from queue import Empty, Queue
from typing import Callable, Dict, List, Optional, Tuple

from .types import TFObject


class Conflict(Exception):
    """Resource-version conflict (optimistic concurrency)."""


class NotFound(Exception):
    pass


class AlreadyExists(Exception):
    pass


Event = Tuple[str, TFObject]  # ("ADDED"|"MODIFIED"|"DELETED", obj)


class Watch:
    def __init__(self, store: "Store", kind: str, q: Queue):
        self._store = store
        self._kind = kind
        self._q = q
        self.closed = False

    def next(self, timeout: Optional[float] = None) -> Optional[Event]:
        try:
            return self._q.get(timeout=timeout)
        except Empty:
            return None

    def __iter__(self):
        while not self.closed:
            ev = self.next(timeout=0.2)
            if ev is not None:
                yield ev

    def stop(self):
        self.closed = True
        self._store._drop_watch(self._kind, self._q)


class Store:
    def __init__(self, persist_dir: Optional[str] = None):
        self._lock = threading.RLock()
        self._objs: Dict[str, Dict[str, TFObject]] = {}  # kind -> key -> obj
        self._rv = 0
        self._watches: Dict[str, List[Queue]] = {}
        self._handlers: Dict[str, List[Callable[[str, TFObject], None]]] = {}
        self._persist_dir = persist_dir
        # per-thread re-entrancy depth + deferred handler events: handlers
        # must run OUTSIDE the store lock (informer contract) or a handler
        # that takes its own lock can deadlock against a thread holding
        # that lock while calling into the store
        self._tls = threading.local()
        if persist_dir:
            os.makedirs(persist_dir, exist_ok=True)
            self._load()

    class _Mutate:
        """Context manager: store lock + deferred handler dispatch.

        Watch-queue puts happen inline (Queue.put never calls back into
        user code); on_change handlers collected during the mutation fire
        after the OUTERMOST lock release, preserving event order."""

        def __init__(self, store: "Store"):
            self.store = store

        def __enter__(self):
            s = self.store
            s._lock.acquire()
            s._tls.depth = getattr(s._tls, "depth", 0) + 1
            if s._tls.depth == 1:
                s._tls.pending = []
            return self

        def __exit__(self, *exc):
            s = self.store
            s._tls.depth -= 1
            outermost = s._tls.depth == 0
            pending = s._tls.pending if outermost else None
            if outermost:
                s._tls.pending = []
            s._lock.release()
            if pending:
                for event, obj, handlers in pending:
                    for h in handlers:
                        try:
                            h(event, copy.deepcopy(obj))
                        except Exception:  # handlers must not break callers
                            import traceback
                            traceback.print_exc()
            return False

    # ------------------------------------------------------------- CRUD

    def create(self, obj: TFObject) -> TFObject:
        with self._Mutate(self):
            kind = obj.kind
            key = obj.meta.key
            bucket = self._objs.setdefault(kind, {})
            if key in bucket:
                raise AlreadyExists(f"{kind} {key}")
            self._rv += 1
            obj.meta.resource_version = self._rv
            # one deepcopy, not two: the stored copy is private to the
            # store; the caller keeps its own object (same isolation,
            # half the cost — admission throughput, tools/bench_webhook)
            bucket[key] = copy.deepcopy(obj)
            self._persist(kind)
            self._notify("ADDED", bucket[key])
            return obj

    def get(self, kind: str, name: str, namespace: str = "") -> TFObject:
        key = f"{namespace}/{name}" if namespace else name
        with self._lock:
            bucket = self._objs.get(kind, {})
            if key not in bucket:
                raise NotFound(f"{kind} {key}")
            return copy.deepcopy(bucket[key])

    def try_get(self, kind: str, name: str, namespace: str = "") -> Optional[TFObject]:
        try:
            return self.get(kind, name, namespace)
        except NotFound:
            return None

    def update(self, obj: TFObject, check_rv: bool = True) -> TFObject:
        with self._Mutate(self):
            bucket = self._objs.setdefault(obj.kind, {})
            key = obj.meta.key
            cur = bucket.get(key)
            if cur is None:
                raise NotFound(f"{obj.kind} {key}")
            if check_rv and obj.meta.resource_version != cur.meta.resource_version:
                raise Conflict(f"{obj.kind} {key}: rv {obj.meta.resource_version} "
                               f"!= {cur.meta.resource_version}")
            self._rv += 1
            obj.meta.resource_version = self._rv
            bucket[key] = copy.deepcopy(obj)
            self._persist(obj.kind)
            self._notify("MODIFIED", bucket[key])
            return obj

    def patch(self, kind: str, name: str, namespace: str,
              fn: Callable[[TFObject], None], retries: int = 8) -> TFObject:
        """Atomic read-modify-write (the controllers' idiom). The store is
        in-process, so the whole RMW runs under the store lock — unlike a
        remote apiserver there is no window for a conflicting writer, and
        concurrent patches can never exhaust retries and drop updates
        (found by tests/test_allocator_properties.py fuzzing the old
        bounded-retry loop under thread contention)."""

        with self._Mutate(self):
            key = f"{namespace}/{name}" if namespace else name
            bucket = self._objs.get(kind, {})
            cur = bucket.get(key)
            if cur is None:
                raise NotFound(f"{kind} {key}")
            obj = copy.deepcopy(cur)
            fn(obj)
            # no-op detection against the STORED original (dataclass
            # field-wise eq) — saves the old second "before" deepcopy
            if obj == cur:
                return obj
            self._rv += 1
            obj.meta.resource_version = self._rv
            bucket[key] = obj  # private copy: fn ran on it under the lock
            self._persist(kind)
            self._notify("MODIFIED", obj)
            # callers get their own copy so later mutation of the return
            # value cannot alias the stored object
            return copy.deepcopy(obj)

    def delete(self, kind: str, name: str, namespace: str = "") -> None:
        # Embedded-mode note: deletion is immediate (DELETED events carry
        # the final object state, so dealloc-on-event is exact). Real
        # finalizer deferral — deletionTimestamp until controllers strip
        # metadata.finalizers — lives in the k8s plane (K8sStore +
        # fake_apiserver), where an apiserver owns object lifetime.
        key = f"{namespace}/{name}" if namespace else name
        with self._Mutate(self):
            bucket = self._objs.get(kind, {})
            obj = bucket.pop(key, None)
            if obj is None:
                raise NotFound(f"{kind} {key}")
            self._persist(kind)
            self._notify("DELETED", obj)
        # owner-reference GC (kube garbage collector semantics): objects
        # whose meta.owner points at the deleted object go with it
        ref = f"{kind}/{namespace}/{name}" if namespace else f"{kind}//{name}"
        owned = []
        with self._lock:
            for k2, b2 in self._objs.items():
                for o in b2.values():
                    if getattr(o.meta, "owner", "") == ref:
                        owned.append((k2, o.meta.name, o.meta.namespace))
        for k2, n2, ns2 in owned:
            try:
                self.delete(k2, n2, ns2)
            except NotFound:
                pass

    def list(self, kind: str, namespace: Optional[str] = None,
             labels: Optional[Dict[str, str]] = None) -> List[TFObject]:
        with self._lock:
            out = []
            for obj in self._objs.get(kind, {}).values():
                if namespace is not None and obj.meta.namespace != namespace:
                    continue
                if labels and any(obj.meta.labels.get(k) != v for k, v in labels.items()):
                    continue
                out.append(copy.deepcopy(obj))
            return out

    # ------------------------------------------------------------ watch

    def watch(self, kind: str) -> Watch:
        q: Queue = Queue()
        with self._lock:
            self._watches.setdefault(kind, []).append(q)
        return Watch(self, kind, q)

    def on_change(self, kind: str, handler: Callable[[str, TFObject], None]):
        """Informer-style synchronous handler. Handlers run AFTER the store
        lock is released (outermost mutation exit), so they may freely take
        their own locks and call back into the store."""

        with self._lock:
            self._handlers.setdefault(kind, []).append(handler)

    def _drop_watch(self, kind: str, q: Queue):
        with self._lock:
            try:
                self._watches.get(kind, []).remove(q)
            except ValueError:
                pass

    def _notify(self, event: str, obj: TFObject):
        for q in self._watches.get(obj.kind, []):
            q.put((event, copy.deepcopy(obj)))
        handlers = list(self._handlers.get(obj.kind, []))
        if not handlers:
            return
        if getattr(self._tls, "depth", 0) > 0:
            # defer to the outermost _Mutate exit (fires outside the lock)
            self._tls.pending.append((event, copy.deepcopy(obj), handlers))
        else:
            for h in handlers:
                try:
                    h(event, copy.deepcopy(obj))
                except Exception:  # handlers must not break the store
                    import traceback
                    traceback.print_exc()

    # ---------------------------------------------------------- persist

    def _persist(self, kind: str):
        if not self._persist_dir:
            return
        path = os.path.join(self._persist_dir, f"{kind}.jsonl")
        tmp = path + ".tmp"
        with open(tmp, "w") as f:
            for obj in self._objs.get(kind, {}).values():
                f.write(json.dumps({"kind": kind, "obj": _to_dict(obj)}) + "\n")
        os.replace(tmp, path)

    def _load(self):
        from . import types as T
        for fn in os.listdir(self._persist_dir):
            if not fn.endswith(".jsonl"):
                continue
            kind = fn[:-6]
            cls = getattr(T, kind, None)
            if cls is None:
                continue
            path = os.path.join(self._persist_dir, fn)
            with open(path) as f:
                for line in f:
                    try:
                        rec = json.loads(line)
                        obj = _from_dict(cls, rec["obj"])
                        self._objs.setdefault(kind, {})[obj.meta.key] = obj
                        self._rv = max(self._rv, obj.meta.resource_version)
                    except Exception:
                        continue


def _to_dict(obj):
    return asdict(obj)


def _from_dict(cls, d):
    """Rebuild a dataclass tree from a dict (best-effort, tolerant)."""

    import dataclasses
    import typing

    if not (is_dataclass(cls) and isinstance(d, dict)):
        return d
    kwargs = {}
    hints = typing.get_type_hints(cls)
    for f in dataclasses.fields(cls):
        if f.name not in d:
            continue
        v = d[f.name]
        t = hints.get(f.name, None)
        origin = typing.get_origin(t)
        if is_dataclass(t) and isinstance(v, dict):
            v = _from_dict(t, v)
        elif origin is list and v:
            (et,) = typing.get_args(t)
            if is_dataclass(et):
                v = [_from_dict(et, x) for x in v]
        elif origin is typing.Union and isinstance(v, dict):
            args = [a for a in typing.get_args(t) if a is not type(None)]
            if len(args) == 1 and is_dataclass(args[0]):
                v = _from_dict(args[0], v)
        kwargs[f.name] = v
    return cls(**kwargs)
