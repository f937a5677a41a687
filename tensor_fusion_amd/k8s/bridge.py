"""K8sStore — the embedded Store write-through-backed by a kube-apiserver.

Every control-plane component (allocator, scheduler, controllers, webhook,
autoscaler) speaks to the Store interface; swapping Store for K8sStore
puts the whole operator on a real cluster without touching them:

  writes   create/update/patch/delete push to the apiserver first (spec
           via the main resource, status via the /status subresource, as
           a real apiserver requires) and the returned object — with the
           server's resourceVersion — lands in the local cache.
  reads    served from the local informer-maintained cache (client-go
           semantics; reference controllers read from informer caches
           too, cmd/main.go manager cache).
  watch    per-kind Informers feed remote events into the cache; echoes
           of our own writes dedupe on resourceVersion.

Reference anchors: gpuallocator.go:2157-2621 (dirty-queue status sync to
GPU CRs), kubernetes_backend.go:312 (hypervisor GPU CR publication).
"""
from __future__ import annotations

import threading
from typing import Callable, Dict, List, Optional

from ..api import types as T
from ..api.store import AlreadyExists, Conflict, NotFound, Store
from . import serde
from .client import ApiError, K8sClient
from .informer import Informer

# kinds mirrored by default: every CRD + core pods/nodes
DEFAULT_KINDS = list(serde.CRD_KINDS) + ["Pod", "Node"]


def _diff_merge(old, new):
    """RFC-7386 merge patch turning `old` into `new` (None = delete)."""

    if not (isinstance(old, dict) and isinstance(new, dict)):
        return new if new != old else None
    patch = {}
    for k, v in new.items():
        if k not in old:
            patch[k] = v
        else:
            sub = _diff_merge(old[k], v)
            if sub is not None or old[k] != v:
                patch[k] = sub if sub is not None else v
    for k in old:
        if k not in new:
            patch[k] = None
    return patch or None


class K8sStore(Store):
    def __init__(self, client: K8sClient, kinds: Optional[List[str]] = None,
                 namespace: str = ""):
        super().__init__(persist_dir=None)
        self.client = client
        self.kinds = kinds or DEFAULT_KINDS
        self.namespace = namespace
        self._informers: List[Informer] = []

    # --------------------------------------------------------- lifecycle

    def start(self, wait: bool = True) -> "K8sStore":
        for kind in self.kinds:
            ns = "" if kind in serde.CLUSTER_SCOPED or kind == "Node" \
                else self.namespace
            inf = Informer(self.client, kind, namespace=ns,
                           on_event=self._on_remote_event)
            self._informers.append(inf.start())
        if wait:
            for inf in self._informers:
                inf.wait_synced()
        return self

    def stop(self):
        for inf in self._informers:
            inf.stop()

    # ------------------------------------------------------ remote apply

    def _on_remote_event(self, typ: str, wire: dict):
        try:
            obj = serde.from_k8s(wire)
        except (ValueError, TypeError):
            return
        self._apply_remote(typ, obj)

    def _apply_remote(self, typ: str, obj: T.TFObject):
        with self._Mutate(self):
            bucket = self._objs.setdefault(obj.kind, {})
            key = obj.meta.key
            cur = bucket.get(key)
            if typ == "DELETED":
                if cur is not None:
                    del bucket[key]
                    self._notify("DELETED", cur)
                return
            if cur is not None and \
                    cur.meta.resource_version >= obj.meta.resource_version:
                return  # stale or echo of our own write
            bucket[key] = obj
            self._rv = max(self._rv, obj.meta.resource_version)
            self._notify("ADDED" if cur is None else "MODIFIED", obj)

    # ------------------------------------------------- write-through API

    def create(self, obj: T.TFObject) -> T.TFObject:
        wire = serde.to_k8s(obj)
        wire.get("metadata", {}).pop("resourceVersion", None)
        status = wire.pop("status", None)
        try:
            out = self.client.create(wire)
        except ApiError as e:
            if e.status == 409:
                raise AlreadyExists(f"{obj.kind} {obj.meta.key}")
            raise
        if status and obj.kind not in ("Pod", "Node"):
            # status subresource write (creation status is meaningful for
            # device inventory: hypervisor publishes GPU CRs with status)
            out["status"] = status
            try:
                out = self.client.update_status(out)
            except ApiError:
                pass
        applied = serde.from_k8s(out)
        self._apply_remote("ADDED", applied)
        return applied.deepcopy()

    def update(self, obj: T.TFObject, check_rv: bool = True) -> T.TFObject:
        cur = self.try_get(obj.kind, obj.meta.name, obj.meta.namespace)
        if cur is None:
            raise NotFound(f"{obj.kind} {obj.meta.key}")
        if check_rv and obj.meta.resource_version != cur.meta.resource_version:
            raise Conflict(
                f"{obj.kind} {obj.meta.key}: rv {obj.meta.resource_version}"
                f" != {cur.meta.resource_version}")
        old_wire = serde.to_k8s(cur)
        new_wire = serde.to_k8s(obj)
        old_status = old_wire.pop("status", None)
        new_status = new_wire.pop("status", None)
        # server-owned metadata fields never go into a patch
        for w in (old_wire, new_wire):
            for k in ("resourceVersion", "uid", "creationTimestamp"):
                w.get("metadata", {}).pop(k, None)
        name, ns = obj.meta.name, obj.meta.namespace
        out = None
        if obj.kind == "Pod":
            # a real apiserver only sets spec.nodeName through the binding
            # subresource (how kube-scheduler binds); never patch it
            old_node = old_wire.get("spec", {}).pop("nodeName", "")
            new_node = new_wire.get("spec", {}).pop("nodeName", "")
            if new_node and new_node != old_node:
                self.client.bind_pod(name, ns, new_node)
        spec_patch = _diff_merge(old_wire, new_wire)
        if spec_patch:
            try:
                out = self.client.patch(obj.kind, name, spec_patch,
                                        namespace=ns)
            except ApiError as e:
                if e.conflict:
                    raise Conflict(str(e))
                if e.not_found:
                    raise NotFound(f"{obj.kind} {obj.meta.key}")
                raise
        if new_status is not None and new_status != old_status and \
                obj.kind not in ("Pod", "Node"):
            try:
                out = self.client.patch(obj.kind, name,
                                        {"status": new_status},
                                        namespace=ns, subresource="status")
            except ApiError as e:
                if e.not_found:
                    raise NotFound(f"{obj.kind} {obj.meta.key}")
                raise
        elif obj.kind == "Pod" and new_status != old_status:
            try:
                out = self.client.patch(
                    obj.kind, name,
                    {"status": serde.pod_to_k8s(obj)["status"]},
                    namespace=ns, subresource="status")
            except ApiError:
                pass
        if out is None:
            out = self.client.get(obj.kind, name, ns)
        applied = serde.from_k8s(out)
        self._apply_remote("MODIFIED", applied)
        return applied.deepcopy()

    def patch(self, kind: str, name: str, namespace: str,
              fn: Callable[[T.TFObject], None], retries: int = 8
              ) -> T.TFObject:
        """Optimistic RMW against the apiserver (unlike the embedded
        store, a remote writer can conflict → bounded retries with
        refetch, the reference's controller-runtime retry idiom)."""

        last: Optional[Exception] = None
        for _ in range(max(1, retries)):
            obj = self.get(kind, name, namespace)
            before = serde.to_k8s(obj)
            fn(obj)
            if serde.to_k8s(obj) == before:
                return obj
            try:
                return self.update(obj)
            except Conflict as e:
                last = e
                # refresh cache from the server before retrying
                try:
                    fresh = serde.from_k8s(self.client.get(
                        kind, name, namespace))
                    self._apply_remote("MODIFIED", fresh)
                except ApiError:
                    pass
                continue
        raise last or Conflict(f"{kind} {namespace}/{name}")

    def delete(self, kind: str, name: str, namespace: str = "") -> None:
        try:
            self.client.delete(kind, name, namespace)
        except ApiError as e:
            if e.not_found:
                raise NotFound(f"{kind} {namespace}/{name}")
            raise
        # finalizers defer deletion on a real apiserver: the object then
        # still exists with deletionTimestamp set and MUST stay in the
        # local cache (controllers reconcile it to strip finalizers) —
        # dropping it locally would orphan the wire object forever
        try:
            remote = self.client.get(kind, name, namespace)
        except ApiError:
            remote = None
        cur = self.try_get(kind, name, namespace)
        if remote is not None:
            from . import serde as _serde
            self._apply_remote("MODIFIED", _serde.from_k8s(remote))
            return
        if cur is not None:
            self._apply_remote("DELETED", cur)
        # owner-reference GC: the embedded store cascades in-process; on a
        # real cluster the kube garbage collector does it, but the fake
        # apiserver has no GC controller, so cascade through the client
        # for parity with the embedded store's behavior
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
