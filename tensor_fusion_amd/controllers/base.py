"""Controller runtime — reconciler framework over the embedded store.

Reference: controller-runtime managers in cmd/main.go:414-572. Each
controller watches one primary kind (plus optional extra kinds mapped to
primary keys), keeps a rate-limited work queue, and reconciles one object
key at a time. `ControllerManager.reconcile_now()` drains every queue
synchronously — the deterministic mode tests and the single-node runtime
use; `start()` runs background workers like the operator binary.
"""
from __future__ import annotations

import threading
import time
import traceback
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

from ..api.store import NotFound, Store

RequeueAfter = float  # seconds; 0 = done


@dataclass
class Request:
    kind: str
    name: str
    namespace: str = ""

    @property
    def key(self) -> str:
        return f"{self.namespace}/{self.name}" if self.namespace else self.name


class Reconciler:
    """Subclass contract: set `kind`, implement reconcile()."""

    kind = ""
    # extra kinds whose events enqueue mapped requests
    watches: List[str] = []
    resync_s: float = 0.0  # periodic full resync (0 = off)

    def __init__(self, store: Store):
        self.store = store

    def reconcile(self, req: Request) -> RequeueAfter:
        raise NotImplementedError

    def map_event(self, kind: str, event: str, obj) -> List[Request]:
        """Map a watched-kind event to primary requests. Default: objects
        owned by the primary kind via meta.owner."""

        owner = obj.meta.owner or ""
        if owner.startswith(self.kind + "/"):
            _, ns, name = (owner.split("/", 2) + [""])[:3]
            return [Request(self.kind, name or ns, ns if name else "")]
        return []


class _Queue:
    def __init__(self):
        self._items: Dict[str, Request] = {}
        self._delayed: List[Tuple[float, Request]] = []
        self._cv = threading.Condition()

    def add(self, req: Request, after: float = 0.0):
        with self._cv:
            if after > 0:
                self._delayed.append((time.time() + after, req))
            else:
                self._items[req.key] = req
            self._cv.notify()

    def pop(self, timeout: float = 0.2) -> Optional[Request]:
        with self._cv:
            self._promote()
            if not self._items:
                self._cv.wait(timeout)
                self._promote()
            if not self._items:
                return None
            key = next(iter(self._items))
            return self._items.pop(key)

    def _promote(self):
        now = time.time()
        still = []
        for due, req in self._delayed:
            if due <= now:
                self._items[req.key] = req
            else:
                still.append((due, req))
        self._delayed = still

    def __len__(self):
        with self._cv:
            self._promote()
            return len(self._items)


class ControllerManager:
    def __init__(self, store: Store):
        self.store = store
        self._controllers: List[Reconciler] = []
        self._queues: Dict[str, _Queue] = {}
        self._threads: List[threading.Thread] = []
        self._stop = threading.Event()
        self._errors: List[str] = []

    def register(self, ctrl: Reconciler):
        self._controllers.append(ctrl)
        q = self._queues.setdefault(ctrl.kind, _Queue())

        def primary_handler(event, obj, ctrl=ctrl, q=q):
            q.add(Request(ctrl.kind, obj.meta.name, obj.meta.namespace))

        self.store.on_change(ctrl.kind, primary_handler)
        for extra in ctrl.watches:
            def extra_handler(event, obj, ctrl=ctrl, q=q, extra=extra):
                for req in ctrl.map_event(extra, event, obj):
                    q.add(req)
            self.store.on_change(extra, extra_handler)

    # ----------------------------------------------------- deterministic

    def enqueue_all(self):
        for ctrl in self._controllers:
            for obj in self.store.list(ctrl.kind):
                self._queues[ctrl.kind].add(
                    Request(ctrl.kind, obj.meta.name, obj.meta.namespace))

    def reconcile_now(self, rounds: int = 4) -> int:
        """Drain all queues synchronously; cascading events re-fill queues,
        hence multiple rounds. Returns number of reconciles executed."""

        n = 0
        self.enqueue_all()
        for _ in range(rounds):
            progressed = False
            for ctrl in self._controllers:
                q = self._queues[ctrl.kind]
                # drain only what is queued at round start: reconciles that
                # re-trigger themselves are picked up next round, not forever
                for _ in range(len(q)):
                    req = q.pop(timeout=0)
                    if req is None:
                        break
                    n += 1
                    progressed = True
                    self._run_one(ctrl, req, q)
            if not progressed:
                break
        return n

    def _run_one(self, ctrl: Reconciler, req: Request, q: _Queue):
        try:
            after = ctrl.reconcile(req) or 0.0
            if after > 0:
                q.add(req, after)
        except NotFound:
            pass  # object deleted between enqueue and reconcile
        except Exception:
            self._errors.append(traceback.format_exc())
            q.add(req, 1.0)

    # ----------------------------------------------------------- threads

    def start(self):
        self._stop.clear()
        for ctrl in self._controllers:
            t = threading.Thread(target=self._worker, args=(ctrl,),
                                 daemon=True, name=f"ctrl-{ctrl.kind}")
            t.start()
            self._threads.append(t)
            if ctrl.resync_s > 0:
                rt = threading.Thread(target=self._resync, args=(ctrl,),
                                      daemon=True)
                rt.start()
                self._threads.append(rt)
        self.enqueue_all()

    def _worker(self, ctrl: Reconciler):
        q = self._queues[ctrl.kind]
        while not self._stop.is_set():
            req = q.pop(timeout=0.2)
            if req is not None:
                self._run_one(ctrl, req, q)

    def _resync(self, ctrl: Reconciler):
        q = self._queues[ctrl.kind]
        while not self._stop.wait(ctrl.resync_s):
            for obj in self.store.list(ctrl.kind):
                q.add(Request(ctrl.kind, obj.meta.name, obj.meta.namespace))

    def stop(self):
        self._stop.set()
        for t in self._threads:
            t.join(timeout=2)
        self._threads.clear()

    @property
    def errors(self) -> List[str]:
        return list(self._errors)
