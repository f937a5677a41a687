"""List+watch informer over K8sClient.

Mirrors the client-go informer contract the reference's controllers sit
on: initial LIST applied as ADDED events, then a WATCH from the list's
resourceVersion; on 410 Gone or stream close it relists. Events are
delivered on the informer's own thread.
"""
from __future__ import annotations

import threading
import time
from typing import Callable, Optional

from .client import ApiError, K8sClient

EventFn = Callable[[str, dict], None]  # ("ADDED"|"MODIFIED"|"DELETED", wire obj)


class Informer:
    def __init__(self, client: K8sClient, kind: str, namespace: str = "",
                 label_selector: str = "", on_event: Optional[EventFn] = None):
        self.client = client
        self.kind = kind
        self.namespace = namespace
        self.label_selector = label_selector
        self.on_event = on_event
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.synced = threading.Event()

    def start(self):
        self._thread = threading.Thread(
            target=self._run, daemon=True, name=f"informer-{self.kind}")
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()

    def wait_synced(self, timeout: float = 30.0) -> bool:
        return self.synced.wait(timeout)

    # ------------------------------------------------------------- loop

    def _relist(self) -> str:
        lst = self.client.list(self.kind, self.namespace,
                               self.label_selector)
        for item in lst.get("items", []):
            item.setdefault("kind", self.kind)
            self._deliver("ADDED", item)
        self.synced.set()
        return lst.get("metadata", {}).get("resourceVersion", "")

    def _deliver(self, typ: str, obj: dict):
        if self.on_event is None:
            return
        try:
            self.on_event(typ, obj)
        except Exception:
            import traceback
            traceback.print_exc()

    def _run(self):
        rv = ""
        while not self._stop.is_set():
            try:
                if not rv:
                    rv = self._relist()
                for typ, obj in self.client.watch(
                        self.kind, self.namespace, resource_version=rv,
                        label_selector=self.label_selector, timeout_s=300):
                    obj.setdefault("kind", self.kind)
                    self._deliver(typ, obj)
                    new_rv = obj.get("metadata", {}).get("resourceVersion")
                    if new_rv:
                        rv = new_rv
                    if self._stop.is_set():
                        return
            except ApiError as e:
                if e.gone:
                    rv = ""  # relist
                else:
                    time.sleep(1.0)
            except Exception:
                time.sleep(1.0)
