"""vGPU worker process manager — remote-mode workers under the hypervisor.

The reference's worker pods run the closed-source vgpu worker; here the
hypervisor owns `tf_vgpu_worker` processes directly (single-node mode) or
through worker pods (store backend), and implements the snapshot/resume
HTTP surface that the reference stubs with 501: snapshot quiesces the
worker (SIGUSR1 → VA-stable device dump, see native/remoting) and
migrate restarts it on another device with the client live.
"""
from __future__ import annotations

import os
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, Optional

from ..client.runtime import (WorkerHandle, migrate_worker, snapshot_and_stop,
                              start_worker)


@dataclass
class VgpuWorker:
    key: str  # namespace/pod
    socket_path: str
    device_index: int
    snapshot_path: str
    handle: Optional[WorkerHandle] = None
    started_ts: float = field(default_factory=time.time)
    migrations: int = 0


class VgpuWorkerManager:
    def __init__(self, run_dir: str = "/run/tensor-fusion/vgpu"):
        self.run_dir = run_dir
        self._mu = threading.Lock()
        self.workers: Dict[str, VgpuWorker] = {}
        os.makedirs(run_dir, exist_ok=True)

    def start(self, key: str, device_index: int,
              env: Optional[dict] = None) -> VgpuWorker:
        safe = key.replace("/", "_")
        sock = os.path.join(self.run_dir, f"{safe}.sock")
        snap = os.path.join(self.run_dir, f"{safe}.snap")
        with self._mu:
            if key in self.workers and self.workers[key].handle and \
                    self.workers[key].handle.proc.poll() is None:
                return self.workers[key]
            w = VgpuWorker(key=key, socket_path=sock, device_index=device_index,
                           snapshot_path=snap)
            w.handle = start_worker(sock, device_index=device_index,
                                    env=env, snapshot_path=snap)
            self.workers[key] = w
            return w

    def stop(self, key: str):
        with self._mu:
            w = self.workers.pop(key, None)
        if w and w.handle:
            w.handle.stop()

    def snapshot(self, key: str) -> str:
        """Quiesce + dump device state; worker exits (frozen-to-disk)."""

        with self._mu:
            w = self.workers.get(key)
        if w is None or w.handle is None:
            raise KeyError(f"no vGPU worker {key}")
        return snapshot_and_stop(w.handle, w.snapshot_path)

    def resume(self, key: str, device_index: Optional[int] = None
               ) -> VgpuWorker:
        """Restart from the worker's snapshot (same socket — a live client
        re-attaches automatically)."""

        with self._mu:
            w = self.workers.get(key)
        if w is None:
            raise KeyError(f"no vGPU worker {key}")
        dev = device_index if device_index is not None else w.device_index
        w.handle = start_worker(
            w.socket_path, device_index=dev,
            env={"TF_WORKER_RESTORE_PATH": w.snapshot_path},
            snapshot_path=w.snapshot_path)
        w.device_index = dev
        w.migrations += 1
        return w

    def migrate(self, key: str, new_device_index: int) -> VgpuWorker:
        """Live migration: snapshot on the current device, restore on
        `new_device_index` under the live client."""

        with self._mu:
            w = self.workers.get(key)
        if w is None or w.handle is None:
            raise KeyError(f"no vGPU worker {key}")
        w.handle = migrate_worker(w.handle, w.snapshot_path,
                                  new_device_index=new_device_index)
        w.device_index = new_device_index
        w.migrations += 1
        return w

    def status(self) -> Dict[str, dict]:
        with self._mu:
            return {k: {
                "socket": w.socket_path,
                "device": w.device_index,
                "alive": bool(w.handle and w.handle.proc.poll() is None),
                "migrations": w.migrations,
            } for k, w in self.workers.items()}
