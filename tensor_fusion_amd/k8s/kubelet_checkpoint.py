"""Kubelet device-manager checkpoint detector — device-plugin coexistence.

Reference: pkg/hypervisor/backend/kubernetes/external_dp/
kubelet_checkpoint.go:82-519 watches kubelet's device-manager checkpoint
file and marks GPUs already handed out by a foreign device plugin
(`usedBy: nvidia-device-plugin`) so the allocator never double-books them
— the progressive-migration path for clusters moving from a plain device
plugin onto tensor-fusion.

MI355X equivalents of the foreign resource names: `amd.com/gpu` (the AMD
device plugin) plus the NVIDIA name for mixed fleets. The checkpoint is
JSON (`kubelet_internal_checkpoint`): RegisteredDevices lists device IDs
per resource, PodDeviceEntries the per-pod grants.
"""
from __future__ import annotations

import json
import os
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Set

DEFAULT_CHECKPOINT = \
    "/var/lib/kubelet/device-plugins/kubelet_internal_checkpoint"

# resource-name prefixes owned by OTHER device plugins (reference
# nvdp_detector/generic_detector per-vendor prefixes)
FOREIGN_GPU_RESOURCES = ("amd.com/gpu", "nvidia.com/gpu")

EXTERNAL_USED_BY = "external-device-plugin"


@dataclass
class CheckpointState:
    """Parsed view of the checkpoint file."""

    registered: Dict[str, List[str]] = field(default_factory=dict)
    # device IDs granted to running pods, per resource
    in_use: Dict[str, Set[str]] = field(default_factory=dict)

    def foreign_gpu_ids(self) -> Set[str]:
        out: Set[str] = set()
        for res, ids in self.registered.items():
            if any(res.startswith(p) for p in FOREIGN_GPU_RESOURCES):
                out.update(ids)
        return out

    def foreign_in_use_ids(self) -> Set[str]:
        out: Set[str] = set()
        for res, ids in self.in_use.items():
            if any(res.startswith(p) for p in FOREIGN_GPU_RESOURCES):
                out.update(ids)
        return out


def parse_checkpoint(path: str) -> Optional[CheckpointState]:
    try:
        with open(path) as f:
            doc = json.load(f)
    except (OSError, ValueError):
        return None
    data = doc.get("Data") or {}
    st = CheckpointState()
    for res, ids in (data.get("RegisteredDevices") or {}).items():
        st.registered[res] = list(ids or [])
    for entry in data.get("PodDeviceEntries") or []:
        res = entry.get("ResourceName", "")
        ids = entry.get("DeviceIDs")
        flat: List[str] = []
        if isinstance(ids, dict):  # {numaNode: [ids]}
            for v in ids.values():
                flat.extend(v or [])
        elif isinstance(ids, list):
            flat.extend(ids)
        st.in_use.setdefault(res, set()).update(flat)
    return st


class CheckpointDetector:
    """Polls the checkpoint file and flips GPU CRs' used_by between
    "tensor-fusion" and EXTERNAL_USED_BY.

    Matching order per GPU: device UUID in the foreign ID set, else the
    device index as a trailing integer (AMD device plugin IDs are often
    card indexes or /dev/dri paths)."""

    def __init__(self, store, node: str, path: str = DEFAULT_CHECKPOINT,
                 interval_s: float = 10.0):
        self.store = store
        self.node = node
        self.path = path
        self.interval_s = interval_s
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._last_mtime = 0.0

    # ------------------------------------------------------------ logic

    @staticmethod
    def _matches(gpu, ids: Set[str]) -> bool:
        if gpu.status.uuid and gpu.status.uuid in ids:
            return True
        for i in ids:
            tail = i.rsplit("-", 1)[-1].rsplit("D", 1)[-1]
            if tail.isdigit() and int(tail) == gpu.status.index and (
                    i.startswith("card") or i.startswith("/dev/dri")
                    or i.startswith("gpu")):
                return True
        return False

    def sync_once(self) -> int:
        """One detection pass; returns the number of GPU CRs flipped."""

        st = parse_checkpoint(self.path)
        if st is None:
            return 0
        foreign = st.foreign_gpu_ids() | st.foreign_in_use_ids()
        flipped = 0
        for gpu in self.store.list("GPU"):
            if gpu.status.node != self.node:
                continue
            external = self._matches(gpu, foreign)
            want = EXTERNAL_USED_BY if external else "tensor-fusion"
            if gpu.status.used_by != want:
                def _set(o, w=want):
                    o.status.used_by = w
                self.store.patch("GPU", gpu.meta.name,
                                 gpu.meta.namespace, _set)
                flipped += 1
        return flipped

    # ------------------------------------------------------------- loop

    def start(self) -> "CheckpointDetector":
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="kubelet-checkpoint")
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()

    def _run(self):
        while not self._stop.wait(self.interval_s):
            try:
                mtime = os.path.getmtime(self.path)
            except OSError:
                continue
            if mtime != self._last_mtime:
                self._last_mtime = mtime
                try:
                    self.sync_once()
                except Exception:
                    pass
