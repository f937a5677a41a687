"""Dynamic config — hot-reloaded YAML, GPU info tables, scheduler weights.

Reference: internal/config/ — global_config.go (fsnotify-watched
config.yaml: metrics TTL, alert rules, preemption mode), gpu_info.go:7-59
(model → fp16 TFLOPS/cost + partition templates), scheduler_config.go
(GPUFitConfig vram/tflops weights, topology mode). The watch here is a
poll on mtime (no inotify dependency); consumers register callbacks.
"""
from __future__ import annotations

import os
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional

import yaml

from . import constants as C


@dataclass
class GPUFitConfig:
    vram_weight: float = 0.7
    tflops_weight: float = 0.3
    placement_mode: str = "CompactFirst"
    topo_mode: str = "soft"  # soft | hard
    max_worker_per_node: int = C.MaxWorkersPerNode


@dataclass
class GpuInfo:
    model: str = C.MI355X_MODEL
    vendor: str = "AMD"
    fp16_tflops: float = C.MI355X_BF16_TFLOPS
    vram: int = C.MI355X_VRAM_BYTES
    compute_units: int = C.MI355X_CUS
    cost_per_hour: float = 4.0


@dataclass
class GlobalConfig:
    metrics_ttl_s: float = 7 * 86400
    metrics_interval_s: float = 60.0
    alert_rules: List[Dict[str, Any]] = field(default_factory=list)
    preemption_enabled: bool = True
    erl: Dict[str, float] = field(default_factory=dict)
    gpu_fit: GPUFitConfig = field(default_factory=GPUFitConfig)
    gpu_info: Dict[str, GpuInfo] = field(default_factory=lambda: {
        C.MI355X_MODEL: GpuInfo()})


def _load_yaml(path: str) -> dict:
    with open(path) as f:
        return yaml.safe_load(f) or {}


def parse_config(data: dict) -> GlobalConfig:
    cfg = GlobalConfig()
    cfg.metrics_ttl_s = float(data.get("metricsTTLSeconds", cfg.metrics_ttl_s))
    cfg.metrics_interval_s = float(
        data.get("metricsIntervalSeconds", cfg.metrics_interval_s))
    cfg.alert_rules = data.get("alertRules", [])
    cfg.preemption_enabled = bool(
        data.get("preemptionEnabled", cfg.preemption_enabled))
    cfg.erl = dict(data.get("erl", {}))
    gf = data.get("gpuFit", {})
    cfg.gpu_fit = GPUFitConfig(
        vram_weight=float(gf.get("vramWeight", 0.7)),
        tflops_weight=float(gf.get("tflopsWeight", 0.3)),
        placement_mode=gf.get("placementMode", "CompactFirst"),
        topo_mode=gf.get("topoMode", "soft"),
        max_worker_per_node=int(gf.get("maxWorkerPerNode",
                                       C.MaxWorkersPerNode)))
    for m in data.get("gpuInfo", []):
        info = GpuInfo(
            model=m.get("model", C.MI355X_MODEL),
            vendor=m.get("vendor", "AMD"),
            fp16_tflops=float(m.get("fp16TFlops", C.MI355X_BF16_TFLOPS)),
            vram=int(m.get("vramBytes", C.MI355X_VRAM_BYTES)),
            compute_units=int(m.get("computeUnits", C.MI355X_CUS)),
            cost_per_hour=float(m.get("costPerHour", 4.0)))
        cfg.gpu_info[info.model] = info
    return cfg


class ConfigWatcher:
    """Poll-based hot reload (reference utils.WatchConfigFileChanges)."""

    def __init__(self, path: str, poll_s: float = 1.0):
        self.path = path
        self.poll_s = poll_s
        self.config = GlobalConfig()
        self._mtime = 0.0
        self._callbacks: List[Callable[[GlobalConfig], None]] = []
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.reload()

    def on_change(self, cb: Callable[[GlobalConfig], None]):
        self._callbacks.append(cb)

    def reload(self) -> bool:
        try:
            mtime = os.path.getmtime(self.path)
        except OSError:
            return False
        if mtime == self._mtime:
            return False
        self._mtime = mtime
        try:
            self.config = parse_config(_load_yaml(self.path))
        except Exception:
            return False
        for cb in self._callbacks:
            try:
                cb(self.config)
            except Exception:
                pass
        return True

    def start(self):
        self._stop.clear()

        def loop():
            while not self._stop.wait(self.poll_s):
                self.reload()
        self._thread = threading.Thread(target=loop, daemon=True,
                                        name="config-watcher")
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
