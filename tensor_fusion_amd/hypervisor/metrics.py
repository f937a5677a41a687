"""Hypervisor-side metrics loop.

Reference: pkg/hypervisor/metrics/metrics.go:23-359 — 60 s loop encoding
node/device/worker usage as influx lines to a rolling file (shipped by
Vector in the reference; ingested by the embedded TSDB here when one is
attached).
"""
from __future__ import annotations

import threading
import time
from typing import Optional

from .. import constants as C
from ..metrics.recorder import MetricsRecorder, NodeMetrics, WorkerMetrics
from .device import DeviceController
from .worker import WorkerController


class HypervisorMetrics:
    def __init__(self, node: str, devices: DeviceController,
                 workers: WorkerController, out_dir: str = "",
                 tsdb=None, pool: str = ""):
        self.node = node
        self.pool = pool
        self.devices = devices
        self.workers = workers
        self.recorder = MetricsRecorder(out_dir=out_dir, tsdb=tsdb)
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def collect_once(self) -> int:
        total_busy = 0.0
        vram_used = 0
        devs = self.devices.devices()
        for d in devs:
            m = self.devices.metrics(d.index)
            if m:
                total_busy += float(m.gfx_activity)
                vram_used += int(m.vram_used)
        self.recorder.set_node(NodeMetrics(
            node=self.node, pool=self.pool, gpu_count=len(devs),
            gpu_busy_percent=total_busy / max(1, len(devs)),
            vram_used_bytes=vram_used))
        now_ns = time.time_ns()
        for w in self.workers.worker_metrics():
            key = w.get("worker", "/")
            ns, _, pod = key.partition("/")
            block_ns = int(w.get("block_ns", 0))
            self.recorder.set_worker(WorkerMetrics(
                workload=w.get("workload", ""),
                worker=pod or key,
                namespace=ns,
                pool=self.pool,
                qos=w.get("qos", C.QosMedium),
                device_uuid=w.get("device", ""),
                compute_percent=float(w.get("up_limit", 0)),
                vram_bytes=int(w.get("vram_used", 0)),
                tokens_consumed=float(w.get("launches", 0)),
                throttled_ratio=min(1.0, block_ns / max(1, now_ns))))
        return self.recorder.flush()

    def telemetry_ping(self) -> Optional[dict]:
        """Anonymous usage ping (reference metrics.go:40-46 PostHog).
        OPT-IN here: only fires when TF_TELEMETRY_URL is set and
        TF_TELEMETRY_DISABLED is not — no endpoint is baked in. Payload
        is aggregate-only (node/device counts, no names)."""

        import os
        url = os.environ.get("TF_TELEMETRY_URL", "")
        if not url or os.environ.get("TF_TELEMETRY_DISABLED"):
            return None
        payload = {
            "event": "hypervisor_heartbeat",
            "gpu_count": len(self.devices.devices()),
            "worker_count": len(self.workers.worker_metrics()),
            "version": C.Version,
        }
        try:
            import requests
            requests.post(url, json=payload, timeout=3)
        except Exception:
            pass  # telemetry must never disturb the node plane
        return payload

    def start(self, interval_s: float = 60.0):
        self._stop.clear()

        def loop():
            while not self._stop.wait(interval_s):
                try:
                    self.collect_once()
                    self.telemetry_ping()
                except Exception:
                    pass
        self._thread = threading.Thread(target=loop, daemon=True,
                                        name="hypervisor-metrics")
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
