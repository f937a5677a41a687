"""Metrics recorder + pluggable encoders + billing.

Reference: internal/metrics/recorder.go:48-913 (in-memory tables →
periodic encode to a rolling file consumed by Vector → GreptimeDB),
encoders/strategy.go:5-13 (influx line protocol default, json, otel),
billing rawCost from QoS pricing recorder.go:852-913. The embedded TSDB
(tsdb.py) replaces the external GreptimeDB for the autoscaler/alert read
path in single-node deployments.
"""
from __future__ import annotations

import json
import os
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .. import constants as C

# ------------------------------------------------------------- tables
# Reference types.go:9-168 worker/node/pool/system metrics tables.


@dataclass
class WorkerMetrics:
    workload: str = ""
    worker: str = ""
    namespace: str = ""
    pool: str = ""
    qos: str = C.QosMedium
    device_uuid: str = ""
    compute_percent: float = 0.0  # busy %
    compute_tflops: float = 0.0
    vram_bytes: int = 0
    vram_percent: float = 0.0
    tokens_consumed: float = 0.0
    throttled_ratio: float = 0.0
    raw_cost_per_hour: float = 0.0


@dataclass
class NodeMetrics:
    node: str = ""
    pool: str = ""
    gpu_count: int = 0
    allocated_tflops: float = 0.0
    allocated_vram: int = 0
    allocated_tflops_percent: float = 0.0
    allocated_vram_percent: float = 0.0
    gpu_busy_percent: float = 0.0
    vram_used_bytes: int = 0
    power_watts: float = 0.0
    temperature_c: float = 0.0


@dataclass
class PoolMetrics:
    pool: str = ""
    node_count: int = 0
    gpu_count: int = 0
    total_tflops: float = 0.0
    total_vram: int = 0
    allocated_tflops: float = 0.0
    allocated_vram: int = 0
    assigned_workers: int = 0


@dataclass
class SchedulerMetrics:
    scheduled: int = 0
    unschedulable: int = 0
    preempted: int = 0
    gang_rejected: int = 0
    latency_ms_sum: float = 0.0
    latency_count: int = 0


# ------------------------------------------------------------ encoders


def _esc(s: str) -> str:
    return str(s).replace(" ", "\\ ").replace(",", "\\,").replace("=", "\\=")


class InfluxEncoder:
    """Influx line protocol (the reference's default wire format)."""

    name = "influx"

    def encode(self, measurement: str, tags: Dict[str, str],
               fields: Dict[str, float], ts_ns: int) -> str:
        t = ",".join(f"{_esc(k)}={_esc(v)}" for k, v in sorted(tags.items())
                     if v != "")
        f = ",".join(
            f"{_esc(k)}={v}i" if isinstance(v, int) and not isinstance(v, bool)
            else f"{_esc(k)}={float(v)}"
            for k, v in sorted(fields.items()))
        head = f"{measurement},{t}" if t else measurement
        return f"{head} {f} {ts_ns}"


class JsonEncoder:
    name = "json"

    def encode(self, measurement: str, tags, fields, ts_ns: int) -> str:
        return json.dumps({"m": measurement, "tags": tags, "fields": fields,
                           "ts": ts_ns}, sort_keys=True)


ENCODERS = {"influx": InfluxEncoder, "json": JsonEncoder}


# ------------------------------------------------------------- pricing


@dataclass
class QosPricingTable:
    """$ per unit-hour by QoS (reference qos pricing gpupool_types.go:336)."""

    tflops_per_hour: Dict[str, float] = field(default_factory=lambda: {
        C.QosLow: 0.02, C.QosMedium: 0.04, C.QosHigh: 0.08,
        C.QosCritical: 0.16})
    vram_gb_per_hour: Dict[str, float] = field(default_factory=lambda: {
        C.QosLow: 0.004, C.QosMedium: 0.008, C.QosHigh: 0.016,
        C.QosCritical: 0.032})

    def raw_cost_per_hour(self, qos: str, tflops: float, vram_bytes: int
                          ) -> float:
        """Reference recorder.go:852-913 getWorkerRawCost."""

        return (self.tflops_per_hour.get(qos, 0.04) * tflops
                + self.vram_gb_per_hour.get(qos, 0.008)
                * (vram_bytes / (1 << 30)))


# ------------------------------------------------------------- recorder


class MetricsRecorder:
    """In-memory maps, periodically encoded to a rolling file (and into the
    embedded TSDB when one is attached)."""

    def __init__(self, out_dir: str = "", encoder: str = "influx",
                 pricing: Optional[QosPricingTable] = None, tsdb=None,
                 max_file_bytes: int = 32 << 20, keep_files: int = 3,
                 remote_url: str = "", remote_db: str = "tensor_fusion",
                 remote_auth: str = ""):
        self.out_dir = out_dir
        self.encoder = ENCODERS[encoder]()
        self.pricing = pricing or QosPricingTable()
        self.tsdb = tsdb
        self.max_file_bytes = max_file_bytes
        self.keep_files = keep_files
        # external pipeline: the reference ships metric files via Vector
        # into GreptimeDB's influx-line HTTP write endpoint
        # (charts/templates/vector-config.yaml:36-65). Here the recorder
        # POSTs the same line protocol directly — remote_url is the
        # /v1/influxdb/write (GreptimeDB) or /api/v2/write (influx)
        # endpoint; failures never block the flush (buffered + retried
        # next cycle, bounded).
        self.remote_url = remote_url or os.environ.get(
            "TF_METRICS_REMOTE_URL", "")
        self.remote_db = remote_db
        self.remote_auth = remote_auth or os.environ.get(
            "TF_METRICS_REMOTE_AUTH", "")
        self._remote_backlog: List[str] = []
        self.remote_errors = 0
        self.remote_posts = 0
        self._lock = threading.Lock()
        self.workers: Dict[str, WorkerMetrics] = {}
        self.nodes: Dict[str, NodeMetrics] = {}
        self.pools: Dict[str, PoolMetrics] = {}
        self.scheduler = SchedulerMetrics()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        if out_dir:
            os.makedirs(out_dir, exist_ok=True)

    # ------------------------------------------------------------ write

    def set_worker(self, m: WorkerMetrics):
        m.raw_cost_per_hour = self.pricing.raw_cost_per_hour(
            m.qos, m.compute_tflops, m.vram_bytes)
        with self._lock:
            self.workers[f"{m.namespace}/{m.worker}/{m.device_uuid}"] = m

    def drop_worker(self, namespace: str, worker: str):
        with self._lock:
            for k in list(self.workers):
                if k.startswith(f"{namespace}/{worker}/"):
                    del self.workers[k]

    def set_node(self, m: NodeMetrics):
        with self._lock:
            self.nodes[m.node] = m

    def set_pool(self, m: PoolMetrics):
        with self._lock:
            self.pools[m.pool] = m

    def record_scheduled(self, latency_ms: float):
        with self._lock:
            self.scheduler.scheduled += 1
            self.scheduler.latency_ms_sum += latency_ms
            self.scheduler.latency_count += 1

    def record_unschedulable(self):
        with self._lock:
            self.scheduler.unschedulable += 1

    def record_preempted(self):
        with self._lock:
            self.scheduler.preempted += 1

    def record_gang_rejected(self):
        with self._lock:
            self.scheduler.gang_rejected += 1

    # ------------------------------------------------------------ flush

    def encode_all(self, ts_ns: Optional[int] = None) -> List[str]:
        ts = ts_ns if ts_ns is not None else time.time_ns()
        out = []
        with self._lock:
            for m in self.workers.values():
                out.append(self.encoder.encode(
                    "tf_worker_metrics",
                    {"workload": m.workload, "worker": m.worker,
                     "namespace": m.namespace, "pool": m.pool, "qos": m.qos,
                     "device": m.device_uuid},
                    {"compute_percent": m.compute_percent,
                     "compute_tflops": m.compute_tflops,
                     "vram_bytes": m.vram_bytes,
                     "vram_percent": m.vram_percent,
                     "tokens_consumed": m.tokens_consumed,
                     "throttled_ratio": m.throttled_ratio,
                     "raw_cost": m.raw_cost_per_hour}, ts))
            for n in self.nodes.values():
                out.append(self.encoder.encode(
                    "tf_node_metrics", {"node": n.node, "pool": n.pool},
                    {"gpu_count": n.gpu_count,
                     "allocated_tflops": n.allocated_tflops,
                     "allocated_vram": n.allocated_vram,
                     "allocated_tflops_percent": n.allocated_tflops_percent,
                     "allocated_vram_percent": n.allocated_vram_percent,
                     "gpu_busy_percent": n.gpu_busy_percent,
                     "vram_used": n.vram_used_bytes,
                     "power_watts": n.power_watts,
                     "temperature_c": n.temperature_c}, ts))
            for p in self.pools.values():
                out.append(self.encoder.encode(
                    "tf_pool_metrics", {"pool": p.pool},
                    {"node_count": p.node_count, "gpu_count": p.gpu_count,
                     "total_tflops": p.total_tflops,
                     "total_vram": p.total_vram,
                     "allocated_tflops": p.allocated_tflops,
                     "allocated_vram": p.allocated_vram,
                     "assigned_workers": p.assigned_workers}, ts))
            s = self.scheduler
            out.append(self.encoder.encode(
                "tf_system_metrics", {},
                {"scheduled": s.scheduled, "unschedulable": s.unschedulable,
                 "preempted": s.preempted, "gang_rejected": s.gang_rejected,
                 "avg_latency_ms": (s.latency_ms_sum / s.latency_count)
                 if s.latency_count else 0.0}, ts))
        return out

    def flush(self) -> int:
        lines = self.encode_all()
        if self.tsdb is not None:
            self.tsdb.ingest_lines(lines)
        if self.out_dir:
            path = os.path.join(self.out_dir, "metrics.log")
            self._rotate(path)
            with open(path, "a") as f:
                f.write("\n".join(lines) + "\n")
        if self.remote_url:
            self._ship_remote(lines)
        return len(lines)

    def _ship_remote(self, lines: List[str]):
        """POST influx line protocol to the external TSDB; backlog on
        failure (bounded to ~4k lines so a dead sink can't grow RAM)."""

        import requests
        payload = self._remote_backlog + lines
        if not payload:
            return
        headers = {"Content-Type": "text/plain"}
        if self.remote_auth:
            headers["Authorization"] = self.remote_auth
        try:
            r = requests.post(self.remote_url,
                              params={"db": self.remote_db},
                              data="\n".join(payload).encode(),
                              headers=headers, timeout=10)
            if r.status_code >= 300:
                raise RuntimeError(f"sink {r.status_code}")
            self._remote_backlog = []
            self.remote_posts += 1
        except Exception:
            self.remote_errors += 1
            self._remote_backlog = payload[-4096:]

    def _rotate(self, path: str):
        try:
            if os.path.getsize(path) < self.max_file_bytes:
                return
        except OSError:
            return
        for i in range(self.keep_files - 1, 0, -1):
            src = f"{path}.{i}" if i > 1 else path
            dst = f"{path}.{i + 1}" if i > 1 else f"{path}.1"
            if os.path.exists(src):
                os.replace(src, dst)

    # ------------------------------------------------------------- loop

    def start(self, interval_s: float = 60.0):
        self._stop.clear()

        def loop():
            while not self._stop.wait(interval_s):
                try:
                    self.flush()
                except Exception:
                    pass
        self._thread = threading.Thread(target=loop, daemon=True,
                                        name="metrics-recorder")
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
