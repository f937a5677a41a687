"""Workload autoscaler — VPA-style vertical recommendations.

Reference: internal/autoscaler/autoscaler.go:47-239 (per-workload loop)
with three recommenders: percentile_recommender.go:66-500 (decaying
histogram percentile of TSDB usage + margin), cron_recommender.go
(time-windowed rules) and external_recommender.go (webhook). The chosen
recommendation lands in workload.status.recommendation; the webhook
applies it to new pods, and AdjustAllocation applies it in place.
"""
from __future__ import annotations

import json
import threading
import time
import urllib.request
from dataclasses import dataclass
from typing import Dict, Optional

from .. import constants as C
from ..api.store import Store
from ..api.types import (Recommendation, Requirements, Resource,
                         TensorFusionWorkload)

# ------------------------------------------------------- histogram


class DecayingHistogram:
    """Exponentially-decaying bucketed histogram (the reference's VPA
    histogram: buckets grow 5% geometrically, half-life decay)."""

    FIRST_BUCKET = 0.01
    GROWTH = 1.05
    N_BUCKETS = 400

    def __init__(self, half_life_s: float = 3600.0):
        self.weights = [0.0] * self.N_BUCKETS
        self.half_life_s = half_life_s
        self.ref_ts = time.time()
        self.total = 0.0

    def _bucket(self, v: float) -> int:
        if v <= self.FIRST_BUCKET:
            return 0
        import math
        i = int(math.log(v / self.FIRST_BUCKET) / math.log(self.GROWTH)) + 1
        return min(i, self.N_BUCKETS - 1)

    def _bucket_value(self, i: int) -> float:
        return self.FIRST_BUCKET * (self.GROWTH ** i)

    def add(self, value: float, ts: Optional[float] = None, weight: float = 1.0):
        ts = ts if ts is not None else time.time()
        decay = 2.0 ** ((ts - self.ref_ts) / self.half_life_s)
        self.weights[self._bucket(value)] += weight * decay
        self.total += weight * decay
        if decay > 1e6:  # re-normalize to keep floats sane
            for i in range(self.N_BUCKETS):
                self.weights[i] /= decay
            self.total /= decay
            self.ref_ts = ts

    def percentile(self, p: float) -> float:
        if self.total <= 0:
            return 0.0
        target = self.total * p
        acc = 0.0
        for i, w in enumerate(self.weights):
            acc += w
            if acc >= target:
                return self._bucket_value(i)
        return self._bucket_value(self.N_BUCKETS - 1)


# ----------------------------------------------------- recommenders


class PercentileRecommender:
    """Reference percentile_recommender.go:66-500."""

    name = "percentile"

    def __init__(self, tsdb, half_life_s: float = 3600.0):
        self.tsdb = tsdb
        self.half_life_s = half_life_s
        self._hist: Dict[str, Dict[str, DecayingHistogram]] = {}
        self._last_ts: Dict[str, int] = {}

    def recommend(self, wl: TensorFusionWorkload) -> Optional[Recommendation]:
        cfg = wl.profile.auto_scaling
        key = wl.meta.key
        hists = self._hist.setdefault(key, {
            "tflops": DecayingHistogram(self.half_life_s),
            "vram": DecayingHistogram(self.half_life_s)})
        last = self._last_ts.get(key, 0)
        pts_t = self.tsdb.query("tf_worker_metrics", "compute_tflops",
                                tags={"workload": wl.meta.name})
        pts_v = self.tsdb.query("tf_worker_metrics", "vram_bytes",
                                tags={"workload": wl.meta.name})
        newest = last
        for ts, v in pts_t:
            if ts > last:
                hists["tflops"].add(v, ts / 1e9)
                newest = max(newest, ts)
        for ts, v in pts_v:
            if ts > last:
                hists["vram"].add(v / (1 << 30), ts / 1e9)  # GiB buckets
                newest = max(newest, ts)
        self._last_ts[key] = newest
        if hists["tflops"].total <= 0 and hists["vram"].total <= 0:
            return None
        p = cfg.target_percentile or 0.9
        margin = 1.0 + (cfg.margin or 0.15)
        tflops = hists["tflops"].percentile(p) * margin
        vram = int(hists["vram"].percentile(p) * margin * (1 << 30))
        rec = Recommendation(
            resources=Requirements(
                requests=Resource(tflops=tflops, vram=vram),
                limits=Resource(tflops=tflops * 2, vram=vram)),
            reason=f"percentile p{int(p * 100)} x{margin:.2f}",
            ts=time.time())
        return rec


class CronRecommender:
    """Time-windowed static rules: {"start": "HH:MM", "end": "HH:MM",
    "tflops": x, "vram": bytes} (reference cron_recommender.go)."""

    name = "cron"

    def recommend(self, wl: TensorFusionWorkload) -> Optional[Recommendation]:
        rules = wl.profile.auto_scaling.cron_rules
        if not rules:
            return None
        now = time.localtime()
        cur = now.tm_hour * 60 + now.tm_min
        for r in rules:
            try:
                sh, sm = map(int, str(r["start"]).split(":"))
                eh, em = map(int, str(r["end"]).split(":"))
            except (KeyError, ValueError):
                continue
            start, end = sh * 60 + sm, eh * 60 + em
            inside = (start <= cur < end) if start <= end else \
                (cur >= start or cur < end)
            if inside:
                res = Resource(tflops=float(r.get("tflops", 0)),
                               vram=int(r.get("vram", 0)))
                return Recommendation(
                    resources=Requirements(requests=res, limits=res),
                    reason=f"cron window {r['start']}-{r['end']}",
                    ts=time.time())
        return None


class ExternalRecommender:
    """POST workload JSON to an external webhook; expects
    {"tflops": x, "vram": y} (reference external_recommender.go)."""

    name = "external"

    def recommend(self, wl: TensorFusionWorkload) -> Optional[Recommendation]:
        url = wl.profile.auto_scaling.external_url
        if not url:
            return None
        body = json.dumps({
            "workload": wl.meta.name, "namespace": wl.meta.namespace,
            "replicas": wl.replicas,
            "current": {"tflops": wl.profile.resources.requests.tflops,
                        "vram": wl.profile.resources.requests.vram},
        }).encode()
        req = urllib.request.Request(
            url, data=body, headers={"Content-Type": "application/json"})
        try:
            with urllib.request.urlopen(req, timeout=5) as resp:
                data = json.loads(resp.read())
        except Exception:
            return None
        res = Resource(tflops=float(data.get("tflops", 0)),
                       vram=int(data.get("vram", 0)))
        return Recommendation(resources=Requirements(requests=res, limits=res),
                              reason="external", ts=time.time())


# ----------------------------------------------------------- autoscaler


class Autoscaler:
    def __init__(self, store: Store, tsdb=None, allocator=None,
                 apply_in_place: bool = True):
        self.store = store
        self.allocator = allocator
        self.apply_in_place = apply_in_place
        self.recommenders = {
            "percentile": PercentileRecommender(tsdb) if tsdb else None,
            "cron": CronRecommender(),
            "external": ExternalRecommender(),
        }
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def tick(self) -> int:
        """One pass over autoscaling-enabled workloads."""

        n = 0
        for wl in self.store.list("TensorFusionWorkload"):
            if not wl.profile.auto_scaling.enabled:
                continue
            rec = self._recommend(wl)
            if rec is None:
                continue
            n += 1

            def _p(obj, rec=rec):
                obj.status.recommendation = rec
            self.store.patch("TensorFusionWorkload", wl.meta.name,
                             wl.meta.namespace, _p)
            if self.apply_in_place and self.allocator is not None:
                self._apply(wl, rec)
        return n

    def _recommend(self, wl) -> Optional[Recommendation]:
        r = self.recommenders.get(wl.profile.auto_scaling.recommender)
        if r is None:
            return None
        return r.recommend(wl)

    def _apply(self, wl, rec: Recommendation):
        """Vertical scaling via allocator.adjust_allocation (reference
        AdjustAllocation gpuallocator.go:1864)."""

        for pod_name in wl.status.worker_pods:
            key = f"{wl.meta.namespace}/{pod_name}"
            try:
                self.allocator.adjust_allocation(
                    key, rec.resources.requests, rec.resources.limits)
            except Exception:
                continue

    def start(self, interval_s: float = 30.0):
        self._stop.clear()

        def loop():
            while not self._stop.wait(interval_s):
                try:
                    self.tick()
                except Exception:
                    pass
        self._thread = threading.Thread(target=loop, daemon=True,
                                        name="autoscaler")
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
