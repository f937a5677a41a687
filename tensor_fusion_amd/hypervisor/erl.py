"""ERL — elastic rate limit: the PID loop governing soft isolation.

Reference: pkg/hypervisor/worker/computing/quota_controller.go:20-458 —
500 ms loop, per worker×device PID (kp .9 ki .35 kd .10, EMA α .25,
deadband 3%, slew +35/−25%, feed-forward), writes token refill
rate/capacity into the worker's shm page; the in-process limiter consumes
tokens per HIP call and sleeps when dry.

The plant: launches/s admitted → GPU utilization. Its gain varies by orders
of magnitude with kernel size, so the controller output is MULTIPLICATIVE
(rate scales by 1+u each tick) with slew clamps, plus a feed-forward reset
when a worker goes idle.
"""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Dict, Optional

from ..api.types import ElasticRateLimitParams
from . import shm as S
from .device import DeviceController


@dataclass
class PidState:
    integral: float = 0.0
    last_error: float = 0.0
    util_ema: Optional[float] = None
    rate: float = 1000.0


class PidController:
    """One worker×device rate controller (pure logic, unit-tested on CPU)."""

    def __init__(self, params: Optional[ElasticRateLimitParams] = None):
        self.p = params or ElasticRateLimitParams()
        self.state = PidState(rate=self.p.min_rate * 10)

    def step(self, setpoint_percent: float, util_percent: float,
             dt: float) -> float:
        """Returns the new token refill rate (tokens/second)."""

        p, s = self.p, self.state
        if s.util_ema is None:
            s.util_ema = util_percent
        else:
            s.util_ema = (p.ema_alpha * util_percent +
                          (1 - p.ema_alpha) * s.util_ema)
        error = setpoint_percent - s.util_ema  # positive → too slow → grow
        if abs(error) <= p.deadband_percent:
            s.last_error = error
            return s.rate
        # normalized PID (error in percent of setpoint)
        denom = max(setpoint_percent, 1.0)
        e = error / denom
        s.integral = max(-2.0, min(2.0, s.integral + e * dt))
        de = (e - s.last_error / denom) / dt if dt > 0 else 0.0
        u = p.kp * e + p.ki * s.integral + p.kd * de
        s.last_error = error
        # multiplicative update with slew clamps
        up = p.slew_up_percent / 100.0
        down = p.slew_down_percent / 100.0
        factor = 1.0 + max(-down, min(up, u))
        s.rate = max(p.min_rate, min(p.max_rate, s.rate * factor))
        return s.rate

    def feed_forward(self, launches_per_s: float, util_percent: float,
                     setpoint_percent: float):
        """Seed the rate from the observed launches↔util gain so a fresh or
        regime-changed worker converges in one or two ticks."""

        if util_percent > 1.0 and launches_per_s > 0:
            gain = launches_per_s / util_percent  # launches per util-%
            self.state.rate = max(self.p.min_rate, min(
                self.p.max_rate, gain * setpoint_percent))


@dataclass
class WorkerErl:
    shm: S.WorkerShm
    pids: Dict[int, PidController] = field(default_factory=dict)  # device idx
    last_launches: Dict[int, int] = field(default_factory=dict)
    last_busy_ns: Dict[int, int] = field(default_factory=dict)


class ErlQuotaController:
    """The hypervisor-side loop: measure per-worker device utilization via
    the accelerator lib, PID to the up_limit setpoint, write rate/capacity
    into each worker's shm."""

    BURST_S = 0.2  # bucket capacity = rate * BURST_S

    def __init__(self, devices: DeviceController,
                 params: Optional[ElasticRateLimitParams] = None):
        self.devices = devices
        self.params = params or ElasticRateLimitParams()
        self.workers: Dict[str, WorkerErl] = {}  # shm path -> state
        self._mu = threading.Lock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._last_tick = time.monotonic()

    def attach(self, shm_page: S.WorkerShm):
        with self._mu:
            self.workers[shm_page.path] = WorkerErl(shm=shm_page)

    def detach(self, path: str):
        with self._mu:
            self.workers.pop(path, None)

    # ------------------------------------------------------------- tick

    def tick(self, dt: Optional[float] = None):
        now = time.monotonic()
        if dt is None:
            dt = max(1e-3, now - self._last_tick)
        self._last_tick = now
        # one metrics read per device per tick
        dev_metrics = {}
        dev_procs = {}
        for d in self.devices.devices():
            try:
                dev_metrics[d.index] = self.devices.metrics(d.index)
            except Exception:
                continue
        with self._mu:
            workers = list(self.workers.values())
        for w in workers:
            try:
                self._tick_worker(w, dev_metrics, dev_procs, dt)
            except Exception:
                continue

    def _tick_worker(self, w: WorkerErl, dev_metrics, dev_procs, dt: float):
        w.shm.touch_hypervisor_heartbeat()
        n = w.shm.read_u32(S.OFF_DEVICE_COUNT)
        worker_pids = set(w.shm.pids())
        for i in range(min(n, S.MAX_DEVICES)):
            d = w.shm.device(i)
            if not d.active or d.up_limit_percent >= 100:
                continue
            dev_idx = self._device_index_for(d.uuid, i)
            m = dev_metrics.get(dev_idx)
            if m is None:
                continue
            util = float(m.gfx_activity)
            # attribute utilization to this worker when possible
            procs = dev_procs.get(dev_idx)
            if procs is None:
                procs = self.devices.processes(dev_idx)
                dev_procs[dev_idx] = procs
            if procs and worker_pids:
                busy = sum(p.gfx_busy_ns for p in procs
                           if p.pid in worker_pids)
                last = w.last_busy_ns.get(i)
                w.last_busy_ns[i] = busy
                if last is not None and busy >= last:
                    util_attr = 100.0 * (busy - last) / (dt * 1e9)
                    util = min(util, util_attr) if util_attr > 0 else util
            pid = w.pids.get(i)
            if pid is None:
                pid = w.pids[i] = PidController(self.params)
                launches = d.launch_count
                last_l = w.last_launches.get(i, launches)
                pid.feed_forward((launches - last_l) / dt, util,
                                 d.up_limit_percent)
            w.last_launches[i] = d.launch_count
            rate = pid.step(d.up_limit_percent, util, dt)
            w.shm.update_erl(i, rate, max(1.0, rate * self.BURST_S))

    def _device_index_for(self, uuid: str, fallback: int) -> int:
        for d in self.devices.devices():
            if d.uuid == uuid:
                return d.index
        return fallback

    # ------------------------------------------------------------- loop

    def start(self):
        def loop():
            while not self._stop.wait(self.params.loop_interval_s):
                self.tick()
        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
