"""Single-node backend — workers are OS processes, no cluster.

Reference: pkg/hypervisor/backend/single_node/single_node_backend.go:47-823
(JSON file state, spawn/stop/restart with exponential backoff, env merge,
reconcile loop). This is how a bare-metal MI355X box runs fractional vGPUs
without Kubernetes — also the path bench/e2e GPU tests exercise.
"""
from __future__ import annotations

import json
import os
import signal
import subprocess
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ... import constants as C
from ..allocation import WorkerSpec
from ..worker import WorkerController


@dataclass
class ProcessSpec:
    spec: WorkerSpec
    command: List[str]
    env: Dict[str, str] = field(default_factory=dict)
    preload_limiter: bool = True
    restart: bool = True


@dataclass
class ProcessState:
    pspec: ProcessSpec
    proc: Optional[subprocess.Popen] = None
    restarts: int = 0
    backoff_until: float = 0.0
    phase: str = "Pending"  # Pending|Running|Exited|Failed


class SingleNodeBackend:
    MAX_BACKOFF_S = 60.0

    def __init__(self, workers: WorkerController,
                 state_file: Optional[str] = None,
                 limiter_path: Optional[str] = None):
        self.workers = workers
        self.state_file = state_file
        self.limiter_path = limiter_path or os.path.join(
            os.path.dirname(os.path.dirname(os.path.dirname(
                os.path.abspath(__file__)))), "_native", C.LimiterLibName)
        self.procs: Dict[str, ProcessState] = {}
        self._mu = threading.RLock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        if state_file and os.path.exists(state_file):
            self._load_state()

    # ---------------------------------------------------------- lifecycle

    def add(self, pspec: ProcessSpec) -> ProcessState:
        with self._mu:
            st = self.procs.get(pspec.spec.key)
            if st is None:
                st = ProcessState(pspec=pspec)
                self.procs[pspec.spec.key] = st
            self._save_state()
            return st

    def remove(self, key: str):
        with self._mu:
            st = self.procs.pop(key, None)
        if st and st.proc and st.proc.poll() is None:
            try:
                os.killpg(os.getpgid(st.proc.pid), signal.SIGTERM)
            except (ProcessLookupError, PermissionError):
                st.proc.terminate()
            try:
                st.proc.wait(timeout=5)
            except subprocess.TimeoutExpired:
                st.proc.kill()
        self.workers.remove_worker(key)
        self._save_state()

    def _spawn(self, st: ProcessState):
        wst = self.workers.add_worker(st.pspec.spec)
        env = dict(os.environ)
        env.update(wst.allocation.env)
        env.update(st.pspec.env)
        if st.pspec.preload_limiter and os.path.exists(self.limiter_path):
            prev = env.get("LD_PRELOAD", "")
            env["LD_PRELOAD"] = (self.limiter_path + (" " + prev if prev else ""))
        st.proc = subprocess.Popen(st.pspec.command, env=env,
                                   start_new_session=True)
        st.phase = "Running"
        self.workers.register_pid(st.pspec.spec.key, st.proc.pid)

    # ------------------------------------------------------------ reconcile

    def reconcile_once(self):
        with self._mu:
            states = list(self.procs.values())
        now = time.time()
        for st in states:
            if st.proc is None:
                if now >= st.backoff_until:
                    try:
                        self._spawn(st)
                    except Exception:
                        st.phase = "Failed"
                        st.restarts += 1
                        st.backoff_until = now + min(
                            self.MAX_BACKOFF_S, 2.0 ** st.restarts)
                continue
            rc = st.proc.poll()
            if rc is None:
                continue
            st.phase = "Exited" if rc == 0 else "Failed"
            st.proc = None
            if st.pspec.restart and rc != 0:
                st.restarts += 1
                st.backoff_until = now + min(self.MAX_BACKOFF_S,
                                             2.0 ** st.restarts)
            elif rc == 0:
                st.pspec.restart = False

    def start(self, interval_s: float = 1.0):
        def loop():
            while not self._stop.wait(interval_s):
                self.reconcile_once()
                self.workers.sync_once()
        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
        for key in list(self.procs):
            self.remove(key)

    # --------------------------------------------------------------- state

    def _save_state(self):
        if not self.state_file:
            return
        data = []
        for st in self.procs.values():
            s = st.pspec.spec
            data.append({
                "namespace": s.namespace, "name": s.name,
                "gpu_uuids": s.gpu_uuids, "isolation": s.isolation,
                "qos": s.qos, "tflops_limit": s.tflops_limit,
                "vram_limit": s.vram_limit,
                "compute_percent_limit": s.compute_percent_limit,
                "command": st.pspec.command, "env": st.pspec.env,
                "restart": st.pspec.restart,
            })
        os.makedirs(os.path.dirname(self.state_file), exist_ok=True)
        tmp = self.state_file + ".tmp"
        with open(tmp, "w") as f:
            json.dump(data, f)
        os.replace(tmp, self.state_file)

    def _load_state(self):
        try:
            with open(self.state_file) as f:
                data = json.load(f)
        except (OSError, ValueError):
            return
        for rec in data:
            spec = WorkerSpec(
                namespace=rec["namespace"], name=rec["name"],
                gpu_uuids=rec["gpu_uuids"], isolation=rec["isolation"],
                qos=rec["qos"], tflops_limit=rec["tflops_limit"],
                vram_limit=rec["vram_limit"],
                compute_percent_limit=rec["compute_percent_limit"])
            self.procs[spec.key] = ProcessState(pspec=ProcessSpec(
                spec=spec, command=rec["command"], env=rec.get("env", {}),
                restart=rec.get("restart", True)))
