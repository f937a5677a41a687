"""Client-plane runtime: launch vGPU workers and compose client env.

The remote flow (reference §3.1): client pod's stub resolves its
TensorFusionConnection → connectionURL `native+<ip>+<port>+<worker>-<rev>`
via the operator /connection endpoint, then attaches. Same-node transport
is the shm ring (native/remoting/protocol.h); the socket path carries the
bootstrap handshake.
"""
from __future__ import annotations

import os
import re
import subprocess
import time
from dataclasses import dataclass
from typing import Dict, Optional

from .. import constants as C

_NATIVE = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                       "_native")

URL_RE = re.compile(r"^native\+(?P<ip>[^+]+)\+(?P<port>\d+)\+(?P<worker>.+)$")


def parse_connection_url(url: str) -> Dict[str, str]:
    m = URL_RE.match(url)
    if not m:
        raise ValueError(f"bad connection url {url!r}")
    d = m.groupdict()
    worker = d["worker"]
    if "-" in worker:
        name, _, rev = worker.rpartition("-")
        if rev.isdigit():
            d["worker"], d["rev"] = name, rev
    return d


@dataclass
class WorkerHandle:
    proc: subprocess.Popen
    socket_path: str

    def stop(self, timeout: float = 5.0):
        if self.proc.poll() is None:
            self.proc.terminate()
            try:
                self.proc.wait(timeout=timeout)
            except subprocess.TimeoutExpired:
                self.proc.kill()
        try:
            os.unlink(self.socket_path)
        except OSError:
            pass


def start_worker(socket_path: str, device_index: int = 0,
                 env: Optional[Dict[str, str]] = None,
                 oneshot: bool = False, wait_s: float = 30.0,
                 snapshot_path: Optional[str] = None,
                 tcp_port: Optional[int] = None) -> WorkerHandle:
    """Spawn tf_vgpu_worker bound to one GPU; waits for its unix socket,
    or for the TCP port when tcp_port is given (cross-node mode)."""

    exe = os.path.join(_NATIVE, "tf_vgpu_worker")
    if not os.path.exists(exe):
        raise FileNotFoundError(f"{exe} not built (python build_native.py)")
    e = dict(os.environ)
    e["HIP_VISIBLE_DEVICES"] = str(device_index)
    e["ROCR_VISIBLE_DEVICES"] = str(device_index)
    if snapshot_path:
        e["TF_WORKER_SNAPSHOT_PATH"] = snapshot_path
    if oneshot:
        e["TF_WORKER_ONESHOT"] = "1"
    if tcp_port is not None:
        e["TF_WORKER_TCP_PORT"] = str(tcp_port)
    e.update(env or {})
    args = [exe] + ([socket_path] if socket_path else [])
    if socket_path:
        os.makedirs(os.path.dirname(socket_path) or ".", exist_ok=True)
    proc = subprocess.Popen(args, env=e)
    deadline = time.time() + wait_s
    while time.time() < deadline:
        if tcp_port is not None:
            import socket as _socket
            try:
                with _socket.create_connection(("127.0.0.1", tcp_port),
                                               timeout=0.2):
                    return WorkerHandle(proc=proc,
                                        socket_path=socket_path or "")
            except OSError:
                pass
        elif os.path.exists(socket_path):
            return WorkerHandle(proc=proc, socket_path=socket_path)
        if proc.poll() is not None:
            raise RuntimeError(
                f"tf_vgpu_worker exited rc={proc.returncode} before listening")
        time.sleep(0.05)
    proc.kill()
    raise TimeoutError(f"worker {socket_path or tcp_port} never came up")


def client_env(socket_path: str, base: Optional[Dict[str, str]] = None,
               debug: bool = False,
               tcp: Optional[str] = None) -> Dict[str, str]:
    """Environment for a GPU-less client process driving a remote vGPU.
    `tcp="host:port"` selects the cross-node TCP transport."""

    e = dict(base if base is not None else os.environ)
    client_lib = os.path.join(_NATIVE, C.ClientLibName)
    prev = e.get("LD_PRELOAD", "")
    e["LD_PRELOAD"] = client_lib + (" " + prev if prev else "")
    if tcp:
        e["TF_WORKER_TCP"] = tcp
        e.pop("TF_WORKER_SOCKET", None)
    else:
        e["TF_WORKER_SOCKET"] = socket_path
    # the client process must NOT see a GPU: remoting is the only path
    e["HIP_VISIBLE_DEVICES"] = ""
    e["ROCR_VISIBLE_DEVICES"] = ""
    if debug:
        e["TF_CLIENT_DEBUG"] = "1"
    return e


def snapshot_and_stop(handle: WorkerHandle, snapshot_path: str,
                      timeout_s: float = 120.0) -> str:
    """Quiesce + snapshot the worker's device state (VA-stable), then let
    the process exit. The worker must have been started with
    TF_WORKER_SNAPSHOT_PATH pointing at `snapshot_path` (start_worker's
    snapshot_path parameter does this)."""

    import signal as _signal
    handle.proc.send_signal(_signal.SIGUSR1)
    handle.proc.wait(timeout=timeout_s)
    if not os.path.exists(snapshot_path):
        raise RuntimeError(
            f"worker exited rc={handle.proc.returncode} without writing "
            f"{snapshot_path}")
    return snapshot_path


def migrate_worker(handle: WorkerHandle, snapshot_path: str,
                   new_device_index: int = 0,
                   env: Optional[Dict[str, str]] = None) -> WorkerHandle:
    """Live-migrate a vGPU: snapshot worker A, start worker B on
    `new_device_index` restoring that snapshot on the SAME socket path.
    The client process re-attaches automatically (its reconnect thread
    re-sends the shared segment); device pointers stay valid because the
    worker's VMM heap is re-reserved at the same VA base."""

    sock = handle.socket_path
    snapshot_and_stop(handle, snapshot_path)
    e = dict(env or {})
    e["TF_WORKER_RESTORE_PATH"] = snapshot_path
    return start_worker(sock, device_index=new_device_index, env=e)
