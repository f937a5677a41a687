"""Hypervisor daemon entry (reference cmd/hypervisor/main.go:49-189).

    python -m tensor_fusion_amd.hypervisor.main \
        --node node-0 --backend single|store --http-port 8001 \
        [--mock-devices 8] [--shm-root /run/tensor-fusion/shm]
"""
from __future__ import annotations

import argparse
import signal
import threading

from .. import constants as C
from .allocation import AllocationController
from .device import Accelerator, DeviceController
from .erl import ErlQuotaController
from .worker import WorkerController


def build_hypervisor(node: str = "node-0", mock_devices: int = 0,
                     shm_root: str = C.ShmRoot, store=None, pool: str = ""):
    accel = Accelerator(mock_devices=mock_devices)
    devices = DeviceController(accel)
    alloc = AllocationController(devices, shm_root=shm_root)
    erl = ErlQuotaController(devices)
    from .pressure import PressureController

    def _free_fn():
        m = devices.metrics(0)
        total = m.vram_total or C.MI355X_VRAM_BYTES
        return max(total - m.vram_used, 0), total
    pressure = PressureController(_free_fn)
    workers = WorkerController(devices, alloc, erl=erl, shm_root=shm_root,
                               pressure=pressure)
    backend = None
    if store is not None:
        from .backend.store_backend import StoreBackend
        backend = StoreBackend(store, node, devices, workers, pool=pool)
    return devices, workers, erl, backend


def make_dp_resolver(store, node: str, workers):
    """kubelet device-plugin Allocate() resolver: pod index on this node →
    the composed worker environment (reference deviceplugin.go:250-338:
    Allocate keys off the index resource to return the per-pod env +
    device-node set)."""

    from ..k8s.deviceplugin import default_device_nodes

    def resolver(index: int):
        for pod in store.list("Pod"):
            if pod.meta.annotations.get(C.AnnoPodIndex) != str(index):
                continue
            if pod.status.node != node:
                continue
            alloc = workers.allocation_of(pod.meta.key) \
                if hasattr(workers, "allocation_of") else None
            if alloc is None:
                return None
            devices = []
            for dn in alloc.device_nodes:
                if dn == "/dev/dri":
                    devices.extend(d for d in default_device_nodes()
                                   if d["host_path"] != "/dev/kfd")
                else:
                    devices.append({"host_path": dn, "container_path": dn,
                                    "permissions": "rw"})
            return {
                "env": dict(alloc.env),
                "devices": devices,
                "annotations": {C.AnnoPodIndex: str(index)},
                "mounts": [{"host_path": C.DataRoot,
                            "container_path": C.DataRoot}],
            }
        return None
    return resolver


def mount_shm(path: str, size_mb: int = 64, dry_run: bool = False) -> str:
    """`mount-shm` init-container subcommand: mount a tmpfs at the
    limiter-shm root so worker pods and the hypervisor share pages via
    a hostPath (reference shm_init/mount_shm.go:17-92). Idempotent: if
    `path` is already a tmpfs mountpoint it is left alone. Returns the
    action taken ("mounted" | "already-mounted" | "dir-only")."""

    import os
    import subprocess
    os.makedirs(path, exist_ok=True)
    try:
        with open("/proc/mounts") as f:
            for line in f:
                parts = line.split()
                if len(parts) >= 3 and parts[1] == os.path.realpath(path) \
                        and parts[2] == "tmpfs":
                    return "already-mounted"
    except OSError:
        pass
    cmd = ["mount", "-t", "tmpfs", "-o",
           f"size={size_mb}m,mode=0755", "tensor-fusion-shm", path]
    if dry_run or os.geteuid() != 0:
        return "dir-only"  # tests / rootless: plain dir still works
    try:
        subprocess.run(cmd, check=True, capture_output=True)
        return "mounted"
    except (subprocess.CalledProcessError, FileNotFoundError):
        return "dir-only"


def main():
    import sys
    if len(sys.argv) > 1 and sys.argv[1] == "mount-shm":
        mp = argparse.ArgumentParser(prog="hypervisor mount-shm")
        mp.add_argument("--path", default=C.ShmRoot)
        mp.add_argument("--size-mb", type=int, default=64)
        a = mp.parse_args(sys.argv[2:])
        print(mount_shm(a.path, a.size_mb))
        return
    ap = argparse.ArgumentParser()
    ap.add_argument("--node", default="node-0")
    ap.add_argument("--backend", default="single",
                    choices=["single", "store", "kubernetes"])
    ap.add_argument("--http-port", type=int, default=C.HypervisorHTTPPort)
    ap.add_argument("--mock-devices", type=int, default=0)
    ap.add_argument("--shm-root", default=C.ShmRoot)
    ap.add_argument("--store-dir", default="",
                    help="persist dir for store backend")
    ap.add_argument("--device-plugin", action="store_true",
                    help="register tensor-fusion.ai/index-N device "
                         "plugins with kubelet (kubernetes backend)")
    ap.add_argument("--kubelet-socket",
                    default="/var/lib/kubelet/device-plugins/kubelet.sock")
    args = ap.parse_args()

    store = None
    if args.backend == "store":
        from ..api.store import Store
        store = Store(persist_dir=args.store_dir or None)
    elif args.backend == "kubernetes":
        # same StoreBackend logic, state plane = the kube-apiserver
        # (reference pkg/hypervisor/backend/kubernetes)
        from ..k8s.bridge import K8sStore
        from ..k8s.client import K8sClient
        store = K8sStore(K8sClient.auto()).start()
    devices, workers, erl, backend = build_hypervisor(
        node=args.node, mock_devices=args.mock_devices,
        shm_root=args.shm_root, store=store)
    devices.start()
    erl.start()
    workers.pressure.start()
    if backend:
        backend.start()

    dp_mgr = None
    ckpt = None
    if args.backend == "kubernetes":
        if args.device_plugin:
            import os
            from ..k8s.deviceplugin import DevicePluginManager
            dp_mgr = DevicePluginManager(
                make_dp_resolver(store, args.node, workers),
                socket_dir=os.path.dirname(args.kubelet_socket),
                kubelet_socket=args.kubelet_socket)
            dp_mgr.start()
        from ..k8s.kubelet_checkpoint import CheckpointDetector
        ckpt = CheckpointDetector(store, node=args.node).start()

    # periodic shm sync / orphan sweep
    stop = threading.Event()

    def sync_loop():
        while not stop.wait(5.0):
            workers.sync_once()

    threading.Thread(target=sync_loop, daemon=True).start()

    # metrics loop (reference pkg/hypervisor/metrics 60 s cadence)
    from .metrics import HypervisorMetrics
    hyp_metrics = HypervisorMetrics(args.node, devices, workers,
                                    out_dir="/var/log/tensor-fusion")
    hyp_metrics.start()

    from .server import attach_vgpu_manager, build_app
    from .vgpu_manager import VgpuWorkerManager
    import uvicorn
    app = build_app(devices, workers)
    attach_vgpu_manager(app, VgpuWorkerManager())
    signal.signal(signal.SIGTERM, lambda *_: stop.set())
    uvicorn.run(app, host="0.0.0.0", port=args.http_port, log_level="warning")


if __name__ == "__main__":
    main()
