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
    workers = WorkerController(devices, alloc, erl=erl, shm_root=shm_root)
    backend = None
    if store is not None:
        from .backend.store_backend import StoreBackend
        backend = StoreBackend(store, node, devices, workers, pool=pool)
    return devices, workers, erl, backend


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--node", default="node-0")
    ap.add_argument("--backend", default="single", choices=["single", "store"])
    ap.add_argument("--http-port", type=int, default=C.HypervisorHTTPPort)
    ap.add_argument("--mock-devices", type=int, default=0)
    ap.add_argument("--shm-root", default=C.ShmRoot)
    ap.add_argument("--store-dir", default="",
                    help="persist dir for store backend")
    args = ap.parse_args()

    store = None
    if args.backend == "store":
        from ..api.store import Store
        store = Store(persist_dir=args.store_dir or None)
    devices, workers, erl, backend = build_hypervisor(
        node=args.node, mock_devices=args.mock_devices,
        shm_root=args.shm_root, store=store)
    devices.start()
    erl.start()
    if backend:
        backend.start()

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
