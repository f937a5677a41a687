#!/usr/bin/env python3
"""BASELINE config 4 demo: N vGPUs whose caps oversubscribe one MI355X,
running concurrently with host-DRAM tiering + the hypervisor pressure
controller.

    python tools/demo_oversub.py --tenants 4 --cap-gb 96 --alloc-gb 85 \
        --hot-gb 8 --duration 40
    python tools/demo_oversub.py --scaled       # small CI shape

Per tenant (child process, LD_PRELOAD limiter, own shm page):
  - allocates ~alloc_gb of buffers ("model weights" + cold filler); the
    slice past the dynamic HBM budget lands in the managed host tier
  - runs a decode-like loop: matmuls over a hot working set, touching it
    so the LRU keeps it HBM-resident
  - reports iterations/s + tier stats (tf_limiter_tier_stats2)

Main process = the hypervisor role: creates the shm pages, runs
PressureController against the real device free-memory reading
(accelerator lib / amdsmi — no HIP context), prints the budget/pressure
trace and the per-tenant results, incl. slowdown vs a solo baseline.
"""
from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys
import threading
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import tensor_fusion_amd.constants as C  # noqa: E402
from tensor_fusion_amd.hypervisor import shm as S  # noqa: E402
from tensor_fusion_amd.hypervisor.pressure import PressureController  # noqa: E402

GB = 1 << 30

TENANT = r"""
import ctypes, json, os, sys, time
import torch

alloc_gb = float(os.environ["DEMO_ALLOC_GB"])
hot_gb = float(os.environ["DEMO_HOT_GB"])
duration = float(os.environ["DEMO_DURATION"])

torch.cuda.init()
lim = ctypes.CDLL(None)
lim.tf_limiter_tier_stats2.argtypes = [ctypes.POINTER(ctypes.c_ulonglong)]

def tier_stats():
    out = (ctypes.c_ulonglong * 8)()
    lim.tf_limiter_tier_stats2(out)
    return list(out)

# "weights": the hot working set the decode loop reads every step
n_hot = int(hot_gb * (1 << 30)) // 2  # bf16
W = torch.randn(n_hot // 4096, 4096, dtype=torch.bfloat16, device="cuda")
x = torch.randn(64, 4096, dtype=torch.bfloat16, device="cuda")

# cold filler in 1 GiB slabs up to alloc_gb (over-budget slabs -> host tier)
filler = []
slab = (1 << 30) // 2
target = int((alloc_gb - hot_gb) * (1 << 30)) // 2
got = 0
while got < target:
    try:
        t = torch.zeros(min(slab, target - got), dtype=torch.bfloat16,
                        device="cuda")
    except torch.cuda.OutOfMemoryError:
        break
    filler.append(t)
    got += t.numel()
torch.cuda.synchronize()
print(json.dumps({"phase": "ready", "alloc_gb": round(
    (got + n_hot) * 2 / (1 << 30), 1)}), flush=True)

# decode-ish loop: read all of W each step (matmul), occasionally touch
# one cold slab (realistic working-set churn)
it = 0
t0 = time.perf_counter()
deadline = t0 + duration
while time.perf_counter() < deadline:
    for chunk in range(0, W.shape[0], 65536):
        y = x @ W[chunk:chunk + 65536].T
    if it % 32 == 0 and filler:
        filler[it // 32 % len(filler)][:1024].add_(1.0)
    it += 1
torch.cuda.synchronize()
dt = time.perf_counter() - t0
s = tier_stats()
print(json.dumps({
    "phase": "done", "iters": it, "it_s": round(it / dt, 2),
    "tier": {"device_res": s[0], "host_res": s[1], "demoted": s[2],
             "promoted": s[3],
             "demote_gbps": round(s[2] / max(s[4], 1), 2),
             "promote_gbps": round(s[3] / max(s[5], 1), 2),
             "n_ranges": s[7]},
}), flush=True)
"""


def read_free():
    """Device free/total via the accelerator lib (amdsmi), no HIP ctx."""

    from tensor_fusion_amd.hypervisor.device import Accelerator
    acc = Accelerator()
    def fn():
        m = acc.metrics(0)
        total = m.vram_total or C.MI355X_VRAM_BYTES
        return max(total - m.vram_used, 0), total
    return fn


def run_tenant(idx: int, shm_path: str, cap: int, args, results, solo=False):
    env = dict(os.environ)
    env.update({
        "LD_PRELOAD": os.path.join(REPO, "tensor_fusion_amd", "_native",
                                   "libtfhip_limiter.so"),
        "TF_SHM_PATH": shm_path,
        "TF_VRAM_EXPAND": "1",
        "DEMO_ALLOC_GB": str(args.alloc_gb),
        "DEMO_HOT_GB": str(args.hot_gb),
        "DEMO_DURATION": str(args.duration),
    })
    out = subprocess.run([sys.executable, "-c", TENANT], env=env,
                         capture_output=True, text=True,
                         timeout=args.duration + 600)
    recs = [json.loads(l) for l in out.stdout.splitlines()
            if l.startswith("{")]
    done = next((r for r in recs if r.get("phase") == "done"), None)
    results[idx] = {"rc": out.returncode, "result": done,
                    "stderr": out.stderr[-1500:] if out.returncode else ""}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--tenants", type=int, default=4)
    ap.add_argument("--cap-gb", type=float, default=96.0)
    ap.add_argument("--alloc-gb", type=float, default=85.0)
    ap.add_argument("--hot-gb", type=float, default=8.0)
    ap.add_argument("--duration", type=float, default=40.0)
    ap.add_argument("--reserve-gb", type=float, default=8.0)
    ap.add_argument("--shm-root", default="/tmp/tf-demo-shm")
    ap.add_argument("--qos-mix", default="",
                    help="comma list per tenant, e.g. critical,high,"
                         "medium,low (default: all medium)")
    ap.add_argument("--scaled", action="store_true",
                    help="CI shape: 3 tenants x 3 GB cap, 4 GB alloc")
    args = ap.parse_args()
    if args.scaled:
        args.tenants, args.cap_gb = 3, 3.0
        args.alloc_gb, args.hot_gb = 4.0, 1.0
        args.duration, args.reserve_gb = 8.0, 2.0

    cap = int(args.cap_gb * GB)
    free_fn = read_free()
    pc = PressureController(free_fn,
                            reserve_bytes=int(args.reserve_gb * GB),
                            interval_s=0.5)

    # ---- solo baseline (one tenant, full cap, no contention)
    page0 = S.WorkerShm.create(os.path.join(args.shm_root, "solo", "shm"))
    page0.set_device(0, "solo", up_limit_percent=100, mem_limit_bytes=cap,
                     total_cus=256, refill_rate=0.0, capacity=0.0)
    solo_res: dict = {}
    run_tenant(0, page0.path, cap, args, solo_res, solo=True)
    solo = solo_res[0]
    solo_its = solo["result"]["it_s"] if solo.get("result") else 0.0
    print(json.dumps({"solo_baseline_it_s": solo_its,
                      "solo_rc": solo["rc"]}), flush=True)

    # ---- N concurrent tenants under the pressure controller
    qos_mix = ([q.strip() for q in args.qos_mix.split(",")]
               if args.qos_mix else [C.QosMedium] * args.tenants)
    pages = []
    for i in range(args.tenants):
        p = S.WorkerShm.create(os.path.join(args.shm_root, f"t{i}", "shm"))
        p.set_device(0, f"tenant-{i}", up_limit_percent=100,
                     mem_limit_bytes=cap, total_cus=256,
                     refill_rate=0.0, capacity=0.0)
        pc.attach(p, qos=qos_mix[i % len(qos_mix)],
                  provisioned_bytes=cap)
        pages.append(p)
    pc.start()

    results: dict = {}
    threads = [threading.Thread(target=run_tenant,
                                args=(i, pages[i].path, cap, args, results))
               for i in range(args.tenants)]
    t0 = time.time()
    for t in threads:
        t.start()
    while any(t.is_alive() for t in threads):
        time.sleep(2.0)
        free, total = free_fn()
        print(json.dumps({
            "t": round(time.time() - t0, 1),
            "free_gb": round(free / GB, 1),
            "budgets_gb": [round(b / GB, 1)
                           for b in pc.last.budgets.values()],
            "pressured": len(pc.last.pressured),
        }), flush=True)
    for t in threads:
        t.join()
    pc.stop()

    tenant_its = [results[i]["result"]["it_s"]
                  for i in range(args.tenants)
                  if results[i].get("result")]
    summary = {
        "config": {"tenants": args.tenants, "cap_gb": args.cap_gb,
                   "alloc_gb": args.alloc_gb, "hot_gb": args.hot_gb,
                   "provisioned_total_gb": args.tenants * args.cap_gb},
        "solo_it_s": solo_its,
        "qos_mix": qos_mix,
        "tenant_it_s": tenant_its,
        "mean_slowdown_x": round(
            solo_its / (sum(tenant_its) / len(tenant_its)), 2)
        if tenant_its and solo_its else None,
        "spread_pct": round(100 * (max(tenant_its) - min(tenant_its))
                            / max(tenant_its), 1) if tenant_its else None,
        "tiers": [results[i]["result"]["tier"]
                  for i in range(args.tenants) if results[i].get("result")],
        "failures": {i: results[i] for i in range(args.tenants)
                     if results[i]["rc"] != 0},
    }
    print("SUMMARY " + json.dumps(summary), flush=True)


if __name__ == "__main__":
    main()
