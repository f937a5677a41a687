#!/usr/bin/env python3
"""Webhook admission throughput over real HTTP (reference
scripts/benchmark.sh benches the mutating webhook with bombardier at
c=10/c=100; Python load generators are slow, so this drives the server
with several forked aiohttp processes and reports the aggregate).

    python tools/bench_webhook.py [--n 4000] [--procs 4] [--conc 10]
"""
from __future__ import annotations

import argparse
import asyncio
import json
import multiprocessing as mp
import os
import socket
import statistics
import sys
import threading
import time

import uvicorn

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import tensor_fusion_amd.constants as C  # noqa: E402
from tensor_fusion_amd.api.store import Store  # noqa: E402
from tensor_fusion_amd.server.webhook_server import create_webhook_app  # noqa: E402
from tensor_fusion_amd.webhook import PodMutator  # noqa: E402


def review(i: int) -> bytes:
    pod = {"metadata": {"name": f"app-{i}", "namespace": "default",
                        "labels": {C.LabelEnabled: "true"},
                        "annotations": {C.AnnoTflopsRequest: "100",
                                        C.AnnoVramRequest: str(8 << 30)}},
           "spec": {"containers": [{"name": "main", "image": "app:1"}]}}
    return json.dumps({
        "apiVersion": "admission.k8s.io/v1", "kind": "AdmissionReview",
        "request": {"uid": f"u-{i}", "namespace": "default",
                    "object": pod}}).encode()


def _load_proc(port: int, n: int, conc: int, seed: int, q):
    import aiohttp

    async def main():
        lat = []
        async with aiohttp.ClientSession() as s:
            sem = asyncio.Semaphore(conc)

            async def one(i):
                body = review(seed * 1000000 + i)
                async with sem:
                    t0 = time.perf_counter()
                    async with s.post(
                            f"http://127.0.0.1:{port}/mutate-v1-pod",
                            data=body,
                            headers={"content-type": "application/json"}
                    ) as r:
                        await r.read()
                        assert r.status == 200
                    lat.append(time.perf_counter() - t0)
            t0 = time.perf_counter()
            await asyncio.gather(*[one(i) for i in range(n)])
            wall = time.perf_counter() - t0
        return wall, lat

    wall, lat = asyncio.run(main())
    q.put((wall, lat))


def run(n_total: int, procs: int, conc: int,
        server_workers: int = 1) -> dict:
    app = create_webhook_app(PodMutator(Store()))
    with socket.socket() as sk:
        sk.bind(("127.0.0.1", 0))
        port = sk.getsockname()[1]
    server = None
    if server_workers > 1:
        from tensor_fusion_amd.server.webhook_server import \
            serve_multiprocess
        lsock, worker_pids = serve_multiprocess(
            app, port, workers=server_workers, host="127.0.0.1")
    else:
        server = uvicorn.Server(uvicorn.Config(
            app, host="127.0.0.1", port=port, log_level="error"))
        threading.Thread(target=server.run, daemon=True).start()
    import requests
    for _ in range(200):
        try:
            if requests.get(f"http://127.0.0.1:{port}/healthz",
                            timeout=1).ok:
                break
        except Exception:
            time.sleep(0.02)

    per = n_total // procs
    ctx = mp.get_context("fork")
    q = ctx.Queue()
    ps = [ctx.Process(target=_load_proc, args=(port, per, conc, k, q))
          for k in range(procs)]
    t0 = time.perf_counter()
    for p in ps:
        p.start()
    results = [q.get(timeout=300) for _ in ps]
    for p in ps:
        p.join()
    wall = time.perf_counter() - t0
    if server is not None:
        server.should_exit = True
    else:
        import signal
        for pid in worker_pids:
            os.kill(pid, signal.SIGTERM)
        lsock.close()
    lat = sorted(x for (_, ls) in results for x in ls)
    return {
        "qps": per * procs / wall,
        "p50_ms": statistics.median(lat) * 1e3,
        "p99_ms": lat[int(len(lat) * 0.99) - 1] * 1e3,
        "n": per * procs, "procs": procs, "conc_per_proc": conc,
        "server_workers": server_workers,
    }


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=4000)
    ap.add_argument("--procs", type=int, default=4)
    ap.add_argument("--conc", type=int, default=10)
    ap.add_argument("--server-workers", type=int, default=1)
    args = ap.parse_args()
    r = run(args.n, args.procs, args.conc, args.server_workers)
    print(json.dumps({k: round(v, 2) if isinstance(v, float) else v
                      for k, v in r.items()}))
