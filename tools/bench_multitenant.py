"""Multi-tenant pooling efficiency: N concurrent GPU-over-IP tenants on
ONE MI355X, each a GPU-less client driving its own vGPU worker running
the Llama-3-8B decode workload. Reports per-tenant and aggregate tok/s
vs the single-tenant number — the pooling deployment's headline
question (does virtualized sharing cost aggregate throughput?).

Usage (GPU box): python tools/bench_multitenant.py [--tenants 2]
"""
import argparse
import json
import os
import subprocess
import sys
import threading

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from tensor_fusion_amd.client.runtime import client_env, start_worker  # noqa


def run_tenant(i, args, results):
    sock = f"/tmp/tf-mt-{os.getpid()}-{i}.sock"
    w = start_worker(sock, device_index=0)
    try:
        env = client_env(sock)
        out = subprocess.run(
            [sys.executable, "-m", "tensor_fusion_amd.models.llama",
             "--model", args.model, "--batch", str(args.batch),
             "--ctx", str(args.ctx), "--steps", str(args.steps),
             "--warmup", str(args.warmup)],
            env=env, capture_output=True, text=True, timeout=1800, cwd=REPO)
        if out.returncode != 0:
            results[i] = {"error": out.stderr[-800:]}
        else:
            results[i] = json.loads(out.stdout.strip().splitlines()[-1])
    finally:
        w.stop()


def measure(n, args):
    results = {}
    threads = [threading.Thread(target=run_tenant, args=(i, args, results))
               for i in range(n)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    for i, r in results.items():
        if "error" in r:
            raise RuntimeError(f"tenant {i}: {r['error']}")
    return [results[i]["tok_s"] for i in range(n)]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--tenants", type=int, default=2)
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--ctx", type=int, default=512)
    ap.add_argument("--steps", type=int, default=32)
    ap.add_argument("--warmup", type=int, default=6)
    args = ap.parse_args()

    solo = measure(1, args)[0]
    multi = measure(args.tenants, args)
    print(json.dumps({
        "solo_tok_s": round(solo, 1),
        "tenants": args.tenants,
        "per_tenant_tok_s": [round(v, 1) for v in multi],
        "aggregate_tok_s": round(sum(multi), 1),
        "aggregate_vs_solo": round(sum(multi) / solo, 3),
    }))


if __name__ == "__main__":
    main()
