#!/usr/bin/env python3
"""Why does eager decode now beat hipGraph replay at batch 8?

Times ONE decode step both ways on identical state with GPU events and
prints the split. Run under rocprofv3 --kernel-trace --stats with
TF_DIAG_MODE=eager|graph to get the per-mode kernel census (the mode
loops MANY steps so the trace dominates setup).

    python tools/graph_vs_eager.py            # event timing, both modes
    TF_DIAG_MODE=eager python tools/graph_vs_eager.py --steps 256
"""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# BEFORE the model import: models/llama.py snapshots TF_FUSED_OPS at
# import time (bench children get it from their process env)
os.environ.setdefault("TF_FUSED_OPS", "1")
from tensor_fusion_amd.models.llama import build_model  # noqa: E402


@torch.no_grad()
def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--ctx", type=int, default=512)
    ap.add_argument("--steps", type=int, default=64)
    args = ap.parse_args()
    mode = os.environ.get("TF_DIAG_MODE", "")

    model = build_model(args.model, device="cuda", dtype=torch.bfloat16)
    cfg = model.cfg
    total = args.ctx + args.steps + 8
    caches = model.make_kv_cache(args.batch, total, "cuda", torch.bfloat16)
    toks = torch.randint(0, cfg.vocab, (args.batch, args.ctx), device="cuda")
    model(toks, pos=torch.arange(args.ctx, device="cuda"), caches=caches,
          pos_end=args.ctx)
    cur = torch.randint(0, cfg.vocab, (args.batch, 1), device="cuda")
    pos_buf = torch.tensor([args.ctx], dtype=torch.long, device="cuda")
    mask = torch.full((1, 1, 1, total), float("-inf"), device="cuda",
                      dtype=torch.bfloat16)
    mask[..., :args.ctx] = 0.0
    zero = torch.zeros(1, device="cuda", dtype=torch.bfloat16)
    one = torch.ones(1, dtype=torch.long, device="cuda")

    def step():
        mask.view(-1).index_copy_(0, pos_buf, zero)
        logits = model(cur, pos=pos_buf, caches=caches, mask=mask)
        cur.copy_(logits.argmax(-1))
        pos_buf.add_(one)

    if os.environ.get("TF_DIAG_WARM", ""):
        # ramp SCLK/power state the way bench's sustained child runs do
        a = torch.randn(8192, 8192, device="cuda", dtype=torch.bfloat16)
        t0 = time.perf_counter()
        while time.perf_counter() - t0 < float(
                os.environ.get("TF_DIAG_WARM", "20")):
            a @ a
        torch.cuda.synchronize()

    def time_gpu(fn, n=24):
        for _ in range(6):
            fn()
        torch.cuda.synchronize()
        evs = [(torch.cuda.Event(enable_timing=True),
                torch.cuda.Event(enable_timing=True)) for _ in range(n)]
        t0 = time.perf_counter()
        for e0, e1 in evs:
            e0.record()
            fn()
            e1.record()
        torch.cuda.synchronize()
        wall = (time.perf_counter() - t0) / n
        gpu = sorted(e0.elapsed_time(e1) for e0, e1 in evs)[n // 2]
        return gpu, wall * 1e3

    if mode:  # rocprof census mode: run ONE mode many times, exit
        if mode == "graph":
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):
                    step()
            torch.cuda.current_stream().wait_stream(side)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                step()
            fn = g.replay
        else:
            fn = step
        torch.cuda.synchronize()
        for _ in range(args.steps):
            fn()
        torch.cuda.synchronize()
        print(json.dumps({"mode": mode, "steps": args.steps}))
        return

    # sliced-KV eager step — what bench.py's eager row actually runs
    # (host-known position, no additive mask)
    pos_host = [args.ctx + 1]

    def step_sliced():
        pos_buf.copy_(torch.tensor([pos_host[0]]), non_blocking=True)
        logits = model(cur, pos=pos_buf, caches=caches,
                       pos_end=pos_host[0] + 1)
        cur.copy_(logits.argmax(-1))

    sliced_gpu, sliced_wall = time_gpu(step_sliced)
    eager_gpu, eager_wall = time_gpu(step)

    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        for _ in range(3):
            step()
    torch.cuda.current_stream().wait_stream(side)
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        step()
    graph_gpu, graph_wall = time_gpu(g.replay)

    out = {
        "batch": args.batch, "ctx": args.ctx,
        "sliced_eager_gpu_ms": round(sliced_gpu, 4),
        "sliced_eager_wall_ms": round(sliced_wall, 4),
        "tok_s_sliced_eager": round(args.batch / sliced_wall * 1e3, 1),
        "eager_gpu_ms": round(eager_gpu, 4),
        "eager_wall_ms": round(eager_wall, 4),
        "graph_gpu_ms": round(graph_gpu, 4),
        "graph_wall_ms": round(graph_wall, 4),
        "graph_minus_eager_gpu_us": round(
            (graph_gpu - eager_gpu) * 1e3, 1),
        "tok_s_eager": round(args.batch / eager_wall * 1e3, 1),
        "tok_s_graph": round(args.batch / graph_wall * 1e3, 1),
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
