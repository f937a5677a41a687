#!/usr/bin/env python3
"""TP bench — gang-scheduled tensor-parallel Llama over RCCL/xGMI
(BASELINE config 5: TP=4 Llama-3-70B on one node's MI355X GPUs).

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
        --master-addr 127.0.0.1 tools/bench_tp.py \
        --model llama3-70b --batch 4 --ctx 512 --steps 32 --warmup 8

Each rank owns one GPU; the block collectives (2 all-reduces per layer)
ride RCCL over the node's xGMI mesh ("nccl" backend IS RCCL on ROCm).
Rank 0 prints one JSON line with whole-job tok/s. Under the vGPU stack
the ranks run inside gang-scheduled workers; the limiter charges tokens
at launch granularity only, so collectives are never split.
"""
from __future__ import annotations

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-70b")
    ap.add_argument("--batch", type=int, default=4)
    ap.add_argument("--ctx", type=int, default=512)
    ap.add_argument("--steps", type=int, default=32)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--device", default="cuda")
    ap.add_argument("--backend", default=None,
                    help="override (nccl on GPU, gloo for CPU CI)")
    args = ap.parse_args()

    import torch
    import torch.distributed as dist

    backend = args.backend or ("nccl" if args.device == "cuda" else "gloo")
    dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    world = dist.get_world_size()
    if args.device == "cuda":
        dev = os.environ.get("TF_BENCH_DEVICE")  # test override: shared GPU
        torch.cuda.set_device(int(dev) if dev is not None
                              else int(os.environ.get("LOCAL_RANK", rank)))

    from tensor_fusion_amd.models.llama import decode_bench
    from tensor_fusion_amd.parallel.tp import build_tp_model

    dtype = torch.bfloat16 if args.device == "cuda" else torch.float32
    model = build_tp_model(args.model, device=args.device, dtype=dtype)
    dist.barrier()
    if args.device == "cuda":
        torch.cuda.synchronize()
    tok_s, ms = decode_bench(model, args.batch, args.ctx, args.steps,
                             args.warmup, device=args.device, dtype=dtype,
                             sync=args.device == "cuda")
    # whole-job tok/s = batch is replicated across TP ranks (same tokens),
    # so job throughput equals one rank's tok/s; report MAX ms over ranks
    t = torch.tensor([tok_s, ms], dtype=torch.float64)
    gathered = [torch.zeros_like(t) for _ in range(world)]
    dist.all_gather(gathered, t)
    if rank == 0:
        print(json.dumps({
            "metric": f"TP{world} {args.model} decode tok/s",
            "value": round(min(float(g[0]) for g in gathered), 1),
            "ms_per_step": round(max(float(g[1]) for g in gathered), 3),
            "n_gpus": world,
            "parallelism": f"tp{world}",
            "dtype": "bf16" if dtype == torch.bfloat16 else "fp32",
            "data": "synthetic",
        }), flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
