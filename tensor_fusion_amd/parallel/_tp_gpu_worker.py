"""Multi-rank RCCL worker for the GPU de-risk tests (torchrun-style env).

Modes (TF_TP_MODE):
  allreduce  - 2-rank RCCL all-reduce correctness on whatever device each
               rank sees (both ranks may share ONE GPU: RCCL permits
               multiple ranks per device, which is how the distributed
               path is validated before the driver's 8-GPU run)
  decode     - TP llama decode over RCCL: shard reference weights, check
               the TP logits against the single-rank reference, then run
               timed decode steps
Each rank prints one JSON line on rank 0.
"""
import json
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.environ.get("TF_REPO", "."))

mode = os.environ.get("TF_TP_MODE", "allreduce")
backend = os.environ.get("TF_TP_BACKEND", "nccl")
dist.init_process_group(backend=backend)
rank = dist.get_rank()
world = dist.get_world_size()
torch.cuda.set_device(0)  # HIP_VISIBLE_DEVICES picks the physical GPU

if mode == "allreduce":
    x = torch.full((1 << 20,), float(rank + 1), device="cuda")
    dist.all_reduce(x)
    torch.cuda.synchronize()
    want = world * (world + 1) / 2
    ok = bool((x == want).all().item())
    # a second, bucketed round (DDP-style many small tensors)
    parts = [torch.full((4096,), float(rank), device="cuda")
             for _ in range(32)]
    for p in parts:
        dist.all_reduce(p)
    torch.cuda.synchronize()
    ok2 = all(bool((p == sum(range(world))).all().item()) for p in parts)
    if rank == 0:
        print(json.dumps({"ok": ok and ok2, "world": world,
                          "mode": "allreduce"}), flush=True)
elif mode == "ep":
    from tensor_fusion_amd.parallel.ep import ExpertParallelMLP
    DIM, INTER, NEXP, N = 64, 128, 4, 53
    torch.manual_seed(1)
    ref = ExpertParallelMLP(DIM, INTER, NEXP, ep_size=1).float()
    torch.manual_seed(2)
    x = torch.randn(N, DIM)
    with torch.no_grad():
        want = ref(x)
    ep = ExpertParallelMLP(DIM, INTER, NEXP, ep_size=world).float()
    per = NEXP // world
    with torch.no_grad():
        ep.router.weight.copy_(ref.router.weight)
        for i in range(per):
            src = ref.experts[rank * per + i]
            ep.experts[i].gate.weight.copy_(src.gate.weight)
            ep.experts[i].up.weight.copy_(src.up.weight)
            ep.experts[i].down.weight.copy_(src.down.weight)
    ep = ep.to("cuda")
    xg = x.to("cuda")
    with torch.no_grad():
        got = ep(xg).cpu()
    err = (got - want).abs().max().item()
    if rank == 0:
        print(json.dumps({"ok": err < 1e-4, "rel_err": err,
                          "world": world, "mode": "ep"}), flush=True)
elif mode == "decode":
    from tensor_fusion_amd.models.llama import CONFIGS, Llama, decode_bench
    from tensor_fusion_amd.parallel.tp import TPLlama
    cfg = CONFIGS[os.environ.get("TF_TP_MODEL", "tiny")]
    torch.manual_seed(1)
    ref = Llama(cfg).float()
    tokens = torch.randint(0, cfg.vocab, (2, 8),
                           generator=torch.Generator().manual_seed(2))
    with torch.no_grad():
        want = ref(tokens)
    torch.manual_seed(1)
    m = TPLlama(cfg, world).float()
    with torch.no_grad():
        m.embed.weight.copy_(ref.embed.weight)
        m.norm.weight.copy_(ref.norm.weight)
        m.lm_head.weight.copy_(ref.lm_head.weight)
        for b, rb in zip(m.blocks, ref.blocks):
            b.ln1.weight.copy_(rb.ln1.weight)
            b.ln2.weight.copy_(rb.ln2.weight)
            hd = cfg.dim // cfg.heads
            qs = cfg.heads // world * hd
            ks = cfg.kv_heads // world * hd
            b.attn.wq.linear.weight.copy_(
                rb.attn.wq.weight[rank * qs:(rank + 1) * qs])
            b.attn.wk.linear.weight.copy_(
                rb.attn.wk.weight[rank * ks:(rank + 1) * ks])
            b.attn.wv.linear.weight.copy_(
                rb.attn.wv.weight[rank * ks:(rank + 1) * ks])
            b.attn.wo.linear.weight.copy_(
                rb.attn.wo.weight[:, rank * qs:(rank + 1) * qs])
            isz = cfg.intermediate // world
            b.mlp.gate.linear.weight.copy_(
                rb.mlp.gate.weight[rank * isz:(rank + 1) * isz])
            b.mlp.up.linear.weight.copy_(
                rb.mlp.up.weight[rank * isz:(rank + 1) * isz])
            b.mlp.down.linear.weight.copy_(
                rb.mlp.down.weight[:, rank * isz:(rank + 1) * isz])
    m = m.to("cuda")
    tokens = tokens.to("cuda")
    with torch.no_grad():
        got = m(tokens).cpu()
    err = (got - want).abs().max().item()
    scale = want.abs().max().item()
    rel = err / max(scale, 1e-6)
    # timed TP decode (bf16, kv-cache path with RCCL collectives per block)
    m = m.to(torch.bfloat16)
    tok_s, ms = decode_bench(m, batch=2, ctx=16, steps=16, warmup=4,
                             device="cuda", dtype=torch.bfloat16)
    if rank == 0:
        print(json.dumps({"ok": rel < 5e-3, "rel_err": rel,
                          "tok_s": round(tok_s, 1), "world": world,
                          "mode": "decode"}), flush=True)
dist.destroy_process_group()
