"""Tensor-parallel Llama: numerics vs the single-process reference over a
2-rank gloo group on CPU (the multi-process distributed-path test the
driver can run GPU-less; on MI355X the same code rides RCCL over xGMI)."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port() -> str:
    import socket
    with socket.socket() as sk:
        sk.bind(("127.0.0.1", 0))
        return str(sk.getsockname()[1])

WORKER = r"""
import os
import torch
import torch.distributed as dist
dist.init_process_group(backend="gloo")
rank = dist.get_rank()
torch.manual_seed(0)

import sys
sys.path.insert(0, os.environ["TF_REPO"])
from tensor_fusion_amd.models.llama import CONFIGS, Llama, decode_bench
from tensor_fusion_amd.parallel.tp import TPLlama

cfg = CONFIGS["tiny"]
# reference model (replicated full weights, deterministic by seed)
torch.manual_seed(1)
ref = Llama(cfg).float()
tokens = torch.randint(0, cfg.vocab, (2, 8), generator=torch.Generator().manual_seed(2))
with torch.no_grad():
    want = ref(tokens)

# TP model: shard the REFERENCE weights so outputs must match exactly
tp = dist.get_world_size()
torch.manual_seed(1)
m = TPLlama(cfg, tp).float()
with torch.no_grad():
    m.embed.weight.copy_(ref.embed.weight)
    m.norm.weight.copy_(ref.norm.weight)
    m.lm_head.weight.copy_(ref.lm_head.weight)
    for b, rb in zip(m.blocks, ref.blocks):
        b.ln1.weight.copy_(rb.ln1.weight)
        b.ln2.weight.copy_(rb.ln2.weight)
        hd = cfg.dim // cfg.heads
        qs = cfg.heads // tp * hd
        ks = cfg.kv_heads // tp * hd
        b.attn.wq.linear.weight.copy_(rb.attn.wq.weight[rank*qs:(rank+1)*qs])
        b.attn.wk.linear.weight.copy_(rb.attn.wk.weight[rank*ks:(rank+1)*ks])
        b.attn.wv.linear.weight.copy_(rb.attn.wv.weight[rank*ks:(rank+1)*ks])
        b.attn.wo.linear.weight.copy_(
            rb.attn.wo.weight[:, rank*qs:(rank+1)*qs])
        isz = cfg.intermediate // tp
        b.mlp.gate.linear.weight.copy_(rb.mlp.gate.weight[rank*isz:(rank+1)*isz])
        b.mlp.up.linear.weight.copy_(rb.mlp.up.weight[rank*isz:(rank+1)*isz])
        b.mlp.down.linear.weight.copy_(rb.mlp.down.weight[:, rank*isz:(rank+1)*isz])
with torch.no_grad():
    got = m(tokens)
err = (got - want).abs().max().item()
scale = want.abs().max().item()
assert err / max(scale, 1e-6) < 1e-4, f"rank {rank}: rel err {err/scale}"

# decode path with sharded kv cache runs end-to-end
tok_s, ms = decode_bench(m, batch=2, ctx=8, steps=3, warmup=1,
                         device="cpu", dtype=torch.float32, sync=False)
assert tok_s > 0
if rank == 0:
    print("TP_OK", err / scale)
dist.destroy_process_group()
"""


def test_tp_matches_reference_2rank():
    env = dict(os.environ)
    env["TF_REPO"] = REPO
    env["MASTER_ADDR"] = "127.0.0.1"
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", _free_port(), "-m", "tensor_fusion_amd.parallel._tp_test_worker"],
        env=env, capture_output=True, text=True, timeout=600, cwd=REPO)
    assert out.returncode == 0, out.stdout[-3000:] + out.stderr[-3000:]
    assert "TP_OK" in out.stdout, out.stdout


def test_bench_tp_tool_2rank_cpu():
    """tools/bench_tp.py runs end-to-end over gloo on CPU (tiny model)."""

    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", _free_port(), "tools/bench_tp.py", "--model", "tiny",
         "--batch", "2", "--ctx", "8", "--steps", "2", "--warmup", "1",
         "--device", "cpu", "--backend", "gloo"],
        env=env, capture_output=True, text=True, timeout=600, cwd=REPO)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    import json
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["value"] > 0 and r["parallelism"] == "tp2"
