"""Tensor parallelism for the Llama workload — RCCL over xGMI.

BASELINE config 5: gang-scheduled TP=4 Llama-3-70B across one node's
MI355X GPUs. The stack *schedules* such jobs (gang + topology plugins);
this module is the workload side: a Megatron-style column/row-parallel
split of the Llama blocks whose collectives ride RCCL over xGMI
(`torch.distributed` backend "nccl" IS RCCL on ROCm; "gloo" for CPU CI).

xGMI-aware choices: every intra-node GPU pair has a
dedicated point-to-point link (7 × ≈153 GB/s), so ring all-reduce is
bound by ONE link. TP keeps exactly two all-reduces per block (attention
out-proj + MLP down-proj, the Megatron minimum), sized dim×batch — small
enough that latency, not bandwidth, dominates at decode; RCCL's
direct-connected algorithms handle them without NVSwitch-style staging.

Only the vGPU *limiter* interacts here: collectives must never be split
by throttling — the limiter charges tokens at kernel-launch granularity
only (hip_limiter.cpp), so an in-flight all-reduce is never paused
mid-algorithm.
"""
from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from ..models.llama import (CONFIGS, Llama, LlamaConfig, RMSNorm,
                            apply_rope)


def _world():
    return dist.get_world_size() if dist.is_initialized() else 1


def _all_reduce(x: torch.Tensor) -> torch.Tensor:
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.all_reduce(x)
    return x


class ColumnParallelLinear(nn.Module):
    """Weight split along OUT features; no communication on forward."""

    def __init__(self, in_f: int, out_f: int, tp: int):
        super().__init__()
        assert out_f % tp == 0, f"{out_f} not divisible by tp={tp}"
        self.linear = nn.Linear(in_f, out_f // tp, bias=False)

    def forward(self, x):
        return self.linear(x)


class RowParallelLinear(nn.Module):
    """Weight split along IN features; forward ends in one all-reduce."""

    def __init__(self, in_f: int, out_f: int, tp: int):
        super().__init__()
        assert in_f % tp == 0
        self.linear = nn.Linear(in_f // tp, out_f, bias=False)

    def forward(self, x):
        return _all_reduce(self.linear(x))


class TPAttention(nn.Module):
    """Heads sharded across ranks; wo is row-parallel (1 all-reduce)."""

    def __init__(self, cfg: LlamaConfig, tp: int):
        super().__init__()
        assert cfg.heads % tp == 0 and cfg.kv_heads % tp == 0, \
            f"heads {cfg.heads}/{cfg.kv_heads} not divisible by tp={tp}"
        self.cfg = cfg
        self.tp = tp
        self.heads = cfg.heads // tp
        self.kv_heads = cfg.kv_heads // tp
        self.head_dim = cfg.dim // cfg.heads
        self.wq = ColumnParallelLinear(cfg.dim, cfg.heads * self.head_dim, tp)
        self.wk = ColumnParallelLinear(cfg.dim, cfg.kv_heads * self.head_dim,
                                       tp)
        self.wv = ColumnParallelLinear(cfg.dim, cfg.kv_heads * self.head_dim,
                                       tp)
        self.wo = RowParallelLinear(cfg.heads * self.head_dim, cfg.dim, tp)

    def forward(self, x, cos, sin, pos, cache=None, pos_end=None,
                mask=None):
        B, T, _ = x.shape
        q = self.wq(x).view(B, T, self.heads, self.head_dim).transpose(1, 2)
        k = self.wk(x).view(B, T, self.kv_heads, self.head_dim).transpose(1, 2)
        v = self.wv(x).view(B, T, self.kv_heads, self.head_dim).transpose(1, 2)
        q = apply_rope(q, cos, sin, pos)
        k = apply_rope(k, cos, sin, pos)
        if cache is not None:
            k_cache, v_cache = cache
            k_cache[:, :, pos] = k
            v_cache[:, :, pos] = v
            if mask is not None:
                k, v = k_cache, v_cache
            else:
                end = pos_end if pos_end is not None \
                    else int(pos[-1].item()) + 1
                k = k_cache[:, :, :end]
                v = v_cache[:, :, :end]
        rep = self.heads // self.kv_heads
        # grouped SDPA on the mask-free path only (the masked+GQA combo
        # picks a slow backend — profiles/graph_vs_eager_r02.md)
        gqa = rep > 1 and mask is None
        if rep > 1 and not gqa:
            k = k.repeat_interleave(rep, dim=1)
            v = v.repeat_interleave(rep, dim=1)
        o = F.scaled_dot_product_attention(q, k, v, attn_mask=mask,
                                           is_causal=T > 1 and mask is None,
                                           enable_gqa=gqa)
        o = o.transpose(1, 2).reshape(B, T, -1)
        return self.wo(o)


class TPMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp: int):
        super().__init__()
        self.gate = ColumnParallelLinear(cfg.dim, cfg.intermediate, tp)
        self.up = ColumnParallelLinear(cfg.dim, cfg.intermediate, tp)
        self.down = RowParallelLinear(cfg.intermediate, cfg.dim, tp)

    def forward(self, x):
        return self.down(F.silu(self.gate(x)) * self.up(x))


class TPBlock(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp: int):
        super().__init__()
        self.attn = TPAttention(cfg, tp)
        self.mlp = TPMLP(cfg, tp)
        self.ln1 = RMSNorm(cfg.dim, cfg.norm_eps)
        self.ln2 = RMSNorm(cfg.dim, cfg.norm_eps)

    def forward(self, x, cos, sin, pos, cache=None, pos_end=None,
                mask=None):
        x = x + self.attn(self.ln1(x), cos, sin, pos, cache, pos_end,
                          mask=mask)
        x = x + self.mlp(self.ln2(x))
        return x


class TPLlama(nn.Module):
    """Llama with every block tensor-parallel over the process group.
    Embedding + lm_head replicated (cheap vs. the 2-collective blocks)."""

    def __init__(self, cfg: LlamaConfig, tp: int):
        super().__init__()
        self.cfg = cfg
        self.tp = tp
        self.embed = nn.Embedding(cfg.vocab, cfg.dim)
        self.blocks = nn.ModuleList(TPBlock(cfg, tp) for _ in range(cfg.layers))
        self.norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.lm_head = nn.Linear(cfg.dim, cfg.vocab, bias=False)

    forward = Llama.forward  # same driver loop (rope cache etc.)

    def make_kv_cache(self, batch: int, max_seq: int, device, dtype):
        cfg = self.cfg
        hd = cfg.dim // cfg.heads
        kv = cfg.kv_heads // self.tp
        return [(torch.zeros(batch, kv, max_seq, hd, device=device,
                             dtype=dtype),
                 torch.zeros(batch, kv, max_seq, hd, device=device,
                             dtype=dtype)) for _ in range(cfg.layers)]


@torch.no_grad()
def build_tp_model(name: str, device="cuda", dtype=torch.bfloat16,
                   seed: int = 0) -> TPLlama:
    cfg = CONFIGS[name]
    tp = _world()
    torch.manual_seed(seed)  # same seed → consistent replicated weights
    with torch.device("meta"):
        m = TPLlama(cfg, tp)
    m = m.to_empty(device=device)
    for p in m.parameters():
        p.data.normal_(0, 0.02)
    return m.to(dtype)
