"""Minimal Llama-3 implementation — the serving workload for the vGPU bench.

The virtualization stack (limiter / remoting / tiering) is
workload-agnostic; this model exists so bench.py can measure Llama-3-8B
bf16 decode tok/s natively vs through the vGPU path on MI355X
(BASELINE.json config 3). Random-init weights, synthetic tokens.

Run as a module for one benchmark child process:
    python -m tensor_fusion_amd.models.llama --model llama3-8b \
        --batch 8 --ctx 512 --steps 64 --warmup 8
prints one JSON line {"tok_s": ..., "ms_per_step": ...}.
"""
from __future__ import annotations

import argparse
import json
import os
import time
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F


@dataclass
class LlamaConfig:
    vocab: int = 128256
    dim: int = 4096
    layers: int = 32
    heads: int = 32
    kv_heads: int = 8
    intermediate: int = 14336
    rope_theta: float = 500000.0
    max_seq: int = 8192
    norm_eps: float = 1e-5


CONFIGS = {
    "llama3-8b": LlamaConfig(),
    "llama3-70b": LlamaConfig(dim=8192, layers=80, heads=64, kv_heads=8,
                              intermediate=28672),
    # CI-sized config
    "tiny": LlamaConfig(vocab=256, dim=128, layers=2, heads=4, kv_heads=2,
                        intermediate=256, max_seq=512),
}


_FUSED_OPS = os.environ.get("TF_FUSED_OPS") == "1"
# grouped-query SDPA without materializing repeat_interleave'd caches
# (4x cache traffic in decode); TF_SDPA_GQA=0 restores the repeat path
_SDPA_GQA = os.environ.get("TF_SDPA_GQA", "1") != "0"


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))

    def forward(self, x):
        if (_FUSED_OPS and x.is_cuda and x.dtype == torch.bfloat16
                and x.shape[-1] % 8 == 0 and x.is_contiguous()):
            # one gfx950 kernel instead of ~5 eager kernels (ops/fused.py;
            # numerics vs the fp32 reference: tests/test_gpu_fused.py)
            from ..ops import fused
            return fused.rmsnorm(x, self.weight, self.eps)
        dt = x.dtype
        x = x.float()
        x = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + self.eps)
        return (x * self.weight.float()).to(dt)


def precompute_rope(cfg: LlamaConfig, device, dtype=torch.float32):
    head_dim = cfg.dim // cfg.heads
    inv = 1.0 / (cfg.rope_theta ** (
        torch.arange(0, head_dim, 2, device=device, dtype=torch.float32) / head_dim))
    t = torch.arange(cfg.max_seq, device=device, dtype=torch.float32)
    freqs = torch.outer(t, inv)
    return torch.cos(freqs).to(dtype), torch.sin(freqs).to(dtype)


def apply_rope(x, cos, sin, pos):
    # x: [B, H, T, D]; pos: [T]
    c = cos[pos].unsqueeze(0).unsqueeze(0)  # [1,1,T,D/2]
    s = sin[pos].unsqueeze(0).unsqueeze(0)
    x1, x2 = x[..., ::2], x[..., 1::2]
    o1 = x1 * c - x2 * s
    o2 = x2 * c + x1 * s
    out = torch.empty_like(x)
    out[..., ::2] = o1
    out[..., 1::2] = o2
    return out


class Attention(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.head_dim = cfg.dim // cfg.heads
        self.wq = nn.Linear(cfg.dim, cfg.heads * self.head_dim, bias=False)
        self.wk = nn.Linear(cfg.dim, cfg.kv_heads * self.head_dim, bias=False)
        self.wv = nn.Linear(cfg.dim, cfg.kv_heads * self.head_dim, bias=False)
        self.wo = nn.Linear(cfg.heads * self.head_dim, cfg.dim, bias=False)
        self._packed = None  # decode-GEMV weight shuffles (lazy)
        self._wqkv = None    # merged qkv weight (fused path, lazy)

    def _proj(self, lin: nn.Linear, x, idx: int):
        """Decode projections via the packed MFMA GEMV when the token
        batch fits (6.25 TB/s vs hipBLASLt's decode tiles; weights are
        pre-shuffled once per layer — profiles/skinny_gemm_packed_r02.md)."""

        B, T, D = x.shape
        if not (_FUSED_OPS and x.is_cuda and x.dtype == torch.bfloat16
                and B * T <= 16 and D % 32 == 0
                and lin.out_features % 16 == 0):
            return lin(x)
        from ..ops import fused
        if not fused.packed_profitable(lin.out_features, D):
            return lin(x)
        if self._packed is None:
            self._packed = {}
        wp = self._packed.get(idx)
        if wp is None:
            wp = fused.pack_skinny_weight(lin.weight.detach())
            self._packed[idx] = wp
        flat = x.reshape(B * T, D)
        if not flat.is_contiguous():
            flat = flat.contiguous()
        return fused.skinny_gemm_packed(flat, wp, lin.out_features)             .view(B, T, -1)

    def forward(self, x, cos, sin, pos, cache=None, pos_end=None,
                mask=None):
        B, T, _ = x.shape
        cfg = self.cfg
        if (_FUSED_OPS and x.is_cuda and x.dtype == torch.bfloat16
                and cache is not None and self.head_dim % 8 == 0
                and isinstance(pos, torch.Tensor) and cos.is_contiguous()):
            # one kernel: q-rope + k-rope/v straight into the caches
            # (~10 eager kernels/layer collapse; decode census in
            # profiles/pmc_ktrace_r02.md). qkv = ONE merged GEMM (3
            # launches -> 1; the rope kernel reads the column slices
            # with their row stride)
            from ..ops import fused
            if self._wqkv is None:
                self._wqkv = torch.cat(
                    [self.wq.weight, self.wk.weight, self.wv.weight],
                    dim=0).detach().contiguous()
            qkv = F.linear(x, self._wqkv)
            dq = cfg.heads * self.head_dim
            dk = cfg.kv_heads * self.head_dim
            qin = qkv[..., :dq]
            kin = qkv[..., dq:dq + dk]
            vin = qkv[..., dq + dk:]
            k_cache, v_cache = cache
            q = fused.rope_qkv_cache(qin, kin, vin, cos, sin, pos,
                                     k_cache, v_cache, cfg.heads,
                                     cfg.kv_heads, self.head_dim)
            if mask is not None:
                k = k_cache
                v = v_cache
            else:
                end = pos_end if pos_end is not None                     else int(pos[-1].item()) + 1
                k = k_cache[:, :, :end]
                v = v_cache[:, :, :end]
            rep = cfg.heads // cfg.kv_heads
            # grouped SDPA only without a mask: the masked+GQA combo
            # drops to a slow backend (measured 7.4 -> 10.3 ms/step)
            gqa = rep > 1 and _SDPA_GQA and mask is None
            if rep > 1 and not gqa:
                k = k.repeat_interleave(rep, dim=1)
                v = v.repeat_interleave(rep, dim=1)
            o = F.scaled_dot_product_attention(q, k, v, attn_mask=mask,
                                               is_causal=T > 1 and
                                               mask is None,
                                               enable_gqa=gqa)
            o = o.transpose(1, 2).reshape(B, T, -1)
            return self._proj(self.wo, o, 3)
        q = self._proj(self.wq, x, 0).view(B, T, cfg.heads,
                                           self.head_dim).transpose(1, 2)
        k = self._proj(self.wk, x, 1).view(B, T, cfg.kv_heads,
                                           self.head_dim).transpose(1, 2)
        v = self._proj(self.wv, x, 2).view(B, T, cfg.kv_heads,
                                           self.head_dim).transpose(1, 2)
        q = apply_rope(q, cos, sin, pos)
        k = apply_rope(k, cos, sin, pos)
        if cache is not None:
            k_cache, v_cache = cache
            k_cache[:, :, pos] = k
            v_cache[:, :, pos] = v
            if mask is not None:
                # graph mode: fixed-shape full-cache attention; validity is
                # the additive mask (updated outside the captured region)
                k = k_cache
                v = v_cache
            else:
                # pos_end is a host int: a device read (`pos[-1].item()`)
                # would force a D2H sync per layer per token.
                end = pos_end if pos_end is not None \
                    else int(pos[-1].item()) + 1
                k = k_cache[:, :, :end]
                v = v_cache[:, :, :end]
        rep = cfg.heads // cfg.kv_heads
        gqa = rep > 1 and _SDPA_GQA and mask is None
        if rep > 1 and not gqa:
            k = k.repeat_interleave(rep, dim=1)
            v = v.repeat_interleave(rep, dim=1)
        causal = T > 1 and mask is None
        o = F.scaled_dot_product_attention(q, k, v, attn_mask=mask,
                                           is_causal=causal,
                                           enable_gqa=gqa)
        o = o.transpose(1, 2).reshape(B, T, -1)
        return self._proj(self.wo, o, 3)


class MLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.gate = nn.Linear(cfg.dim, cfg.intermediate, bias=False)
        self.up = nn.Linear(cfg.dim, cfg.intermediate, bias=False)
        self.down = nn.Linear(cfg.intermediate, cfg.dim, bias=False)
        self._packed = None
        self._wgu = None  # merged gate+up weight (fused path, lazy)

    _proj = Attention._proj  # same packed-GEMV decode routing

    def forward(self, x):
        if (_FUSED_OPS and x.is_cuda and x.dtype == torch.bfloat16
                and self.gate.out_features % 8 == 0):
            # gate+up = ONE merged GEMM, then one silu*mul kernel over
            # the two column halves
            from ..ops import fused
            if self._wgu is None:
                self._wgu = torch.cat(
                    [self.gate.weight, self.up.weight],
                    dim=0).detach().contiguous()
            gu = F.linear(x, self._wgu)
            h = fused.silu_mul_gu(gu, self.gate.out_features)
            return self._proj(self.down, h, 2)
        g = self._proj(self.gate, x, 0)
        u = self._proj(self.up, x, 1)
        return self._proj(self.down, F.silu(g) * u, 2)


class Block(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.attn = Attention(cfg)
        self.mlp = MLP(cfg)
        self.ln1 = RMSNorm(cfg.dim, cfg.norm_eps)
        self.ln2 = RMSNorm(cfg.dim, cfg.norm_eps)

    def forward(self, x, cos, sin, pos, cache=None, pos_end=None,
                mask=None):
        x = x + self.attn(self.ln1(x), cos, sin, pos, cache, pos_end,
                          mask=mask)
        x = x + self.mlp(self.ln2(x))
        return x


class Llama(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Embedding(cfg.vocab, cfg.dim)
        self.blocks = nn.ModuleList(Block(cfg) for _ in range(cfg.layers))
        self.norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.lm_head = nn.Linear(cfg.dim, cfg.vocab, bias=False)

    def forward(self, tokens, pos=None, caches=None, pos_end=None,
                mask=None):
        device = tokens.device
        if pos is None:
            pos = torch.arange(tokens.shape[1], device=device)
        if not hasattr(self, "_cos") or self._cos.device != device:
            self._cos, self._sin = precompute_rope(self.cfg, device)
        x = self.embed(tokens)
        if (_FUSED_OPS and x.is_cuda and x.dtype == torch.bfloat16
                and x.shape[-1] % 8 == 0):
            return self._forward_fused(x, pos, caches, pos_end, mask)
        for i, blk in enumerate(self.blocks):
            x = blk(x, self._cos, self._sin, pos,
                    caches[i] if caches is not None else None, pos_end,
                    mask=mask)
        return self.lm_head(self.norm(x))

    def _forward_fused(self, x, pos, caches, pos_end, mask):
        """Residual-carry layout: the residual stream lives in one buffer
        and every add+RMSNorm pair is ONE gfx950 kernel (ops/fused.py) —
        the vLLM-style fusion of the decode hot path."""

        from ..ops import fused
        res = x.contiguous()
        normed = fused.rmsnorm(res, self.blocks[0].ln1.weight,
                               self.blocks[0].ln1.eps)
        n = len(self.blocks)
        for i, blk in enumerate(self.blocks):
            a = blk.attn(normed, self._cos, self._sin, pos,
                         caches[i] if caches is not None else None,
                         pos_end, mask=mask)
            normed = fused.add_rmsnorm(a.contiguous(), res, blk.ln2.weight,
                                       blk.ln2.eps)
            m = blk.mlp(normed)
            nxt_w = (self.blocks[i + 1].ln1.weight if i + 1 < n
                     else self.norm.weight)
            nxt_eps = (self.blocks[i + 1].ln1.eps if i + 1 < n
                       else self.norm.eps)
            normed = fused.add_rmsnorm(m.contiguous(), res, nxt_w, nxt_eps)
        # lm_head via the packed-layout MFMA decode GEMV when the token
        # batch fits (M <= 16): 6.25 TB/s vs hipBLASLt's 5.7 on this
        # shape (profiles/skinny_gemm_packed_r02.md). Weights are
        # pre-shuffled once and cached; fall back to the plain matmul
        # for prefill-sized batches.
        B, T, D = normed.shape
        if (B * T <= 16 and D % 32 == 0
                and self.lm_head.weight.shape[0] % 16 == 0):
            if getattr(self, "_lm_packed", None) is None:
                self._lm_packed = fused.pack_skinny_weight(
                    self.lm_head.weight.detach())
            flat = normed.reshape(B * T, D).contiguous()
            out = fused.skinny_gemm_packed(
                flat, self._lm_packed, self.lm_head.weight.shape[0])
            return out.view(B, T, -1)
        return self.lm_head(normed)

    def make_kv_cache(self, batch: int, max_seq: int, device, dtype):
        cfg = self.cfg
        hd = cfg.dim // cfg.heads
        return [(torch.zeros(batch, cfg.kv_heads, max_seq, hd, device=device,
                             dtype=dtype),
                 torch.zeros(batch, cfg.kv_heads, max_seq, hd, device=device,
                             dtype=dtype)) for _ in range(cfg.layers)]


@torch.no_grad()
def build_model(name: str, device="cuda", dtype=torch.bfloat16,
                seed: int = 0) -> Llama:
    cfg = CONFIGS[name]
    torch.manual_seed(seed)
    with torch.device("meta"):
        m = Llama(cfg)
    m = m.to_empty(device=device)
    # cheap random init directly on device (values only affect numerics,
    # not timing; scale keeps activations finite)
    for p in m.parameters():
        p.data.normal_(0, 0.02)
    return m.to(dtype)


@torch.no_grad()
def decode_bench(model: Llama, batch: int, ctx: int, steps: int, warmup: int,
                 device="cuda", dtype=torch.bfloat16, sync=True):
    """Prefill `ctx` tokens, then time `steps` single-token decode steps.
    Returns (tok_s, ms_per_step)."""

    cfg = model.cfg
    caches = model.make_kv_cache(batch, ctx + steps + warmup + 8, device, dtype)
    toks = torch.randint(0, cfg.vocab, (batch, ctx), device=device)
    model(toks, pos=torch.arange(ctx, device=device), caches=caches,
          pos_end=ctx)

    cur = torch.randint(0, cfg.vocab, (batch, 1), device=device)

    pos_buf = torch.empty(1, dtype=torch.long, device=device)

    def step(i):
        # single small H2D per token (no device reads on the host path)
        pos_buf.copy_(torch.tensor([ctx + i]), non_blocking=True)
        logits = model(cur, pos=pos_buf, caches=caches, pos_end=ctx + i + 1)
        return logits.argmax(-1)

    for i in range(warmup):
        cur = step(i)
    if sync and device != "cpu":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(warmup, warmup + steps):
        cur = step(i)
    if sync and device != "cpu":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return batch * steps / dt, dt / steps * 1000.0


@torch.no_grad()
def decode_bench_graphs(model: Llama, batch: int, ctx: int, steps: int,
                        warmup: int, device="cuda", dtype=torch.bfloat16):
    """hipGraph-captured decode: one graph replay + two 1-element H2Ds per
    token instead of ~1,100 eager launches. Works natively and through the
    GPU-over-IP client (capture ops are forwarded to the worker)."""

    cfg = model.cfg
    total = ctx + steps + warmup + 8
    caches = model.make_kv_cache(batch, total, device, dtype)
    toks = torch.randint(0, cfg.vocab, (batch, ctx), device=device)
    model(toks, pos=torch.arange(ctx, device=device), caches=caches,
          pos_end=ctx)

    cur = torch.randint(0, cfg.vocab, (batch, 1), device=device)
    pos_buf = torch.tensor([ctx], dtype=torch.long, device=device)
    # additive validity mask over the FULL cache length
    mask = torch.full((1, 1, 1, total), float("-inf"), device=device,
                      dtype=dtype)
    mask[..., :ctx] = 0.0
    zero = torch.zeros(1, device=device, dtype=dtype)
    one = torch.ones(1, dtype=torch.long, device=device)

    def step_graphable():
        # fully device-driven: position advance and mask validity update
        # are graph nodes, so a decode step is ONE graph replay with no
        # host work at all (remote: one async ring record per token)
        mask.view(-1).index_copy_(0, pos_buf, zero)
        logits = model(cur, pos=pos_buf, caches=caches, mask=mask)
        cur.copy_(logits.argmax(-1))
        pos_buf.add_(one)

    # warmup on a side stream (allocator steady-state), then capture
    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        for _ in range(3):
            step_graphable()
    torch.cuda.current_stream().wait_stream(side)
    torch.cuda.synchronize()

    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        step_graphable()

    for _ in range(1, warmup):
        g.replay()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        g.replay()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return batch * steps / dt, dt / steps * 1000.0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--ctx", type=int, default=512)
    ap.add_argument("--steps", type=int, default=64)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--device", default="cuda")
    ap.add_argument("--dtype", default="bf16")
    ap.add_argument("--graphs", action="store_true",
                    help="hipGraph-captured decode loop")
    args = ap.parse_args()
    dtype = {"bf16": torch.bfloat16, "fp16": torch.float16,
             "fp32": torch.float32}[args.dtype]
    model = build_model(args.model, device=args.device, dtype=dtype)
    if args.graphs or os.environ.get("TF_BENCH_GRAPHS") == "1":
        tok_s, ms = decode_bench_graphs(model, args.batch, args.ctx,
                                        args.steps, args.warmup,
                                        device=args.device, dtype=dtype)
    else:
        tok_s, ms = decode_bench(model, args.batch, args.ctx, args.steps,
                                 args.warmup, device=args.device, dtype=dtype)
    print(json.dumps({"tok_s": tok_s, "ms_per_step": ms, "model": args.model,
                      "batch": args.batch, "ctx": args.ctx}), flush=True)


if __name__ == "__main__":
    main()
