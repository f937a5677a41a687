"""ctypes wrapper over libtfops.so (native/ops/fused_ops.hip).

Fused normalization kernels for the decode hot path. Opt-in for the
model via TF_FUSED_OPS=1 (numerics validated against the fp32 torch
reference in tests/test_gpu_fused.py); fails loudly on a GPU box if the
extension is missing.
"""
from __future__ import annotations

import ctypes
import os
from typing import Optional

_LIB: Optional[ctypes.CDLL] = None


class FusedOpsMissing(RuntimeError):
    pass


def _native_dir() -> str:
    return os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "_native")


def lib() -> ctypes.CDLL:
    global _LIB
    if _LIB is None:
        path = os.path.join(_native_dir(), "libtfops.so")
        if not os.path.exists(path):
            raise FusedOpsMissing(
                f"{path} not built — run python build_native.py")
        _LIB = ctypes.CDLL(path)
        for fn in ("tf_rmsnorm", "tf_add_rmsnorm"):
            getattr(_LIB, fn).restype = ctypes.c_int
        _LIB.tf_rmsnorm.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int,
            ctypes.c_int, ctypes.c_float, ctypes.c_void_p]
        _LIB.tf_add_rmsnorm.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
            ctypes.c_void_p, ctypes.c_int, ctypes.c_int, ctypes.c_float,
            ctypes.c_void_p]
    return _LIB


def rope_qkv_cache(qin, kin, vin, cos_t, sin_t, pos, k_cache, v_cache,
                   heads: int, kv_heads: int, head_dim: int):
    """One launch per attention layer: q-rope to a fresh [B,Hq,T,D]
    tensor; k roped and v copied straight into the caches at `pos`
    (replaces ~10 eager kernels/layer — see the decode census in
    profiles/pmc_ktrace_r02.md)."""

    import torch
    B, T, _ = qin.shape
    S = k_cache.shape[2]
    lb = lib()
    if not hasattr(lb, "_rope_ready"):
        lb.tf_rope_qkv_cache.restype = ctypes.c_int
        lb.tf_rope_qkv_cache.argtypes = ([ctypes.c_void_p] * 9 +
                                         [ctypes.c_int] * 9 +
                                         [ctypes.c_void_p])
        lb._rope_ready = True
    qout = torch.empty(B, heads, T, head_dim, device=qin.device,
                       dtype=qin.dtype)
    # q/k/v may be column slices of ONE merged qkv GEMM output: pass
    # each slice's base pointer + row stride (in elements)
    rc = lb.tf_rope_qkv_cache(
        qin.data_ptr(), kin.data_ptr(), vin.data_ptr(),
        cos_t.data_ptr(), sin_t.data_ptr(), pos.data_ptr(),
        qout.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
        B, T, heads, kv_heads, head_dim, S,
        qin.stride(-2), kin.stride(-2), vin.stride(-2), _stream())
    if rc != 0:
        raise RuntimeError(f"tf_rope_qkv_cache failed: {rc}")
    return qout


def silu_mul_gu(gu, inter: int):
    """out[r,:I] = silu(gu[r,:I]) * gu[r,I:2I] — the two halves of one
    merged gate+up GEMM output, fused in one kernel."""

    import torch
    lb = lib()
    if not hasattr(lb, "_silugu_ready"):
        lb.tf_silu_mul_gu.restype = ctypes.c_int
        lb.tf_silu_mul_gu.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                      ctypes.c_int, ctypes.c_int,
                                      ctypes.c_void_p]
        lb._silugu_ready = True
    rows = gu.numel() // (2 * inter)
    out = torch.empty(*gu.shape[:-1], inter, device=gu.device,
                      dtype=gu.dtype)
    rc = lb.tf_silu_mul_gu(gu.data_ptr(), out.data_ptr(), rows, inter,
                           _stream())
    if rc != 0:
        raise RuntimeError(f"tf_silu_mul_gu failed: {rc}")
    return out


def silu_mul(g, u):
    """out = silu(g) * u in one gfx950 kernel (bf16)."""

    import torch
    lb = lib()
    if not hasattr(lb, "_silu_ready"):
        lb.tf_silu_mul.restype = ctypes.c_int
        lb.tf_silu_mul.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                   ctypes.c_void_p, ctypes.c_longlong,
                                   ctypes.c_void_p]
        lb._silu_ready = True
    out = torch.empty_like(g)
    rc = lb.tf_silu_mul(g.data_ptr(), u.data_ptr(), out.data_ptr(),
                        g.numel(), _stream())
    if rc != 0:
        raise RuntimeError(f"tf_silu_mul failed: {rc}")
    return out


def available() -> bool:
    return os.path.exists(os.path.join(_native_dir(), "libtfops.so"))


def _stream() -> int:
    import torch
    return torch.cuda.current_stream().cuda_stream


def rmsnorm(x, weight, eps: float):
    """out = x * rsqrt(mean(x², -1) + eps) * weight — bf16, any leading
    shape, last dim % 8 == 0."""

    import torch
    assert x.dtype == torch.bfloat16 and x.is_contiguous()
    dim = x.shape[-1]
    rows = x.numel() // dim
    out = torch.empty_like(x)
    rc = lib().tf_rmsnorm(x.data_ptr(), weight.data_ptr(), out.data_ptr(),
                          rows, dim, eps, _stream())
    if rc != 0:
        raise RuntimeError(f"tf_rmsnorm failed: {rc}")
    return out


def add_rmsnorm(x, residual, weight, eps: float):
    """residual += x (in place); returns rmsnorm(residual) * weight."""

    import torch
    assert x.dtype == torch.bfloat16 and x.is_contiguous()
    assert residual.is_contiguous()
    dim = x.shape[-1]
    rows = x.numel() // dim
    out = torch.empty_like(x)
    rc = lib().tf_add_rmsnorm(x.data_ptr(), residual.data_ptr(),
                              weight.data_ptr(), out.data_ptr(), rows, dim,
                              eps, _stream())
    if rc != 0:
        raise RuntimeError(f"tf_add_rmsnorm failed: {rc}")
    return out


def pack_skinny_weight(weight):
    """Pre-shuffle nn.Linear weight[N,K] into the MFMA per-lane fragment
    order consumed by tf_skinny_gemm_packed: P[nt][kt][lane][j] =
    W[nt*16 + (lane&15)][kt*32 + (lane>>4)*8 + j]. One-time cost at
    model load; each compute wave then streams ONE sequential region
    (the layout change that closes the 4.4 vs 5.7 TB/s gap to hipBLASLt
    — profiles/skinny_gemm_packed_r02.md)."""

    import torch
    N, K = weight.shape
    assert N % 16 == 0 and K % 32 == 0, (N, K)
    # (nt, col, kt, kgrp, j) -> (nt, kt, kgrp, col, j); lane = kgrp*16+col
    return (weight.view(N // 16, 16, K // 32, 4, 8)
            .permute(0, 2, 3, 1, 4).contiguous())


def packed_profitable(N: int, K: int) -> bool:
    """Shape routing for the packed decode GEMV, from the measured
    per-shape table (tools/shape_bench.py on MI355X, batch 8):

        N=  4096 K= 4096: packed 19.1us  hipBLASLt 23.4us   -> packed
        N=  1024 K= 4096: packed 15.5us  hipBLASLt 19.1us   -> packed
        N= 14336 K= 4096: packed 24.8us  hipBLASLt 20.2us   -> blas
        N=  4096 K=14336: packed 74.5us  hipBLASLt 19.6us   -> blas
        N=128256 K= 4096: packed 165us   hipBLASLt 185us    -> packed

    The kernel launches N/64 workgroups of 4 waves. Standalone the small
    shapes initially looked competitive, then measured SLOWER in the
    captured decode graph (the microbench was L2-flattered). The
    wave-split-K variant (skinny_gemm_packed_ws_kernel: 4 K-slice
    wave-groups per workgroup, LDS reduction) fixed the per-CU memory
    parallelism at small N:

        N= 4096 K= 4096: 19.1 -> 15.6us (beats blas 18.5)
        N= 1024 K= 4096: 15.5 -> 14.6us (beats blas 19.0)

    Deep-K narrow shapes (down-proj N=4096 K=14336: 55us vs blas 19.3)
    and mid-N (gate/up N=14336: loses to blas 5.8 TB/s) stay on
    hipBLASLt — a grid-level split-K with fp32 atomics could close
    down-proj but costs extra zero/cast launches per projection."""

    if K > 8192:
        return False  # deep-K underfilled: hipBLASLt wins 3x
    if N >= 32768:
        return True
    if N >= 16384 and os.environ.get("TF_PACKED_MID", "0") == "1":
        return True  # chip-filling-ish (gate+up N=28672): A/B gated
    # small-N: wave-split-K wins the STANDALONE microbench (15.6 vs
    # 18.5us at N=4096) but measured consistently slower IN the captured
    # decode graph (A/B x2: 762.5 vs 776.6 tok/s) — hipBLASLt's decode
    # tiles interact better with the surrounding graph than an isolated
    # rep loop predicts. Default off; TF_PACKED_SMALL=1 for experiments.
    return N <= 8192 and os.environ.get("TF_PACKED_SMALL", "0") == "1"


def skinny_gemm_packed(x, wp, N: int):
    """y = x @ W.T with W pre-shuffled by pack_skinny_weight."""

    import torch
    assert x.dtype == torch.bfloat16 and x.is_contiguous()
    M, K = x.shape[-2], x.shape[-1]
    lb = lib()
    if not hasattr(lb, "_skinnyp_ready"):
        lb.tf_skinny_gemm_packed.restype = ctypes.c_int
        lb.tf_skinny_gemm_packed.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int,
            ctypes.c_int, ctypes.c_int, ctypes.c_void_p]
        lb._skinnyp_ready = True
    y = torch.empty(*x.shape[:-1], N, device=x.device, dtype=torch.bfloat16)
    rc = lb.tf_skinny_gemm_packed(x.data_ptr(), wp.data_ptr(), y.data_ptr(),
                                  M, N, K, _stream())
    if rc != 0:
        raise RuntimeError(f"tf_skinny_gemm_packed failed: {rc}")
    return y


def skinny_gemm(x, weight):
    """y = x @ weight.T via the gfx950 MFMA decode-GEMV kernel
    (native/ops/skinny_gemm.hip). x[M,K] bf16 with M ≤ 16; weight[N,K]
    bf16 row-major (nn.Linear layout)."""

    import torch
    assert x.dtype == torch.bfloat16 and x.is_contiguous()
    assert weight.dtype == torch.bfloat16 and weight.is_contiguous()
    M, K = x.shape[-2], x.shape[-1]
    N = weight.shape[0]
    lb = lib()
    if not hasattr(lb, "_skinny_ready"):
        lb.tf_skinny_gemm.restype = ctypes.c_int
        lb.tf_skinny_gemm.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int,
            ctypes.c_int, ctypes.c_int, ctypes.c_void_p]
        lb._skinny_ready = True
    y = torch.empty(*x.shape[:-1], N, device=x.device, dtype=torch.bfloat16)
    rc = lb.tf_skinny_gemm(x.data_ptr(), weight.data_ptr(), y.data_ptr(),
                           M, N, K, _stream())
    if rc != 0:
        raise RuntimeError(f"tf_skinny_gemm failed: {rc}")
    return y
