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
