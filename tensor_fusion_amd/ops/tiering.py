"""ctypes wrapper over libtftier.so (native/tiering/tier_kernels.hip).

Fails loudly on a GPU box if the extension is missing — the HIP path must be
the one that runs (never a silent PyTorch fallback).
"""
from __future__ import annotations

import ctypes
import os
from typing import Optional

_LIB: Optional[ctypes.CDLL] = None


class TieringKernelsMissing(RuntimeError):
    pass


def _native_dir() -> str:
    return os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                        "_native")


def lib() -> ctypes.CDLL:
    global _LIB
    if _LIB is None:
        path = os.path.join(_native_dir(), "libtftier.so")
        if not os.path.exists(path):
            raise TieringKernelsMissing(
                f"{path} not built — run python build_native.py (hipcc "
                "--offload-arch=gfx950); the tiering data path requires the "
                "native kernels, there is no fallback")
        _LIB = ctypes.CDLL(path)
        _LIB.tf_tier_copy.restype = ctypes.c_int
        _LIB.tf_tier_copy.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                      ctypes.c_size_t, ctypes.c_void_p,
                                      ctypes.c_int]
        _LIB.tf_tier_gather.restype = ctypes.c_int
        _LIB.tf_tier_gather.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                        ctypes.c_void_p, ctypes.c_size_t,
                                        ctypes.c_uint32, ctypes.c_void_p,
                                        ctypes.c_int]
        _LIB.tf_tier_scatter.restype = ctypes.c_int
        _LIB.tf_tier_scatter.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                         ctypes.c_void_p, ctypes.c_size_t,
                                         ctypes.c_uint32, ctypes.c_void_p]
        _LIB.tf_tier_synchronize.restype = ctypes.c_int
        _LIB.tf_tier_synchronize.argtypes = [ctypes.c_void_p]
    return _LIB


def _stream_ptr(stream=None) -> int:
    if stream is not None:
        return stream
    import torch
    return torch.cuda.current_stream().cuda_stream


def _check(rc: int, what: str):
    if rc != 0:
        raise RuntimeError(f"{what} failed: hipError {rc}")


def copy(src_ptr: int, dst_ptr: int, bytes_: int, stream=None,
         nontemporal: bool = False):
    """Device-side vectorized copy (compaction / staging moves)."""

    _check(lib().tf_tier_copy(src_ptr, dst_ptr, bytes_, _stream_ptr(stream),
                              1 if nontemporal else 0), "tf_tier_copy")


def copy_tensor(src, dst, nontemporal: bool = False, stream=None):
    assert src.numel() * src.element_size() == dst.numel() * dst.element_size()
    assert src.is_contiguous() and dst.is_contiguous()
    copy(src.data_ptr(), dst.data_ptr(), src.numel() * src.element_size(),
         stream=stream, nontemporal=nontemporal)


def gather_pages(base_ptr: int, staging_ptr: int, idx_dev_ptr: int,
                 page_bytes: int, npages: int, stream=None,
                 nontemporal: bool = True):
    """Gather scattered device pages into a contiguous staging arena
    (eviction pre-pass so the PCIe SDMA transfer is one contiguous copy)."""

    _check(lib().tf_tier_gather(base_ptr, staging_ptr, idx_dev_ptr, page_bytes,
                                npages, _stream_ptr(stream),
                                1 if nontemporal else 0), "tf_tier_gather")


def scatter_pages(staging_ptr: int, base_ptr: int, idx_dev_ptr: int,
                  page_bytes: int, npages: int, stream=None):
    _check(lib().tf_tier_scatter(staging_ptr, base_ptr, idx_dev_ptr,
                                 page_bytes, npages, _stream_ptr(stream)),
           "tf_tier_scatter")


def synchronize(stream=None):
    _check(lib().tf_tier_synchronize(_stream_ptr(stream)), "tf_tier_sync")
