"""PMC profiling target: runs each hot custom kernel a fixed number of
times so rocprofv3 --pmc can attribute counters per kernel.

Usage (on a GPU box):
  cd /tmp && export TMPDIR=/tmp
  rocprofv3 --pmc MfmaUtil VALUBusy --stats -d $GRAFT_REPO_ROOT/gpurun_out/pmc \
      -- python $GRAFT_REPO_ROOT/tools/pmc_target.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from tensor_fusion_amd.ops import fused  # noqa: E402

REPS = 50


def main():
    torch.manual_seed(0)
    dev = "cuda"
    # skinny GEMM at the Llama-3-8B lm_head decode shape
    x = (torch.randn(8, 4096, device=dev, dtype=torch.bfloat16) * 0.05)
    w = (torch.randn(128256, 4096, device=dev, dtype=torch.bfloat16) * 0.05)
    for _ in range(REPS):
        fused.skinny_gemm(x, w)
    torch.cuda.synchronize()

    # packed-layout GEMV (round 2: the hipBLASLt-beating variant)
    wp = fused.pack_skinny_weight(w)
    for _ in range(REPS):
        fused.skinny_gemm_packed(x, wp, w.shape[0])
    torch.cuda.synchronize()

    # fused rmsnorm / add+rmsnorm at the decode row shape
    h = torch.randn(8, 4096, device=dev, dtype=torch.bfloat16)
    r = torch.randn(8, 4096, device=dev, dtype=torch.bfloat16)
    g = torch.randn(4096, device=dev, dtype=torch.bfloat16)
    for _ in range(REPS):
        fused.rmsnorm(h, g, 1e-5)
        fused.add_rmsnorm(h, r, g, 1e-5)
    torch.cuda.synchronize()

    # tiering copy at 1 GiB (the VRAM-expansion bulk path)
    from tensor_fusion_amd.ops import tiering
    src = torch.empty(1 << 30, device=dev, dtype=torch.uint8)
    dst = torch.empty(1 << 30, device=dev, dtype=torch.uint8)
    for _ in range(10):
        tiering.copy_tensor(src, dst)
    tiering.synchronize()
    print("pmc target done")


if __name__ == "__main__":
    main()
