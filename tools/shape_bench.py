import sys, time
sys.path.insert(0, ".")
import torch
from tensor_fusion_amd.ops import fused
torch.manual_seed(0)
for (N, K) in [(6144, 4096), (28672, 4096), (14336, 4096), (128256, 4096)]:
    x = torch.randn(8, K, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    wp = fused.pack_skinny_weight(w)
    def bw(fn, reps=200):
        for _ in range(20): fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps): fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / reps * 1e6
    us_p = bw(lambda: fused.skinny_gemm_packed(x, wp, N))
    us_b = bw(lambda: x @ w.T)
    gb = 2*N*K/1e9
    print(f"N={N:6d} K={K:5d}: packed {us_p:7.1f}us ({gb/us_p*1e3:5.2f} TB/s)  blas {us_b:7.1f}us ({gb/us_b*1e3:5.2f} TB/s)")
