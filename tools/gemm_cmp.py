import sys, time, json
import torch
torch.cuda.init()
n = 8192
a = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
b = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
for _ in range(10): a @ b
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(50): a @ b
torch.cuda.synchronize()
dt = time.perf_counter() - t0
tflops = 2 * n**3 * 50 / dt / 1e12
print(json.dumps({"gemm_tflops": round(tflops, 1)}))
