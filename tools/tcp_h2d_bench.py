import json, os, sys, time
import torch
torch.cuda.init()
N = 256 << 20  # 256 MiB per transfer
src = torch.randn(N // 4, pin_memory=False)
dst = torch.empty(N // 4, device="cuda")
# warm
dst.copy_(src); torch.cuda.synchronize()
t0 = time.perf_counter()
reps = 12
for _ in range(reps):
    dst.copy_(src)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(json.dumps({"gbps": N * reps / dt / 1e9,
                  "zerocopy": os.environ.get("TF_TCP_ZEROCOPY", "1")}))
