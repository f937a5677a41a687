"""LD_PRELOAD limiter against the real HIP runtime + PyTorch on MI355X."""
import json
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIMITER = os.path.join(REPO, "tensor_fusion_amd", "_native",
                       "libtfhip_limiter.so")

CHILD = r"""
import json, sys, time
import torch
mode = sys.argv[1]
if mode == "vram":
    torch.cuda.init()
    got_oom = False
    try:
        x = torch.empty(int(2.0 * (1<<30)) // 2, dtype=torch.float16,
                        device="cuda")  # 2 GiB
    except torch.OutOfMemoryError:
        got_oom = True
    # under the cap a small alloc must still work
    y = torch.empty(1024, device="cuda")
    print(json.dumps({"oom": got_oom}))
elif mode == "matmul":
    a = torch.randn(1024, 1024, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(1024, 1024, device="cuda", dtype=torch.bfloat16)
    for _ in range(20):
        a @ b
    torch.cuda.synchronize()
    n = 400
    t0 = time.perf_counter()
    for _ in range(n):
        a @ b
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(json.dumps({"elapsed_s": dt, "launches": n}))
"""


def run_child(mode, env_extra, timeout=240):
    env = dict(os.environ)
    env["LD_PRELOAD"] = LIMITER
    env.pop("TF_SHM_PATH", None)
    env.update(env_extra)
    out = subprocess.run([sys.executable, "-c", CHILD, mode], env=env,
                         capture_output=True, text=True, timeout=timeout)
    assert out.returncode == 0, out.stderr[-2000:]
    return json.loads(out.stdout.strip().splitlines()[-1])


def test_vram_cap_causes_torch_oom():
    r = run_child("vram", {"TF_VRAM_LIMIT_BYTES": str(1 << 30)})
    assert r["oom"] is True


def test_no_cap_no_oom():
    r = run_child("vram", {"TF_VRAM_LIMIT_BYTES": str(64 << 30)})
    assert r["oom"] is False


def test_erl_throttles_real_launches():
    # unthrottled
    fast = run_child("matmul", {"TF_UP_LIMIT_PERCENT": "100"})
    # 400 launches at 200/s ≈ 2s forced pacing
    slow = run_child("matmul", {"TF_UP_LIMIT_PERCENT": "25",
                                "TF_ERL_RATE": "200",
                                "TF_ERL_CAPACITY": "20"})
    assert slow["elapsed_s"] > 4 * fast["elapsed_s"]
    assert slow["elapsed_s"] > 1.5
