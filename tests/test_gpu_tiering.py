"""Tiering HIP kernels on a real MI355X (gather/scatter/copy + bandwidth)."""
import time

import pytest
import torch

pytestmark = pytest.mark.gpu

from tensor_fusion_amd.ops import tiering  # noqa: E402


@pytest.fixture(scope="module", autouse=True)
def _cuda():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.cuda.init()


def test_copy_roundtrip():
    src = torch.randn(8 << 20, device="cuda")  # 32 MiB
    dst = torch.empty_like(src)
    tiering.copy_tensor(src, dst)
    tiering.synchronize()
    assert torch.equal(src, dst)


def test_copy_nontemporal():
    src = torch.randn(8 << 20, device="cuda")
    dst = torch.empty_like(src)
    tiering.copy_tensor(src, dst, nontemporal=True)
    tiering.synchronize()
    assert torch.equal(src, dst)


def test_gather_scatter_pages():
    page = 1 << 20  # 1 MiB pages
    npages_total, npick = 64, 16
    base = torch.randn(npages_total * page // 4, device="cuda")
    staging = torch.zeros(npick * page // 4, device="cuda")
    # uint32 on the C side; int32 works for values < 2^31
    idx32 = torch.arange(npages_total - 1, -1, -4, dtype=torch.int32,
                         device="cuda")[:npick].contiguous()
    tiering.gather_pages(base.data_ptr(), staging.data_ptr(), idx32.data_ptr(),
                         page, npick)
    tiering.synchronize()
    for i in range(npick):
        j = int(idx32[i])
        assert torch.equal(staging[i * page // 4:(i + 1) * page // 4],
                           base[j * page // 4:(j + 1) * page // 4])

    # scatter back to different slots and verify
    base2 = torch.zeros_like(base)
    tiering.scatter_pages(staging.data_ptr(), base2.data_ptr(),
                          idx32.data_ptr(), page, npick)
    tiering.synchronize()
    for i in range(npick):
        j = int(idx32[i])
        assert torch.equal(base2[j * page // 4:(j + 1) * page // 4],
                           staging[i * page // 4:(i + 1) * page // 4])


def test_copy_bandwidth_tb_s():
    """D2D copy must be HBM-class (guide: ~6.3 TB/s achievable; require >3)."""

    n = 1 << 30  # 4 GiB read + 4 GiB write
    src = torch.empty(n // 4, device="cuda")
    dst = torch.empty_like(src)
    tiering.copy_tensor(src, dst)  # warm
    tiering.synchronize()
    t0 = time.perf_counter()
    reps = 5
    for _ in range(reps):
        tiering.copy_tensor(src, dst)
    tiering.synchronize()
    dt = time.perf_counter() - t0
    tb_s = 2 * n * reps / dt / 1e12
    print(f"tier copy: {tb_s:.2f} TB/s")
    assert tb_s > 3.0, f"copy kernel too slow: {tb_s:.2f} TB/s"


def test_graph_decode_native_matches_eager():
    """Graph-mode decode numerics: greedy tokens from the captured loop
    must match the eager loop (same seed/model/prompt)."""

    import torch

    from tensor_fusion_amd.models.llama import (build_model, CONFIGS)
    cfg = CONFIGS["tiny"]
    torch.manual_seed(7)
    model = build_model("tiny", device="cuda", dtype=torch.float32)
    B, CTX, N = 2, 16, 12

    def run(graphs):
        torch.manual_seed(9)
        toks = torch.randint(0, cfg.vocab, (B, CTX), device="cuda")
        caches = model.make_kv_cache(B, CTX + N + 8, "cuda", torch.float32)
        model(toks, pos=torch.arange(CTX, device="cuda"), caches=caches,
              pos_end=CTX)
        cur = toks[:, -1:].clone()
        outs = []
        if not graphs:
            for i in range(N):
                pos = torch.tensor([CTX + i], device="cuda")
                logits = model(cur, pos=pos, caches=caches, pos_end=CTX+i+1)
                cur = logits.argmax(-1)
                outs.append(cur.clone())
        else:
            total = CTX + N + 8
            mask = torch.full((1, 1, 1, total), float("-inf"),
                              device="cuda", dtype=torch.float32)
            mask[..., :CTX] = 0.0
            pos_buf = torch.empty(1, dtype=torch.long, device="cuda")
            static_cur = cur.clone()
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            # NOTE: warmup would pollute the kv cache; capture directly
            # (allocator already warm from prefill)
            torch.cuda.current_stream().wait_stream(side)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                logits = model(static_cur, pos=pos_buf, caches=caches,
                               mask=mask)
                static_cur.copy_(logits.argmax(-1))
            for i in range(N):
                pos_buf.copy_(torch.tensor([CTX + i]))
                mask[..., CTX + i] = 0.0
                g.replay()
                torch.cuda.synchronize()
                outs.append(static_cur.clone())
        return torch.stack(outs)

    eager = run(False)
    graphed = run(True)
    assert torch.equal(eager, graphed), (eager.flatten(), graphed.flatten())


# ---------------------------------------------------------------------------
# Tier engine: LRU working-set migration + stats on the managed-memory
# host tier (hip_limiter.cpp). A VMM host tier was probed and REJECTED —
# ROCm 7.2 backs "host-located" handles with HBM (vmm_host_probe4.cpp),
# so managed memory + SDMA prefetch is the genuine mechanism.

LRU_CHILD = r"""
import ctypes, json, sys
import torch
torch.cuda.init()

lim = ctypes.CDLL(None)
lim.tf_limiter_tier_stats2.argtypes = [ctypes.POINTER(ctypes.c_ulonglong)]
lim.tf_limiter_touch.argtypes = [ctypes.c_void_p]
lim.tf_limiter_promote.restype = ctypes.c_ulonglong
lim.tf_limiter_promote.argtypes = [ctypes.c_ulonglong]
lim.tf_limiter_demote_all.restype = ctypes.c_ulonglong
lim.tf_limiter_tier_range.restype = ctypes.c_int
lim.tf_limiter_tier_range.argtypes = [
    ctypes.c_void_p, ctypes.POINTER(ctypes.c_ulonglong),
    ctypes.POINTER(ctypes.c_int), ctypes.POINTER(ctypes.c_ulonglong)]

def rng(t):
    b = ctypes.c_ulonglong(); d = ctypes.c_int(); lu = ctypes.c_ulonglong()
    ok = lim.tf_limiter_tier_range(ctypes.c_void_p(t.data_ptr()),
                                   ctypes.byref(b), ctypes.byref(d),
                                   ctypes.byref(lu))
    return ok, d.value, lu.value

def stats():
    out = (ctypes.c_ulonglong * 8)()
    lim.tf_limiter_tier_stats2(out)
    return list(out)

MB = 1 << 20
# fill most of the 1 GiB cap, then three over-cap ranges -> host tier
pad = torch.zeros(int(0.8 * (1 << 30)) // 4, device="cuda")
a = torch.full((300 * MB // 4,), 1.0, device="cuda")
b = torch.full((300 * MB // 4,), 2.0, device="cuda")
c = torch.full((300 * MB // 4,), 3.0, device="cuda")
torch.cuda.synchronize()
for t in (a, b, c):
    ok, dev, _ = rng(t)
    assert ok == 1 and dev == 0, (ok, dev)

# heat: touch a then c (c is hottest; b never touched)
lim.tf_limiter_touch(ctypes.c_void_p(a.data_ptr()))
_ = float(a.sum())  # ticks the epoch via governed ops
lim.tf_limiter_touch(ctypes.c_void_p(c.data_ptr()))

# promote with budget for ONE range -> must pick c (hottest)
lim.tf_limiter_promote(ctypes.c_ulonglong(310 * MB))
ra, rb, rc = rng(a), rng(b), rng(c)
assert rc[1] == 1, (ra, rb, rc)
assert ra[1] == 0 and rb[1] == 0, (ra, rb, rc)

# promote one more -> a (b is coldest, stays)
lim.tf_limiter_promote(ctypes.c_ulonglong(310 * MB))
ra, rb, rc = rng(a), rng(b), rng(c)
assert ra[1] == 1 and rc[1] == 1 and rb[1] == 0, (ra, rb, rc)

# pressure demote of everything; data must be intact afterwards
lim.tf_limiter_demote_all()
torch.cuda.synchronize()
assert float(a.sum()) == a.numel() * 1.0
assert float(b.sum()) == b.numel() * 2.0
assert float(c.sum()) == c.numel() * 3.0
s = stats()
bw_demote = s[2] / max(s[4], 1)   # bytes per ns == GB/s
bw_promote = s[3] / max(s[5], 1)
print(json.dumps({"ok": True, "demoted_total": s[2], "promoted_total": s[3],
                  "demote_gbps": round(bw_demote, 2),
                  "promote_gbps": round(bw_promote, 2),
                  "n_ranges": s[7]}))
"""


def _limiter_env(cap_bytes: int, extra=None):
    import os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.update({
        "LD_PRELOAD": os.path.join(repo, "tensor_fusion_amd", "_native",
                                   "libtfhip_limiter.so"),
        "TF_VRAM_LIMIT_BYTES": str(cap_bytes),
        "TF_VRAM_EXPAND": "1",
        "TF_LIMITER_DEBUG": "1",
    })
    env.pop("TF_SHM_PATH", None)
    env.update(extra or {})
    return env


def test_tier_lru_promotes_hottest_first():
    """LRU working-set migration: with budget for one range, promote must
    pick the most recently touched range; untouched ranges stay cold;
    pressure demote preserves data. Migration bandwidth reported from the
    engine's own byte/ns counters (tf_limiter_tier_stats2)."""

    import json
    import subprocess
    import sys
    out = subprocess.run(
        [sys.executable, "-c", LRU_CHILD],
        env=_limiter_env(1 << 30), capture_output=True, text=True,
        timeout=600)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-4000:]
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["ok"] and r["demoted_total"] >= 600 * (1 << 20)
    assert r["promoted_total"] >= 600 * (1 << 20)
    print(f"\ntier LRU: demote {r['demote_gbps']} GB/s, "
          f"promote {r['promote_gbps']} GB/s over SDMA prefetch")


def test_multi_tenant_oversubscription_with_pressure_controller():
    """BASELINE config 4 (scaled shape): N tenants whose caps
    oversubscribe the device run concurrently; each tenant's over-budget
    slabs live in the host tier, the hypervisor PressureController
    assigns budgets/flags, nobody OOMs, and the per-tenant spread stays
    tight. Full 4x96 GB shape: tools/demo_oversub.py (see
    profiles/oversub_4x96_r02.md)."""

    import json
    import os
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "tools", "demo_oversub.py"),
         "--scaled", "--shm-root", "/tmp/tf-testdemo-shm"],
        capture_output=True, text=True, timeout=900)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-3000:]
    line = next(l for l in out.stdout.splitlines()
                if l.startswith("SUMMARY "))
    r = json.loads(line[len("SUMMARY "):])
    assert not r["failures"], r["failures"]
    assert len(r["tenant_it_s"]) == r["config"]["tenants"]
    assert all(t > 0 for t in r["tenant_it_s"])
    # every tenant exceeded its HBM budget and used the host tier
    assert all(t["host_res"] > 0 for t in r["tiers"])
    # fairness between equal-QoS tenants
    assert r["spread_pct"] < 25.0, r
