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
