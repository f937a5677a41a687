"""Tiny-config Llama correctness on CPU (numerics vs full forward)."""
import torch

from tensor_fusion_amd.models.llama import CONFIGS, Llama, build_model


def test_decode_matches_full_forward():
    torch.manual_seed(0)
    cfg = CONFIGS["tiny"]
    model = Llama(cfg).eval().float()
    B, T = 2, 12
    toks = torch.randint(0, cfg.vocab, (B, T))
    with torch.no_grad():
        full = model(toks)  # [B, T, V]
        caches = model.make_kv_cache(B, 64, "cpu", torch.float32)
        # prefill first 6, then decode one at a time
        model(toks[:, :6], pos=torch.arange(6), caches=caches)
        outs = []
        for t in range(6, T):
            o = model(toks[:, t:t + 1], pos=torch.tensor([t]), caches=caches)
            outs.append(o)
        inc = torch.cat(outs, dim=1)
    assert torch.allclose(full[:, 6:], inc, atol=1e-4), \
        (full[:, 6:] - inc).abs().max()


def test_build_model_cpu():
    m = build_model("tiny", device="cpu", dtype=torch.float32)
    out = m(torch.randint(0, 256, (1, 4)))
    assert out.shape == (1, 4, 256)
    assert torch.isfinite(out).all()
