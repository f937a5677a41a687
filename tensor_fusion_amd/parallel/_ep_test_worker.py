"""2-rank gloo worker for tests/test_ep_cpu.py: EP-sharded MoE must
reproduce the single-process dense-MoE reference exactly."""
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.environ["TF_REPO"])
dist.init_process_group(backend="gloo")
rank = dist.get_rank()
world = dist.get_world_size()

from tensor_fusion_amd.parallel.ep import Expert, ExpertParallelMLP

DIM, INTER, NEXP, N = 32, 64, 4, 37
torch.manual_seed(1)
# reference: all experts in one process
ref = ExpertParallelMLP(DIM, INTER, NEXP, ep_size=1).float()
torch.manual_seed(2)
x = torch.randn(N, DIM)
with torch.no_grad():
    want = ref(x)

# EP model: shard the REFERENCE experts across ranks; same router
ep = ExpertParallelMLP(DIM, INTER, NEXP, ep_size=world).float()
with torch.no_grad():
    ep.router.weight.copy_(ref.router.weight)
    per = NEXP // world
    for i in range(per):
        src = ref.experts[rank * per + i]
        ep.experts[i].gate.weight.copy_(src.gate.weight)
        ep.experts[i].up.weight.copy_(src.up.weight)
        ep.experts[i].down.weight.copy_(src.down.weight)
with torch.no_grad():
    got = ep(x)
err = (got - want).abs().max().item()
assert err < 1e-5, f"rank {rank}: err {err}"
if rank == 0:
    # sanity: tokens actually crossed ranks (both ranks used)
    ids = ref.router(x).argmax(-1)
    used = {int(i) // per for i in ids}
    assert len(used) == world, used
    print("EP_OK", err)

# edge case: EVERY token routes to expert 0 — the non-owning rank's
# experts receive ZERO tokens and the all-to-all count handshake must
# move empty payloads without deadlock or shape errors
with torch.no_grad():
    # zero router weights: every logit ties at 0 and argmax picks
    # expert 0 deterministically for every token
    ref.router.weight.zero_()
    ep.router.weight.copy_(ref.router.weight)
    want2 = ref(x)
    got2 = ep(x)
err2 = (got2 - want2).abs().max().item()
assert err2 < 1e-5, f"rank {rank}: starved-expert err {err2}"
ids2 = ref.router(x).argmax(-1)
assert set(ids2.tolist()) == {0}
if rank == 0:
    print("EP_EMPTY_OK", err2)
dist.destroy_process_group()
