"""Expert parallelism — MoE token routing over RCCL all-to-all on xGMI.

SURVEY §2.5: the reference schedules multi-GPU jobs but implements no
collectives; the MI355X plan routes gang-scheduled EP workloads over
RCCL's all-to-all, which maps perfectly onto the xGMI full mesh (every
intra-node GPU pair has a dedicated point-to-point link, so all-to-all
is the one collective that uses ALL 7 links of every GPU at once —
unlike ring all-reduce, which is single-link-bound).

ExpertParallelMLP: a switch-routed (top-1) MoE layer whose experts are
sharded one-group-per-rank. forward():
    router logits → expert id per token
    all_to_all_single (token counts) → exchange counts
    all_to_all_single (tokens)       → tokens to their expert's rank
    local expert MLPs                → computed where the weights live
    all_to_all_single (results)      → back to the owning rank

Like TP, the vGPU limiter charges launches only — an in-flight
all-to-all is never split by throttling. Backend "nccl"
is RCCL on ROCm; "gloo" runs the same code on CPU for CI.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F


def _world():
    return dist.get_world_size() if dist.is_initialized() else 1


def _rank():
    return dist.get_rank() if dist.is_initialized() else 0


class Expert(nn.Module):
    def __init__(self, dim: int, inter: int):
        super().__init__()
        self.gate = nn.Linear(dim, inter, bias=False)
        self.up = nn.Linear(dim, inter, bias=False)
        self.down = nn.Linear(inter, dim, bias=False)

    def forward(self, x):
        return self.down(F.silu(self.gate(x)) * self.up(x))


class ExpertParallelMLP(nn.Module):
    """num_experts total, evenly sharded across the EP group; each rank
    holds num_experts // world experts and computes tokens routed to
    them."""

    def __init__(self, dim: int, inter: int, num_experts: int,
                 ep_size: Optional[int] = None):
        super().__init__()
        self.dim = dim
        self.num_experts = num_experts
        self.ep = ep_size or _world()
        assert num_experts % self.ep == 0, (num_experts, self.ep)
        self.local_experts = num_experts // self.ep
        self.router = nn.Linear(dim, num_experts, bias=False)
        self.experts = nn.ModuleList(
            Expert(dim, inter) for _ in range(self.local_experts))

    # ---------------------------------------------------------- routing

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        orig_shape = x.shape
        t = x.reshape(-1, self.dim)
        n = t.shape[0]
        logits = self.router(t)
        expert_of = logits.argmax(-1)  # top-1 switch routing
        dest_rank = expert_of // self.local_experts

        if self.ep == 1:
            out = torch.empty_like(t)
            for e in range(self.local_experts):
                sel = expert_of == e
                if sel.any():
                    out[sel] = self.experts[e](t[sel])
            return out.reshape(orig_shape)

        # sort tokens by destination rank so each rank's slice is
        # contiguous for all_to_all_single
        order = torch.argsort(dest_rank, stable=True)
        t_sorted = t[order]
        send_counts = torch.bincount(dest_rank, minlength=self.ep)

        recv_counts = torch.empty_like(send_counts)
        dist.all_to_all_single(recv_counts, send_counts)
        in_splits: List[int] = send_counts.tolist()
        out_splits: List[int] = recv_counts.tolist()

        received = t_sorted.new_empty(sum(out_splits), self.dim)
        dist.all_to_all_single(received, t_sorted.contiguous(),
                               out_splits, in_splits)
        # which local expert for each received token: re-run the router
        # slice for received tokens (weights are replicated) — avoids a
        # second index exchange
        local_ids = (self.router(received).argmax(-1)
                     - _rank() * self.local_experts)
        computed = torch.empty_like(received)
        for e in range(self.local_experts):
            sel = local_ids == e
            if sel.any():
                computed[sel] = self.experts[e](received[sel])

        returned = t_sorted.new_empty(n, self.dim)
        dist.all_to_all_single(returned, computed, in_splits, out_splits)
        # undo the destination sort
        out = torch.empty_like(returned)
        out[order] = returned
        return out.reshape(orig_shape)
