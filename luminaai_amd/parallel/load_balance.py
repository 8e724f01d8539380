"""Expert load balancing across EP ranks.

Counterpart of the vendored ColossalAI LoadBalancer (SURVEY.md §2.4:
moe/load_balance.py:15 — beam-search expert re-placement applied via
apply_load_balance). Here the placement is a permutation of experts over
EP-rank slots chosen by greedy LPT (longest-processing-time) bin packing —
at the 8-slot scale of one xGMI node LPT is within a few percent of optimal
and costs microseconds.

Applying a placement exchanges expert weights with ONE all-gather per MoE
layer and re-slices locally; the optimizer must be rebuilt afterwards
(Trainer.apply_expert_load_balance does both). Fresh Adam moments for the
moved experts are the documented cost, as with expert add/prune.
"""

from __future__ import annotations

from typing import List, Sequence

import torch
import torch.distributed as dist


def plan_placement(loads: Sequence[float], ep_size: int) -> List[int]:
    """loads[e] = routed-token share of expert e. Returns a permutation
    `order` (len E): order[slot] = expert id, where slots [r*EL,(r+1)*EL)
    live on EP rank r — chosen so per-rank load sums are near-equal."""
    E = len(loads)
    assert E % ep_size == 0
    per = E // ep_size
    ranked = sorted(range(E), key=lambda e: -loads[e])
    bins: List[List[int]] = [[] for _ in range(ep_size)]
    sums = [0.0] * ep_size
    for e in ranked:
        # fullest-first tie-break keeps the permutation stable
        cand = [r for r in range(ep_size) if len(bins[r]) < per]
        r = min(cand, key=lambda r: sums[r])
        bins[r].append(e)
        sums[r] += loads[e]
    order: List[int] = []
    for b in bins:
        order.extend(sorted(b))
    return order


def imbalance(loads: Sequence[float], order: Sequence[int],
              ep_size: int) -> float:
    """max-rank-load / mean-rank-load under a placement."""
    E = len(order)
    per = E // ep_size
    sums = [sum(loads[e] for e in order[r * per:(r + 1) * per])
            for r in range(ep_size)]
    mean = sum(sums) / ep_size
    return max(sums) / max(mean, 1e-12)


@torch.no_grad()
def apply_placement(moe_layer, order: Sequence[int]):
    """Reorder a MoEFFNLayer's experts into `order` (bucket slot s holds
    expert order[s]). Under EP this all-gathers the full expert weights and
    keeps the local slice; without EP it is a local permutation. The layer's
    `placement` buffer maps global expert id -> bucket slot for routing."""
    E = moe_layer.num_experts
    ep = moe_layer.ep_size
    EL = moe_layer.num_local_experts
    device = moe_layer.w_gate_up.device
    order_t = torch.as_tensor(list(order), dtype=torch.long, device=device)
    inv = torch.empty_like(order_t)
    inv[order_t] = torch.arange(E, device=device)   # expert -> slot

    cur = getattr(moe_layer, "placement", None)
    if cur is None:
        cur_order = torch.arange(E, device=device)
    else:
        cur_order = torch.empty(E, dtype=torch.long, device=device)
        cur_order[cur] = torch.arange(E, device=device)  # slot -> expert

    if ep > 1:
        full_gu = torch.empty(E, *moe_layer.w_gate_up.shape[1:],
                              dtype=moe_layer.w_gate_up.dtype, device=device)
        full_dn = torch.empty(E, *moe_layer.w_down.shape[1:],
                              dtype=moe_layer.w_down.dtype, device=device)
        dist.all_gather_into_tensor(full_gu, moe_layer.w_gate_up.data.contiguous(),
                                    group=moe_layer.ep_group) \
            if dist.get_backend(moe_layer.ep_group) != "gloo" else \
            dist.all_gather(list(full_gu.chunk(ep)),
                            moe_layer.w_gate_up.data.contiguous(),
                            group=moe_layer.ep_group)
        if dist.get_backend(moe_layer.ep_group) == "gloo":
            dist.all_gather(list(full_dn.chunk(ep)),
                            moe_layer.w_down.data.contiguous(),
                            group=moe_layer.ep_group)
        else:
            dist.all_gather_into_tensor(full_dn, moe_layer.w_down.data.contiguous(),
                                        group=moe_layer.ep_group)
        # slot s currently holds expert cur_order[s]; want expert order[s]
        src_experts = cur_order            # gathered tensors are in slot order
        # expert -> current slot
        cur_slot = torch.empty(E, dtype=torch.long, device=device)
        cur_slot[cur_order] = torch.arange(E, device=device)
        from ..parallel.mesh import get_mesh
        r = get_mesh().ep_rank
        my_slots = torch.arange(r * EL, (r + 1) * EL, device=device)
        take = cur_slot[order_t[my_slots]]
        moe_layer.w_gate_up.data.copy_(full_gu[take])
        moe_layer.w_down.data.copy_(full_dn[take])
    else:
        cur_slot = torch.empty(E, dtype=torch.long, device=device)
        cur_slot[cur_order] = torch.arange(E, device=device)
        take = cur_slot[order_t]
        moe_layer.w_gate_up.data.copy_(moe_layer.w_gate_up.data[take].clone())
        moe_layer.w_down.data.copy_(moe_layer.w_down.data[take].clone())

    moe_layer.placement = inv          # expert id -> bucket slot
