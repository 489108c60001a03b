"""Gradient clipping by global L2 norm.

Capability analog of reference megatron/core/optimizer/clip_grads.py
(multi-tensor l2norm + all-reduce over the model-parallel group).

Norm accounting rule: a gradient element must be counted exactly once across
the whole job.  TP-sharded params are distinct per TP rank (count each);
TP-duplicated params (norm weights, biases) are counted only on tp_rank 0;
the squared norm is all-reduced over the model-parallel (tp x cp x pp) group.
The distributed optimizer additionally shards over DP and all-reduces there.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from megatron_amd import ops
from megatron_amd.parallel import grid as G


def param_is_not_tensor_parallel_duplicate(param) -> bool:
    if getattr(param, "tensor_parallel", False):
        return True
    if not G.grid_initialized():
        return True
    if getattr(param, "is_expert_parallel", False):
        # expert params are duplicated over etp (not tp) when unsharded
        return G.get_grid().rank_in("etp") == 0
    return G.get_tensor_model_parallel_rank() == 0


def _local_sq(grads: List[torch.Tensor], device) -> torch.Tensor:
    if grads:
        local = ops.l2_norm(grads)
        return (local * local).to(torch.float32)
    return torch.zeros((), dtype=torch.float32, device=device)


def _reduce_sq(sq: torch.Tensor, groups: list) -> torch.Tensor:
    for g in groups:
        if g is not None and dist.is_initialized() and dist.get_world_size(group=g) > 1:
            dist.all_reduce(sq, group=g)
    return sq


def get_grad_norm(
    grads: List[torch.Tensor],
    extra_groups: Optional[list] = None,
    expert_grads: Optional[List[torch.Tensor]] = None,
    expert_extra_groups: Optional[list] = None,
) -> torch.Tensor:
    """grads: dense-param grad tensors already filtered for duplicates;
    expert_grads: expert-parallel param grads (sharded over etp x ep).
    Returns the global L2 norm.

    Dense contributions are reduced over tp x pp (NOT cp: grads are already
    reduced/sharded over dp_cp which spans cp) plus ``extra_groups`` (e.g.
    dp_cp for the distributed optimizer's shards).  Expert contributions are
    reduced over etp x ep x pp plus ``expert_extra_groups`` (e.g. edp)."""
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    if grads:
        dev = grads[0].device
    elif expert_grads:
        dev = expert_grads[0].device
    sq = _local_sq(grads, dev)
    groups = []
    if G.grid_initialized():
        groups.append(G.get_grid().group("tp_pp"))
    if extra_groups:
        groups.extend(extra_groups)
    sq = _reduce_sq(sq, groups)
    if expert_grads is not None and (expert_grads or G.grid_initialized()):
        esq = _local_sq(expert_grads, dev)
        egroups = []
        if G.grid_initialized():
            egroups.append(G.get_grid().group("etp_ep_pp"))
        if expert_extra_groups:
            egroups.extend(expert_extra_groups)
        esq = _reduce_sq(esq, egroups)
        sq = sq + esq
    return torch.sqrt(sq)


def split_grads_for_norm(params, grads):
    """Filter TP duplicates and split (dense, expert) grads for get_grad_norm."""
    dense, expert = [], []
    for p, g in zip(params, grads):
        if not param_is_not_tensor_parallel_duplicate(p):
            continue
        (expert if getattr(p, "is_expert_parallel", False) else dense).append(g)
    return dense, expert


def clip_grads_by_total_norm(grads: List[torch.Tensor], max_norm: float, total_norm: torch.Tensor):
    clip_coeff = max_norm / (total_norm + 1.0e-6)
    if clip_coeff.item() < 1.0:
        torch._foreach_mul_(grads, clip_coeff.item())
