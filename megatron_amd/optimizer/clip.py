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
    return (not G.grid_initialized()) or G.get_tensor_model_parallel_rank() == 0


def get_grad_norm(grads: List[torch.Tensor], extra_groups: Optional[list] = None) -> torch.Tensor:
    """grads: tensors already filtered for duplicates; returns global L2 norm,
    reduced over the model-parallel group plus any ``extra_groups`` (e.g. the
    DP group for the distributed optimizer)."""
    if grads:
        local = ops.l2_norm(grads)
        sq = (local * local).to(torch.float32)
    else:
        dev = "cuda" if torch.cuda.is_available() else "cpu"
        sq = torch.zeros((), dtype=torch.float32, device=dev)
    groups = []
    if G.grid_initialized():
        groups.append(G.get_grid().group("mp"))
    if extra_groups:
        groups.extend(extra_groups)
    for g in groups:
        if g is not None and dist.is_initialized() and dist.get_world_size(group=g) > 1:
            dist.all_reduce(sq, group=g)
    return torch.sqrt(sq)


def clip_grads_by_total_norm(grads: List[torch.Tensor], max_norm: float, total_norm: torch.Tensor):
    clip_coeff = max_norm / (total_norm + 1.0e-6)
    if clip_coeff.item() < 1.0:
        torch._foreach_mul_(grads, clip_coeff.item())
