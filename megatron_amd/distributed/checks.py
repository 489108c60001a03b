"""Cross-replica consistency checks (reference core/utils.py
check_param_hashes_across_dp_replicas, used by
--check-weight-hash-across-dp-replicas-interval)."""

from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist

from megatron_amd.parallel import grid as G


def _param_fingerprint(p: torch.Tensor) -> torch.Tensor:
    """Cheap order-independent fingerprint (sum + sumsq in fp64)."""
    x = p.detach().double()
    return torch.stack([x.sum(), (x * x).sum()])


def check_param_hashes_across_dp_replicas(models: List[torch.nn.Module]) -> bool:
    """True iff every parameter is bitwise-consistent (by fingerprint) across
    the DP group. Desynced replicas indicate silent corruption or a missed
    broadcast — the reference logs and optionally aborts."""
    group = G.get_data_parallel_group()
    if group is None or dist.get_world_size(group) == 1:
        return True
    fps = []
    for m in models:
        for p in m.parameters():
            fps.append(_param_fingerprint(p))
    if not fps:
        return True
    mine = torch.stack(fps)
    ref = mine.clone()
    dist.broadcast(ref, src=dist.get_process_group_ranks(group)[0], group=group)
    ok = torch.equal(mine, ref)
    agree = torch.tensor([1 if ok else 0])
    dist.all_reduce(agree, op=dist.ReduceOp.MIN, group=group)
    return bool(agree.item())


def check_grads_finite(models: List[torch.nn.Module]) -> bool:
    """NaN/Inf sweep over main grads (reference param_and_grad_buffer
    check_grads analog)."""
    for m in models:
        for p in m.parameters():
            g = getattr(p, "main_grad", None)
            if g is None:
                g = p.grad
            if g is not None and not torch.isfinite(g).all():
                return False
    return True
