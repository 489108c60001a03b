"""Hybrid (per-sample) context parallelism: wiring the BalancedCPScheduler
into the ring-attention path.

Capability analog of reference megatron/core/pipeline_parallel/
hybrid_cp_schedule.py + THD PackedSeqParams usage: variable-length samples
each get their own CP group size (power of two); this module owns

  * the subgroup communicators (one per aligned power-of-two rank window),
  * the per-rank batch builder (each rank gets its zigzag slice of every
    sample assigned to it, packed with cu_seqlens-style metadata),
  * the attention runner: every sample's ring attention runs on ITS
    subgroup; ranks walk their assignments in global sample order so the
    members of each subgroup enter the same ring together.

On one MI355X node the aligned windows keep each subgroup's KV ring on
direct xGMI links.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist

from megatron_amd.parallel.balanced_cp import BalancedCPScheduler, CPAssignment
from megatron_amd.parallel.context_parallel import ring_attention, slice_for_cp_rank


class HybridCPGroups:
    """Pre-created communicators for every aligned power-of-two window.

    Group creation is collective, so EVERY rank builds the full set once."""

    def __init__(self, world_size: int):
        self.world = world_size
        self._groups: Dict[Tuple[int, int], object] = {}
        size = 2
        while size <= world_size:
            for start in range(0, world_size, size):
                ranks = list(range(start, start + size))
                self._groups[(size, start)] = (
                    dist.new_group(ranks=ranks) if dist.is_initialized() else None)
            size *= 2

    def group_for(self, ranks: Sequence[int]):
        ranks = tuple(ranks)
        if len(ranks) == 1:
            return None
        return self._groups[(len(ranks), ranks[0])]


@dataclass
class HybridCPBatch:
    """This rank's packed slices: one entry per assignment (sample order)."""

    assignments: List[CPAssignment]
    slices: List[torch.Tensor]          # each [s_i/cp_i, ...] zigzag slice
    cu_seqlens: torch.Tensor            # local packed offsets

    def packed(self) -> torch.Tensor:
        return torch.cat(self.slices, dim=0) if self.slices else torch.empty(0)


def build_hybrid_cp_batch(samples: Sequence[torch.Tensor],
                          per_rank: Dict[int, List[CPAssignment]],
                          rank: int, seq_dim: int = 0) -> HybridCPBatch:
    """Slice every sample assigned to `rank` for its (cp_rank, cp_size)
    using the load-balanced zigzag layout (sample length must divide by
    2*cp_size, the ring-attention contract)."""
    assigns = sorted(per_rank.get(rank, []), key=lambda a: a.sample)
    slices = []
    lens = [0]
    for a in assigns:
        t = samples[a.sample]
        if a.cp_size == 1:
            sl = t
        else:
            sl = slice_for_cp_rank(t, a.cp_rank, a.cp_size, seq_dim=seq_dim, mode="p2p")
        slices.append(sl)
        lens.append(lens[-1] + sl.shape[seq_dim])
    return HybridCPBatch(assigns, slices, torch.tensor(lens, dtype=torch.long))


def hybrid_cp_attention(batch_qkv: List[Tuple[torch.Tensor, torch.Tensor, torch.Tensor]],
                        assignments: List[CPAssignment],
                        groups: HybridCPGroups,
                        scale: Optional[float] = None) -> List[torch.Tensor]:
    """Run each assigned sample's causal attention on its own CP subgroup.

    batch_qkv[i] are this rank's zigzag q/k/v slices for assignments[i]
    ([s_i/cp_i, b, h, d]).  Assignments must be in global sample order on
    every rank (build_hybrid_cp_batch guarantees it), so subgroup members
    pair up ring-step for ring-step."""
    outs = []
    for (q, k, v), a in zip(batch_qkv, assignments):
        if a.cp_size == 1:
            from megatron_amd import ops

            outs.append(ops.flash_attention(q, k, v, causal=True, scale=scale))
        else:
            outs.append(ring_attention(q, k, v, scale=scale,
                                       group=groups.group_for(a.ranks)))
    return outs
