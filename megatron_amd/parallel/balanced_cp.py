"""Balanced hybrid context parallelism scheduling.

Capability analog of reference megatron/core/pipeline_parallel/
hybrid_cp_schedule.py:14 (`BalancedCPScheduler`): with variable-length
samples, a fixed CP degree wastes ranks on short samples (attention work is
quadratic in sequence length).  Instead each sample is given its own CP
group size (a power of two, larger for longer samples) and samples are
packed onto the DPxCP rank grid so every rank carries a near-equal amount
of quadratic attention work.

This module is pure scheduling: it emits per-rank assignments
(sample, cp_size, cp_rank) that the training loop feeds to the existing
ring-attention context parallelism (parallel/context_parallel.py) with a
per-sample group.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Sequence


@dataclass(frozen=True)
class CPAssignment:
    sample: int      # index into the global batch
    cp_size: int     # CP group size for this sample (power of two)
    cp_rank: int     # this rank's position inside the sample's CP group
    ranks: tuple     # the full (aligned, contiguous) CP group


def pick_cp_size(seq_len: int, max_cp: int, chunk_target: int) -> int:
    """Smallest power-of-two CP degree that brings the per-rank chunk down
    to `chunk_target` tokens (reference: per-sample CP group sizes)."""
    cp = 1
    while cp < max_cp and seq_len // cp > chunk_target:
        cp *= 2
    return cp


def quadratic_cost(seq_len: int) -> float:
    """Relative attention work for a causal sample of length s (~s^2/2)."""
    return 0.5 * float(seq_len) * float(seq_len)


class BalancedCPScheduler:
    """Greedy LPT packing of variable-cp-size samples onto `world` ranks.

    Samples are sorted by descending per-rank cost; each is placed on the
    aligned `cp_size`-wide rank window whose current maximum load is lowest
    (alignment keeps CP groups inside natural power-of-two boundaries, which
    on one MI355X node keeps a group's xGMI traffic on direct links).
    """

    def __init__(self, world_size: int, max_cp: int, chunk_target: int = 4096):
        assert world_size & (world_size - 1) == 0, "world must be a power of two"
        assert max_cp <= world_size
        self.world = world_size
        self.max_cp = max_cp
        self.chunk_target = chunk_target

    def schedule(self, seq_lens: Sequence[int]) -> Dict[int, List[CPAssignment]]:
        loads = [0.0] * self.world
        per_rank: Dict[int, List[CPAssignment]] = {r: [] for r in range(self.world)}

        order = sorted(
            range(len(seq_lens)),
            key=lambda i: quadratic_cost(seq_lens[i]) / pick_cp_size(seq_lens[i], self.max_cp, self.chunk_target),
            reverse=True,
        )
        for i in order:
            s = seq_lens[i]
            cp = pick_cp_size(s, self.max_cp, self.chunk_target)
            share = quadratic_cost(s) / cp
            # candidate windows: contiguous, cp-aligned
            best_start, best_load = 0, float("inf")
            for start in range(0, self.world, cp):
                window_max = max(loads[start : start + cp])
                if window_max < best_load:
                    best_load, best_start = window_max, start
            ranks = tuple(range(best_start, best_start + cp))
            for j, r in enumerate(ranks):
                loads[r] += share
                per_rank[r].append(CPAssignment(sample=i, cp_size=cp, cp_rank=j, ranks=ranks))
        return per_rank

    def balance_ratio(self, seq_lens: Sequence[int]) -> float:
        """max/mean per-rank quadratic load of the produced schedule."""
        per_rank = self.schedule(seq_lens)
        loads = []
        for r, assigns in per_rank.items():
            loads.append(sum(quadratic_cost(seq_lens[a.sample]) / a.cp_size for a in assigns))
        mean = sum(loads) / len(loads)
        return max(loads) / mean if mean > 0 else 1.0
