"""Generic named-dimension communicator grid.

Capability analog of reference megatron/core/hyper_comm_grid.py:46
(`HyperCommGrid`): an N-D factoring of the world with caller-chosen dim
names, producing a process group (or rank list) for any subset of dims —
used by multi-module models (vision encoder on its own sub-grid next to
the LLM's tp x dp grid) without touching the global training grid
singleton (parallel/grid.py remains the tp-cp-ep-dp-pp fast path).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import torch.distributed as dist


class HyperCommGrid:
    """dims: ordered (name, size) pairs, fastest-varying first.

    rank = sum_i idx[i] * stride[i], stride fastest-first — the same
    factoring convention as the training grid."""

    def __init__(self, dim_names: Sequence[str], dim_sizes: Sequence[int],
                 world_size: Optional[int] = None, rank_offset: int = 0,
                 create_groups: bool = True):
        assert len(dim_names) == len(set(dim_names)), "duplicate dim names"
        assert len(dim_names) == len(dim_sizes)
        self.names = list(dim_names)
        self.sizes = list(dim_sizes)
        total = 1
        for s in self.sizes:
            total *= s
        self.total = total
        self.rank_offset = rank_offset
        if world_size is not None:
            assert total <= world_size - rank_offset, \
                f"grid of {total} ranks does not fit world {world_size} at offset {rank_offset}"
        self.strides = []
        acc = 1
        for s in self.sizes:
            self.strides.append(acc)
            acc *= s
        self._groups: Dict[Tuple[str, ...], object] = {}
        self._create = create_groups and dist.is_initialized()

    # -- coordinates ---------------------------------------------------------
    def coords_of(self, global_rank: int) -> Dict[str, int]:
        r = global_rank - self.rank_offset
        assert 0 <= r < self.total, f"rank {global_rank} not in this grid"
        return {n: (r // st) % s for n, s, st in zip(self.names, self.sizes, self.strides)}

    def rank_at(self, **coords) -> int:
        r = 0
        for n, s, st in zip(self.names, self.sizes, self.strides):
            c = coords.get(n, 0)
            assert 0 <= c < s, (n, c, s)
            r += c * st
        return r + self.rank_offset

    # -- rank enumeration ----------------------------------------------------
    def ranks_for(self, dims: Sequence[str], anchor_rank: int) -> List[int]:
        """Global ranks of the group spanned by `dims` that contains
        `anchor_rank` (all other dims held at the anchor's coordinates)."""
        anchor = self.coords_of(anchor_rank)
        span = [n for n in self.names if n in dims]
        assert set(dims) == set(span), f"unknown dims in {dims}"
        ranks = []

        def rec(i, coords):
            if i == len(span):
                ranks.append(self.rank_at(**coords))
                return
            for c in range(self.sizes[self.names.index(span[i])]):
                coords[span[i]] = c
                rec(i + 1, coords)

        rec(0, dict(anchor))
        return sorted(ranks)

    def all_groups_for(self, dims: Sequence[str]) -> List[List[int]]:
        """Every disjoint `dims` group in the grid (rank lists)."""
        seen, out = set(), []
        for r in range(self.rank_offset, self.rank_offset + self.total):
            key = tuple(self.ranks_for(dims, r))
            if key not in seen:
                seen.add(key)
                out.append(list(key))
        return out

    def group_for(self, dims: Sequence[str], anchor_rank: Optional[int] = None):
        """The torch.distributed group over `dims` containing this rank.
        Creates (and caches) groups collectively — every rank must call with
        the same dims."""
        if not self._create:
            return None
        me = dist.get_rank() if anchor_rank is None else anchor_rank
        key = tuple(sorted(dims))
        cached = self._groups.get(key)
        if cached is None:
            mine = None
            for ranks in self.all_groups_for(dims):
                g = dist.new_group(ranks=ranks)
                if me in ranks:
                    mine = g
            self._groups[key] = {"mine": mine}
            return mine
        return cached["mine"]
