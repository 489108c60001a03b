"""MoE token dispatchers.

Capability analog of reference megatron/core/transformer/moe/
token_dispatcher.py (MoEAlltoAllTokenDispatcher :372, MoEAllGatherTokenDispatcher
:230) + the A2A metadata computation.

xGMI note: the node is fully connected, so all_to_all_single is single-hop —
the reference's DeepEP/NVSHMEM fused path exists because NVLink-domain
crossing is expensive; here plain RCCL A2A (C12) is already topology-optimal.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from megatron_amd.parallel import grid as G
from megatron_amd.parallel.mappings import all_to_all


def expert_capacity_mask(top_idx: torch.Tensor, probs: torch.Tensor,
                         num_experts: int, capacity: int,
                         drop_policy: str = "probs") -> torch.Tensor:
    """Keep mask [T*k]: at most `capacity` (token, slot) assignments kept
    per expert (reference router capacity / moe_token_drop_policy).
    'probs' keeps the highest-probability assignments; 'position' the
    earliest tokens."""
    T, k = top_idx.shape
    n = T * k
    flat = top_idx.reshape(-1)
    if drop_policy == "probs":
        order = torch.argsort(probs.reshape(-1).float(), descending=True, stable=True)
    else:
        order = torch.arange(n, device=flat.device)
    priority = torch.empty(n, dtype=torch.long, device=flat.device)
    priority[order] = torch.arange(n, device=flat.device)
    # rank of each assignment within its expert, by priority
    grouped = torch.argsort(flat * (n + 1) + priority)
    counts = torch.bincount(flat, minlength=num_experts)
    starts = torch.cumsum(counts, 0) - counts
    ranks = torch.empty(n, dtype=torch.long, device=flat.device)
    ranks[grouped] = torch.arange(n, device=flat.device) - starts.repeat_interleave(counts)
    return ranks < capacity


def permute(tokens: torch.Tensor, top_idx: torch.Tensor, num_experts: int,
            keep_mask: torch.Tensor = None):
    """Replicate+sort tokens by target expert.

    tokens [T, h], top_idx [T, k] -> (permuted [n_kept, h], sort_order
    [n_kept] (global (token,slot) flat ids in expert order),
    tokens_per_expert [E]).  keep_mask [T*k] drops capacity-overflow
    assignments (their tokens ride the residual only).  HIP gather kernel
    (K12) replaces the index_select later; torch path is a single gather."""
    T, k = top_idx.shape
    flat_experts = top_idx.reshape(-1)  # [T*k] expert id per (token, slot)
    if keep_mask is not None:
        kept = keep_mask.nonzero(as_tuple=True)[0]
        order_within = torch.argsort(flat_experts[kept], stable=True)
        sort_order = kept[order_within]
    else:
        sort_order = torch.argsort(flat_experts, stable=True)
    src_token = sort_order // k  # originating token row
    permuted = tokens.index_select(0, src_token)
    tokens_per_expert = torch.bincount(flat_experts[sort_order], minlength=num_experts)
    return permuted, sort_order, tokens_per_expert


def unpermute(permuted: torch.Tensor, sort_order: torch.Tensor, probs: torch.Tensor, T: int):
    """Inverse of permute with prob weighting: out[t] = sum_k prob[t,k] * x[(t,k)]
    (dropped (t,k) slots contribute zero)."""
    k = probs.shape[1]
    out_flat = torch.zeros(T * k, permuted.shape[-1], dtype=permuted.dtype,
                           device=permuted.device)
    out_flat = out_flat.index_copy(0, sort_order, permuted)
    out_flat = out_flat.view(T, k, -1) * probs.unsqueeze(-1).to(permuted.dtype)
    return out_flat.sum(dim=1)


def pad_to_capacity(permuted: torch.Tensor, tokens_per_expert, capacity: int):
    """Pad/position each expert's rows into a fixed [E*capacity, h] buffer
    (reference moe_pad_expert_input_to_capacity): downstream grouped GEMMs
    see STATIC shapes (graph-capturable).  Returns (padded, row_index) where
    row_index[i] is the padded position of permuted row i (for unpadding)."""
    E = len(tokens_per_expert)
    counts = [int(x) for x in tokens_per_expert]
    total = permuted.shape[0]
    padded = permuted.new_zeros(E * capacity, permuted.shape[1])
    idx = torch.empty(total, dtype=torch.long, device=permuted.device)
    start = 0
    for e, n in enumerate(counts):
        n = min(n, capacity)
        idx[start : start + n] = torch.arange(
            e * capacity, e * capacity + n, device=permuted.device)
        start += n
    padded.index_copy_(0, idx, permuted)
    return padded, idx


def unpad_from_capacity(padded_out: torch.Tensor, row_index: torch.Tensor) -> torch.Tensor:
    return padded_out.index_select(0, row_index)


class MoEAlltoAllTokenDispatcher:
    """permute -> A2A(EP) -> sort-by-local-expert -> experts -> A2A -> unpermute."""

    def __init__(self, config):
        self.config = config
        self.num_experts = config.num_experts
        self.ep = G.get_expert_model_parallel_world_size() if G.grid_initialized() else 1
        self.num_local_experts = self.num_experts // max(self.ep, 1)
        self.group = G.get_grid().group("ep") if G.grid_initialized() else None

    def _keep_mask(self, tokens, probs, top_idx):
        cf = getattr(self.config, "moe_expert_capacity_factor", None)
        if not cf:
            return None
        import math

        T, k = top_idx.shape
        cap = max(1, math.ceil(T * k / self.num_experts * cf))
        return expert_capacity_mask(top_idx, probs, self.num_experts, cap,
                                    getattr(self.config, "moe_token_drop_policy", "probs"))

    def dispatch(self, tokens: torch.Tensor, probs: torch.Tensor, top_idx: torch.Tensor):
        T = tokens.shape[0]
        permuted, sort_order, tokens_per_expert = permute(
            tokens, top_idx, self.num_experts, self._keep_mask(tokens, probs, top_idx))
        self._sort_order = sort_order
        self._T = T
        self._probs = probs
        self._pad_index = None
        if self.ep == 1:
            self._restore = None
            return self._maybe_pad(permuted, tokens_per_expert)

        # per-EP-peer split sizes (each peer owns num_local_experts experts)
        counts_matrix = torch.empty(self.ep * self.num_experts, dtype=tokens_per_expert.dtype,
                                    device=tokens_per_expert.device)
        torch.distributed.all_gather_into_tensor(
            counts_matrix, tokens_per_expert.contiguous(), group=self.group
        )
        # ONE DtoH copy of the [ep, E] counts matrix is the layer's single
        # host sync point; every split list below is derived host-side
        # (reference cuda_sync_point discipline, token_dispatcher.py:453-460)
        counts_host = counts_matrix.view(self.ep, self.num_experts).cpu()  # [src_rank, expert]
        rank = torch.distributed.get_rank(group=self.group)
        my_slice = counts_host[:, rank * self.num_local_experts : (rank + 1) * self.num_local_experts]
        self._input_splits = counts_host[rank].view(self.ep, self.num_local_experts).sum(dim=1).tolist()
        self._output_splits = my_slice.sum(dim=1).tolist()  # tokens arriving from each peer

        recv = all_to_all(self.group, permuted, self._output_splits, self._input_splits)

        # received tokens are grouped by (src_rank, local_expert); resort to
        # (local_expert, src_rank) so each expert's tokens are contiguous
        # (reference sort_chunks_by_idxs moe_utils.py:628)
        self._chunk_sizes = my_slice.reshape(-1).tolist()  # [ep * n_local] in (rank, expert) order
        # _chunk_perm[i] indexes (rank-major) chunks in (expert, rank) order
        self._chunk_perm = [
            r * self.num_local_experts + e
            for e in range(self.num_local_experts)
            for r in range(self.ep)
        ]
        chunks = torch.split(recv, self._chunk_sizes)
        reordered = torch.cat([chunks[i] for i in self._chunk_perm], dim=0) if len(chunks) > 1 else recv
        tokens_per_local_expert = my_slice.sum(dim=0)  # host tensor: experts sync-free downstream
        self._restore = True
        return self._maybe_pad(reordered, tokens_per_local_expert)

    def _maybe_pad(self, permuted, tokens_per_expert):
        if not getattr(self.config, "moe_pad_expert_input_to_capacity", False):
            return permuted, tokens_per_expert
        import math

        cf = getattr(self.config, "moe_expert_capacity_factor", None)
        assert cf, "moe_pad_expert_input_to_capacity requires moe_expert_capacity_factor"
        T_total = self._T * self._probs.shape[1]
        cap = max(1, math.ceil(T_total / self.num_experts * cf))
        if self.ep > 1:
            # after the a2a each LOCAL expert holds up to `cap` tokens from
            # EACH of the ep source ranks
            cap *= self.ep
        padded, self._pad_index = pad_to_capacity(permuted, tokens_per_expert, cap)
        static_counts = torch.full((len(tokens_per_expert),), cap, dtype=torch.long)
        return padded, static_counts

    def combine(self, expert_out: torch.Tensor) -> torch.Tensor:
        if self._pad_index is not None:
            expert_out = unpad_from_capacity(expert_out, self._pad_index)
            self._pad_index = None
        if self.ep > 1:
            # invert the (expert, rank) reorder, then A2A back
            sizes = [self._chunk_sizes[i] for i in self._chunk_perm]
            chunks = torch.split(expert_out, sizes)
            inv = [0] * len(self._chunk_perm)
            for pos, orig in enumerate(self._chunk_perm):
                inv[orig] = pos
            expert_out = torch.cat([chunks[inv[i]] for i in range(len(inv))], dim=0) if len(chunks) > 1 else expert_out
            expert_out = all_to_all(self.group, expert_out, self._input_splits, self._output_splits)
        return unpermute(expert_out, self._sort_order, self._probs, self._T)


class _GatherTokensEP(torch.autograd.Function):
    """all-gather along dim 0 over the EP group; backward reduce-scatters."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        world = torch.distributed.get_world_size(group=group)
        out = torch.empty(world * x.shape[0], *x.shape[1:], dtype=x.dtype, device=x.device)
        torch.distributed.all_gather_into_tensor(out, x.contiguous(), group=group)
        return out

    @staticmethod
    def backward(ctx, g):
        world = torch.distributed.get_world_size(group=ctx.group)
        out = torch.empty(g.shape[0] // world, *g.shape[1:], dtype=g.dtype, device=g.device)
        torch.distributed.reduce_scatter_tensor(out, g.contiguous(), group=ctx.group)
        return out, None


class _ReduceScatterTokensEP(torch.autograd.Function):
    """reduce-scatter along dim 0 over the EP group; backward all-gathers."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        world = torch.distributed.get_world_size(group=group)
        out = torch.empty(x.shape[0] // world, *x.shape[1:], dtype=x.dtype, device=x.device)
        torch.distributed.reduce_scatter_tensor(out, x.contiguous(), group=group)
        return out

    @staticmethod
    def backward(ctx, g):
        world = torch.distributed.get_world_size(group=ctx.group)
        out = torch.empty(world * g.shape[0], *g.shape[1:], dtype=g.dtype, device=g.device)
        torch.distributed.all_gather_into_tensor(out, g.contiguous(), group=ctx.group)
        return out, None


class MoEAllGatherTokenDispatcher:
    """Every rank all-gathers the EP group's tokens, computes its LOCAL
    experts on the full set, and reduce-scatters the prob-weighted combine
    (reference MoEAllGatherTokenDispatcher :230).  At ep=1 it degrades to a
    pure local permute and doubles as the alltoall equivalence oracle."""

    def __init__(self, config):
        self.config = config
        self.num_experts = config.num_experts
        self.ep = G.get_expert_model_parallel_world_size() if G.grid_initialized() else 1
        self.num_local_experts = self.num_experts // max(self.ep, 1)
        self.group = G.get_grid().group("ep") if (G.grid_initialized() and self.ep > 1) else None

    def _keep_mask(self, tokens, probs, top_idx):
        cf = getattr(self.config, "moe_expert_capacity_factor", None)
        if not cf:
            return None
        import math

        T, k = top_idx.shape
        cap = max(1, math.ceil(T * k / self.num_experts * cf))
        return expert_capacity_mask(top_idx, probs, self.num_experts, cap,
                                    getattr(self.config, "moe_token_drop_policy", "probs"))

    def dispatch(self, tokens, probs, top_idx):
        if self.group is None:
            permuted, sort_order, tokens_per_expert = permute(
                tokens, top_idx, self.num_experts, self._keep_mask(tokens, probs, top_idx))
            self._sort_order, self._T, self._probs = sort_order, tokens.shape[0], probs
            return permuted, tokens_per_expert

        # EP>1: every rank sees the group's full token set and runs only its
        # local experts; tokens/probs gather differentiably (router grads
        # flow back to the owning rank via the backward reduce-scatter)
        rank = torch.distributed.get_rank(group=self.group)
        full_tokens = _GatherTokensEP.apply(tokens, self.group)          # [ep*T, h]
        full_probs = _GatherTokensEP.apply(probs, self.group)            # [ep*T, k]
        full_idx = torch.empty(self.ep * top_idx.shape[0], top_idx.shape[1],
                               dtype=top_idx.dtype, device=top_idx.device)
        torch.distributed.all_gather_into_tensor(full_idx, top_idx.contiguous(), group=self.group)

        keep = self._keep_mask(full_tokens, full_probs, full_idx)
        lo = rank * self.num_local_experts
        hi = lo + self.num_local_experts
        mine = ((full_idx >= lo) & (full_idx < hi)).reshape(-1)
        keep = mine if keep is None else (keep & mine)
        permuted, sort_order, tokens_per_expert = permute(
            full_tokens, full_idx, self.num_experts, keep)
        self._sort_order = sort_order
        self._T = full_tokens.shape[0]
        self._probs = full_probs
        return permuted, tokens_per_expert[lo:hi]

    def combine(self, expert_out):
        out = unpermute(expert_out, self._sort_order, self._probs, self._T)
        if self.group is None:
            return out
        return _ReduceScatterTokensEP.apply(out, self.group)
