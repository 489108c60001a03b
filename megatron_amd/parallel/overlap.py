"""Chunked GEMM <-> collective overlap (TE userbuffers analog).

Capability analog of reference TE userbuffers integration (SURVEY.md §8.5:
`TELayerNormColumnParallelLinear(ub_overlap_ag=...)` etc.): under sequence
parallelism the big TP collectives (all-gather activations before a
column-linear, reduce-scatter after a row-linear) serialize with the GEMMs
they feed.  Decomposing the collective into tp-1 ring steps lets each
received chunk's GEMM run while the next chunk is in flight — on one MI355X
node each ring hop is a single direct xGMI link, and the GEMM chunks keep
the MFMA pipes busy behind it.

Functional building blocks (used by the overlap-enabled linear paths and
benchmarked standalone):
  * ring_allgather_gemm: y = allgather(x_shard) @ w.T  without ever
    materializing the gather as a blocking step;
  * gemm_ring_reducescatter: y_local = reducescatter_seq(x @ w.T) with each
    sequence chunk's partial reduced while the next chunk's GEMM runs.

On CPU (gloo) the "overlap" degenerates to interleaved execution with
identical numerics, which is what the unit tests pin down.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


def _ranks(group):
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    # global ranks of the group in group-rank order
    if group is None or group is dist.group.WORLD:
        glob = list(range(world))
    else:
        glob = dist.get_process_group_ranks(group)
    return world, rank, glob


def ring_allgather_gemm(x_shard: torch.Tensor, weight: torch.Tensor,
                        group=None, bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """x_shard [n/tp, k] (this rank's sequence chunk), weight [m, k] ->
    y [n, m] == allgather(x_shard) @ weight.T (+bias).

    Ring: at step s the chunk originally from rank (r+s) mod tp arrives;
    its GEMM is issued while the chunk is forwarded to the next peer."""
    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        y = x_shard @ weight.t()
        return y + bias if bias is not None else y
    world, rank, glob = _ranks(group)
    nloc, k = x_shard.shape
    out = torch.empty(world * nloc, weight.shape[0], dtype=x_shard.dtype,
                      device=x_shard.device)
    send_to = glob[(rank + 1) % world]
    recv_from = glob[(rank - 1) % world]
    cur = x_shard.contiguous()
    for step in range(world):
        src = (rank - step) % world  # owner of `cur`
        if step < world - 1:
            nxt = torch.empty_like(cur)
            reqs = dist.batch_isend_irecv([
                dist.P2POp(dist.isend, cur, send_to, group=group),
                dist.P2POp(dist.irecv, nxt, recv_from, group=group),
            ])
        # GEMM overlaps the in-flight ring hop
        y = cur @ weight.t()
        if bias is not None:
            y = y + bias
        out[src * nloc:(src + 1) * nloc] = y
        if step < world - 1:
            for r in reqs:
                r.wait()
            cur = nxt
    return out


def gemm_ring_reducescatter(x: torch.Tensor, weight: torch.Tensor,
                            group=None) -> torch.Tensor:
    """x [n, k] (full sequence, partial-k inputs on each rank), weight
    [m, k] -> y_local [n/tp, m] == reduce_scatter_seq(x @ weight.T).

    The sequence is cut into tp chunks; chunk j's partial GEMM result is
    reduced to its owner asynchronously while chunk j+1's GEMM runs."""
    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        return x @ weight.t()
    world, rank, glob = _ranks(group)
    n = x.shape[0]
    assert n % world == 0
    nloc = n // world
    pending = []
    partials: List[torch.Tensor] = []
    for j in range(world):
        part = x[j * nloc:(j + 1) * nloc] @ weight.t()  # [nloc, m] partial sum
        partials.append(part)
        # async reduce to the chunk's owner; later GEMMs overlap the wire time
        work = dist.reduce(part, dst=glob[j], op=dist.ReduceOp.SUM,
                           group=group, async_op=True)
        pending.append(work)
    for w in pending:
        w.wait()
    return partials[rank]
