"""Chunked GEMM<->collective overlap numerics (gloo TP=2)."""

import torch
import torch.distributed as dist

from megatron_amd.parallel import grid as G
from megatron_amd.parallel.overlap import gemm_ring_reducescatter, ring_allgather_gemm

from tests.utils import init_single, spawn_dist


def test_single_rank_passthrough():
    init_single()
    x = torch.randn(6, 8)
    w = torch.randn(10, 8)
    assert torch.allclose(ring_allgather_gemm(x, w), x @ w.t())
    assert torch.allclose(gemm_ring_reducescatter(x, w), x @ w.t())


def _tp2_case(rank, world):
    G.initialize_model_parallel(tensor_parallel_size=2)
    group = G.get_grid().group("tp")
    torch.manual_seed(0)  # same full tensors on both ranks
    x_full = torch.randn(8, 16)
    w = torch.randn(12, 16)

    # AG + GEMM: each rank holds its sequence shard
    shard = x_full[rank * 4:(rank + 1) * 4]
    y = ring_allgather_gemm(shard, w, group=group)
    assert torch.allclose(y, x_full @ w.t(), atol=1e-5)

    # GEMM + RS: per-rank partial inputs (split k) must sum then scatter
    k_shard = x_full[:, rank * 8:(rank + 1) * 8].contiguous()
    w_shard = w[:, rank * 8:(rank + 1) * 8].contiguous()
    y_local = gemm_ring_reducescatter(k_shard, w_shard, group=group)
    full = x_full @ w.t()  # == sum of per-rank partials
    assert torch.allclose(y_local, full[rank * 4:(rank + 1) * 4], atol=1e-5), \
        (y_local - full[rank * 4:(rank + 1) * 4]).abs().max()


def test_tp2_overlap_equivalence():
    spawn_dist(_tp2_case, 2)
