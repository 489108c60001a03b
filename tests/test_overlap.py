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


def _tp2_linear_overlap_case(rank, world):
    from megatron_amd.config import TransformerConfig
    from megatron_amd.parallel.layers import ColumnParallelLinear
    from megatron_amd.parallel.random import model_parallel_seed

    G.initialize_model_parallel(tensor_parallel_size=2)
    model_parallel_seed(3)
    base = dict(num_layers=1, hidden_size=16, num_attention_heads=2, vocab_size=32,
                ffn_hidden_size=32, tensor_parallel_size=2, sequence_parallel=True,
                gradient_accumulation_fusion=False)
    cfg_plain = TransformerConfig(**base)
    cfg_ovl = TransformerConfig(**base, tp_comm_overlap=True)

    torch.manual_seed(7)
    lin_a = ColumnParallelLinear(16, 24, config=cfg_plain, bias=False)
    lin_b = ColumnParallelLinear(16, 24, config=cfg_ovl, bias=False)
    with torch.no_grad():
        lin_b.weight.copy_(lin_a.weight)

    torch.manual_seed(11)  # same shard input on both paths
    x = torch.randn(4, 2, 16)  # [s/tp, b, h]
    xa = x.clone().requires_grad_(True)
    xb = x.clone().requires_grad_(True)
    ya, _ = lin_a(xa)
    yb, _ = lin_b(xb)
    assert torch.allclose(ya, yb, atol=1e-6), (ya - yb).abs().max()
    ya.square().sum().backward()
    yb.square().sum().backward()
    assert torch.allclose(xa.grad, xb.grad, atol=1e-5)
    assert torch.allclose(lin_a.weight.grad, lin_b.weight.grad, atol=1e-5)


def test_tp2_column_linear_overlap_equivalence():
    spawn_dist(_tp2_linear_overlap_case, 2)
