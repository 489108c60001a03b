"""TP layer numerics: TP=2 over gloo must match a single-process reference.

Pattern mirrors reference tests/unit_tests/tensor_parallel/test_layers.py:
build the full weight everywhere (same seed), shard it by rank, compare
outputs and input-grads against the unsharded computation.
"""

import torch
import torch.distributed as dist

from megatron_amd.config import TransformerConfig
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.cross_entropy import vocab_parallel_cross_entropy
from megatron_amd.parallel.layers import ColumnParallelLinear, RowParallelLinear, VocabParallelEmbedding
from megatron_amd.parallel.random import model_parallel_seed

from tests.utils import assert_close, init_single, spawn_dist


def _cfg(tp, sp=False):
    return TransformerConfig(
        num_layers=1, hidden_size=32, num_attention_heads=4, vocab_size=64,
        tensor_parallel_size=tp, sequence_parallel=sp,
        gradient_accumulation_fusion=False, async_tensor_model_parallel_allreduce=True,
    )


def _column_parallel_case(rank, world, sp):
    G.initialize_model_parallel(tensor_parallel_size=world)
    model_parallel_seed(1234)
    cfg = _cfg(world, sp)
    torch.manual_seed(7)
    full_w = torch.randn(48, 32)
    x = torch.randn(8, 2, 32, requires_grad=True)  # [s, b, h]

    layer = ColumnParallelLinear(32, 48, config=cfg, bias=False)
    with torch.no_grad():
        layer.weight.copy_(full_w.chunk(world, dim=0)[rank])

    inp = x.chunk(world, dim=0)[rank].detach().requires_grad_(True) if sp else x
    out, _ = layer(inp)
    ref = torch.matmul(x, full_w.t())
    ref_local = ref.chunk(world, dim=-1)[rank]
    assert_close(out, ref_local.detach())

    g = torch.ones_like(out)
    out.backward(g)
    ref.backward(torch.ones_like(ref))
    if sp:
        assert_close(inp.grad, x.grad.chunk(world, dim=0)[rank])
    else:
        assert_close(inp.grad, x.grad)


def test_column_parallel_tp2():
    spawn_dist(_column_parallel_case, 2, False)


def test_column_parallel_tp2_sp():
    spawn_dist(_column_parallel_case, 2, True)


def _row_parallel_case(rank, world, sp):
    G.initialize_model_parallel(tensor_parallel_size=world)
    model_parallel_seed(1234)
    cfg = _cfg(world, sp)
    torch.manual_seed(7)
    full_w = torch.randn(32, 48)
    x = torch.randn(8, 2, 48, requires_grad=True)

    layer = RowParallelLinear(48, 32, config=cfg, bias=False)
    with torch.no_grad():
        layer.weight.copy_(full_w.chunk(world, dim=1)[rank])

    x_local = x.detach().chunk(world, dim=-1)[rank].requires_grad_(True)
    out, _ = layer(x_local)
    ref = torch.matmul(x, full_w.t())
    if sp:
        assert_close(out, ref.detach().chunk(world, dim=0)[rank], rtol=1e-4, atol=1e-4)
        out.backward(torch.ones(out.shape))
    else:
        assert_close(out, ref.detach(), rtol=1e-4, atol=1e-4)
        out.backward(torch.ones_like(out))
    ref.backward(torch.ones_like(ref))
    assert_close(x_local.grad, x.grad.chunk(world, dim=-1)[rank])


def test_row_parallel_tp2():
    spawn_dist(_row_parallel_case, 2, False)


def test_row_parallel_tp2_sp():
    spawn_dist(_row_parallel_case, 2, True)


def _vocab_embedding_case(rank, world):
    G.initialize_model_parallel(tensor_parallel_size=world)
    model_parallel_seed(1234)
    cfg = _cfg(world)
    torch.manual_seed(7)
    full_w = torch.randn(64, 32)
    ids = torch.randint(0, 64, (2, 8))

    emb = VocabParallelEmbedding(64, 32, config=cfg)
    with torch.no_grad():
        emb.weight.copy_(full_w.chunk(world, dim=0)[rank])
    out = emb(ids)  # [s, b, h]
    ref = torch.nn.functional.embedding(ids, full_w).transpose(0, 1)
    assert_close(out, ref)


def test_vocab_embedding_tp2():
    spawn_dist(_vocab_embedding_case, 2)


def _vocab_ce_case(rank, world):
    G.initialize_model_parallel(tensor_parallel_size=world)
    model_parallel_seed(1234)
    torch.manual_seed(7)
    logits = torch.randn(6, 2, 64)
    target = torch.randint(0, 64, (6, 2))
    local = logits.chunk(world, dim=-1)[rank].detach().requires_grad_(True)
    loss = vocab_parallel_cross_entropy(local, target)
    ref = torch.nn.functional.cross_entropy(
        logits.reshape(-1, 64), target.reshape(-1), reduction="none"
    ).view(6, 2)
    assert_close(loss, ref, rtol=1e-5, atol=1e-5)
    loss.sum().backward()
    logits_ref = logits.detach().requires_grad_(True)
    torch.nn.functional.cross_entropy(
        logits_ref.reshape(-1, 64), target.reshape(-1), reduction="sum"
    ).backward()
    assert_close(local.grad, logits_ref.grad.chunk(world, dim=-1)[rank], rtol=1e-5, atol=1e-5)


def test_vocab_parallel_cross_entropy_tp2():
    spawn_dist(_vocab_ce_case, 2)


def test_vocab_parallel_cross_entropy_tp1():
    init_single(tp=1)
    torch.manual_seed(7)
    logits = torch.randn(6, 2, 64, requires_grad=True)
    target = torch.randint(0, 64, (6, 2))
    loss = vocab_parallel_cross_entropy(logits, target)
    ref = torch.nn.functional.cross_entropy(logits.detach().reshape(-1, 64), target.reshape(-1), reduction="none").view(6, 2)
    assert_close(loss, ref, rtol=1e-5, atol=1e-5)


def test_global_memory_buffer_reuse():
    import torch

    from megatron_amd.parallel.memory_buffer import GlobalMemoryBuffer

    gmb = GlobalMemoryBuffer()
    a = gmb.get_tensor((4, 8), torch.float32, "x")
    ptr = a.data_ptr()
    b = gmb.get_tensor((2, 8), torch.float32, "x")  # smaller: same storage
    assert b.data_ptr() == ptr
    c = gmb.get_tensor((8, 8), torch.float32, "x")  # bigger: regrown
    assert c.numel() == 64
    d = gmb.get_tensor((4, 8), torch.float64, "x")  # dtype keyed separately
    assert d.dtype == torch.float64
