"""FSDP (ZeRO-3) tests: DP=2 training equivalence vs a single-process
model, sharded memory behavior, state_dict round trip."""

import torch
import torch.distributed as dist
import torch.nn as nn

from megatron_amd.config import TransformerConfig
from megatron_amd.distributed.fsdp import FullyShardedDataParallel
from megatron_amd.models.gpt import GPTModel

from tests.utils import assert_close, init_single, spawn_dist


def _cfg(**kw):
    d = dict(num_layers=2, hidden_size=32, num_attention_heads=4, vocab_size=64,
             max_position_embeddings=64)
    d.update(kw)
    return TransformerConfig(**d)


def _make_model(seed=11):
    torch.manual_seed(seed)
    return GPTModel(_cfg())


def _fsdp_worker(rank, world, reshard):
    from megatron_amd.parallel import grid as G
    from megatron_amd.parallel.random import model_parallel_seed

    G.destroy_model_parallel()
    G.initialize_model_parallel()
    model_parallel_seed(1234)

    model = _make_model(seed=100 + rank)  # deliberately different per-rank init
    fsdp = FullyShardedDataParallel(model, reshard_after_forward=reshard)
    opt = torch.optim.AdamW(fsdp.shard_parameters(), lr=1e-2)

    # reference: single-process twin loaded from FSDP's full state dict,
    # fed both ranks' data
    ref = _make_model(seed=999)
    ref.load_state_dict(fsdp.state_dict())
    ref_opt = torch.optim.AdamW(ref.parameters(), lr=1e-2)

    for step in range(3):
        g = torch.Generator().manual_seed(7 * step)
        ids_all = torch.randint(0, 64, (2 * world, 16), generator=g)
        labels_all = torch.randint(0, 64, (2 * world, 16), generator=g)
        ids = ids_all[rank * 2 : rank * 2 + 2]
        labels = labels_all[rank * 2 : rank * 2 + 2]

        fsdp.zero_grad_buffer()
        loss = fsdp(input_ids=ids, labels=labels).mean()
        loss.backward()
        opt.step()
        fsdp.update_model_shards()

        ref_opt.zero_grad()
        # mean over the union batch == DP-average of per-rank means
        ref_loss = ref(input_ids=ids_all, labels=labels_all).mean()
        ref_loss.backward()
        ref_opt.step()

        full_loss = loss.detach().clone()
        dist.all_reduce(full_loss)
        full_loss /= world
        assert_close(full_loss, ref_loss.detach(), rtol=1e-4, atol=1e-5,
                     msg=f"step {step}")


def test_fsdp_dp2_matches_single_process():
    spawn_dist(_fsdp_worker, world_size=2, reshard=False)


def test_fsdp_reshard_after_forward():
    spawn_dist(_fsdp_worker, world_size=2, reshard=True)


def _accum_worker(rank, world):
    from megatron_amd.parallel import grid as G
    from megatron_amd.parallel.random import model_parallel_seed

    G.destroy_model_parallel()
    G.initialize_model_parallel()
    model_parallel_seed(1234)
    model = _make_model(seed=100 + rank)
    fsdp = FullyShardedDataParallel(model)
    g = torch.Generator().manual_seed(3)
    ids = torch.randint(0, 64, (4, 16), generator=g)
    labels = torch.randint(0, 64, (4, 16), generator=g)

    # 2 microbatches with accumulation == 1 big microbatch (both pre-scaled)
    fsdp.zero_grad_buffer()
    with fsdp.no_last_microbatch():
        (fsdp(input_ids=ids[:2], labels=labels[:2]).mean() / 2).backward()
    (fsdp(input_ids=ids[2:], labels=labels[2:]).mean() / 2).backward()
    accum = [u.grad_shard.clone() for u in fsdp.units]

    fsdp.zero_grad_buffer()
    fsdp(input_ids=ids, labels=labels).mean().backward()
    for a, u in zip(accum, fsdp.units):
        assert_close(a, u.grad_shard, rtol=1e-4, atol=1e-5)

    n = fsdp.clip_grad_norm(1e9)
    assert torch.isfinite(n)


def test_fsdp_grad_accumulation():
    spawn_dist(_accum_worker, world_size=2)


def _sd_worker(rank, world):
    from megatron_amd.parallel import grid as G
    from megatron_amd.parallel.random import model_parallel_seed

    G.destroy_model_parallel()
    G.initialize_model_parallel()
    model_parallel_seed(1234)
    model = _make_model(seed=100 + rank)
    fsdp = FullyShardedDataParallel(model, reshard_after_forward=True)
    sd = fsdp.state_dict()
    # full state dict identical on every rank (= rank-0 init)
    for k, v in sorted(sd.items()):
        ref = v.clone()
        dist.broadcast(ref, src=0)
        assert torch.equal(ref, v), k
    # outside forward, params are stubs (sharded memory)
    for u in fsdp.units:
        if u.reshard_after_forward:
            assert all(p.data.numel() == 0 for p in u.params)


def test_fsdp_state_dict():
    spawn_dist(_sd_worker, world_size=2)


def _prefetch_case(rank, world):
    """AG-prefetch: forward of unit i must launch unit i+1's async gather."""
    import torch.nn as nn

    from megatron_amd.distributed.fsdp import FullyShardedDataParallel, _FSDPUnit
    from megatron_amd.parallel import grid as G

    G.initialize_model_parallel()
    torch.manual_seed(7)
    model = nn.Sequential(nn.Linear(32, 32), nn.Linear(32, 32), nn.Linear(32, 32))
    async_calls = []
    orig = _FSDPUnit.unshard

    def spy(self, async_op=False):
        if async_op and self._flat is None:
            async_calls.append(self.name)
        return orig(self, async_op)

    _FSDPUnit.unshard = spy
    try:
        fsdp = FullyShardedDataParallel(model, unit_classes=(nn.Linear,),
                                        reshard_after_forward=True)
        x = torch.randn(8, 32)
        fsdp(x).sum().backward()
    finally:
        _FSDPUnit.unshard = orig
    assert len(async_calls) >= 2, async_calls  # fwd prefetches 1,2; bwd prefetches back


def test_fsdp_ag_prefetch_pipeline():
    spawn_dist(_prefetch_case, 2)


def _ep_worker(rank, world):
    """FSDP over a MoE model at EP=2 x world 4 (dp=4, edp=2): dense params
    flat-shard over dp, expert params over edp with dp_cp grad averaging.
    Reference twin: plain per-rank replicas with manual group all-reduces."""
    from megatron_amd.parallel import grid as G
    from megatron_amd.parallel.random import model_parallel_seed

    G.destroy_model_parallel()
    G.initialize_model_parallel(expert_parallel_size=2)

    def build():
        model_parallel_seed(1234)
        torch.manual_seed(50 + G.get_grid().rank_in("ep"))  # per-EP-slice init
        return GPTModel(_cfg(num_experts=4, moe_router_topk=2, moe_ffn_hidden_size=32,
                             expert_parallel_size=2, gradient_accumulation_fusion=False))

    model = build()
    ref = build()

    fsdp = FullyShardedDataParallel(model)
    assert any(u.name.endswith(".experts") for u in fsdp.units)
    opt = torch.optim.AdamW(fsdp.shard_parameters(), lr=1e-2)

    # the FSDP wrapper synced params from each unit-group's rank 0; mirror
    # that on the reference replicas
    grid = G.get_grid()
    dp_group, edp_group = grid.group("dp"), grid.group("expert_dp")
    for n, p in ref.named_parameters():
        if getattr(p, "is_expert_parallel", False):
            dist.broadcast(p.data, src=dist.get_process_group_ranks(edp_group)[0],
                           group=edp_group)
        else:
            dist.broadcast(p.data, src=dist.get_process_group_ranks(dp_group)[0],
                           group=dp_group)
    ref_opt = torch.optim.AdamW(ref.parameters(), lr=1e-2)

    dp_cp = dist.get_world_size(dp_group)
    for step in range(3):
        g = torch.Generator().manual_seed(11 * step)
        ids = torch.randint(0, 64, (world * 2, 16), generator=g)[rank * 2 : rank * 2 + 2]

        fsdp.zero_grad_buffer()
        loss = fsdp(input_ids=ids, labels=ids).mean()
        loss.backward()
        opt.step()
        fsdp.update_model_shards()

        ref_opt.zero_grad()
        ref_loss = ref(input_ids=ids, labels=ids).mean()
        ref_loss.backward()
        for p in ref.parameters():
            if p.grad is None:
                p.grad = torch.zeros_like(p)
            if getattr(p, "is_expert_parallel", False):
                dist.all_reduce(p.grad, group=edp_group)
                p.grad /= dp_cp
            else:
                dist.all_reduce(p.grad, group=dp_group)
                p.grad /= dp_cp
        ref_opt.step()
        assert_close(loss.detach(), ref_loss.detach(), rtol=1e-5, atol=1e-6,
                     msg=f"step {step}")

    full = fsdp.state_dict()
    for n, p in ref.named_parameters():
        assert_close(full[n], p.detach(), rtol=1e-4, atol=1e-5, msg=n)


def test_fsdp_expert_parallel_ep2():
    spawn_dist(_ep_worker, world_size=4)
