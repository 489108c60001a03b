"""DDP + optimizer correctness on CPU/gloo.

Invariants:
  * DP=2 training == single-process training on the concatenated batch.
  * distributed optimizer (ZeRO-1) == plain mixed-precision optimizer,
    step for step.
"""

import torch
import torch.distributed as dist

from megatron_amd.config import DDPConfig, OptimizerConfig, TransformerConfig
from megatron_amd.models.gpt import GPTModel
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.random import model_parallel_seed
from megatron_amd.training.training import setup_model_and_optimizer, train_step

from tests.utils import assert_close, init_single, spawn_dist


def _tiny_cfg(**kw):
    base = dict(
        num_layers=2, hidden_size=64, num_attention_heads=4, num_query_groups=2,
        vocab_size=96, ffn_hidden_size=128, gradient_accumulation_fusion=True,
    )
    base.update(kw)
    return TransformerConfig(**base)


def _provider(config, pre_process=True, post_process=True, vp_stage=None):
    torch.manual_seed(42)  # same init on every rank
    return GPTModel(config, pre_process=pre_process, post_process=post_process, vp_stage=vp_stage)


def _make_fwd_step(batches):
    it = iter(batches)

    def forward_step(data_iterator, model):
        batch = next(it)

        def loss_func(loss_sb):
            loss_sum = loss_sb.sum()
            ntok = torch.tensor(loss_sb.numel())
            return loss_sum, ntok, {"loss_sum": loss_sum.detach()}

        out = model(batch["tokens"], labels=batch["labels"])
        return out, loss_func

    return forward_step


def _run_steps(model_chunks, optimizer, cfg, batches, n_steps, mbs_per_step):
    losses = []
    for s in range(n_steps):
        step_batches = batches[s * mbs_per_step : (s + 1) * mbs_per_step]
        fwd = _make_fwd_step(step_batches)
        r = train_step(fwd, None, model_chunks, optimizer, cfg, mbs_per_step, 16, 2)
        losses.append(r["lm_loss"])
    return losses


def _gen_batches(n, mbs=2, seq=16, vocab=96, seed=7):
    g = torch.Generator().manual_seed(seed)
    out = []
    for _ in range(n):
        t = torch.randint(0, vocab, (mbs, seq + 1), generator=g)
        out.append({"tokens": t[:, :-1], "labels": t[:, 1:]})
    return out


def _dp2_case(rank, world, use_dist_opt):
    G.initialize_model_parallel()  # pure DP
    model_parallel_seed(1234)
    cfg = _tiny_cfg()
    opt_cfg = OptimizerConfig(lr=1e-3, weight_decay=0.01, clip_grad=1.0,
                              use_distributed_optimizer=use_dist_opt)
    ddp_cfg = DDPConfig(grad_reduce_in_fp32=True, use_distributed_optimizer=use_dist_opt,
                        bucket_size=10_000)
    chunks, opt = setup_model_and_optimizer(_provider, cfg, opt_cfg, ddp_cfg)

    all_batches = _gen_batches(8)  # 4 steps x 2 microbatches(dp-split: 1 each)
    mine = [all_batches[i] for i in range(len(all_batches)) if i % world == rank]
    losses = _run_steps(chunks, opt, cfg, mine, 4, 1)

    t = torch.tensor(losses)
    dist.broadcast(t, src=0)
    assert torch.allclose(t, torch.tensor(losses), atol=1e-6), "ranks disagree on loss"
    # stash for cross-run comparison via file
    import json, os

    if rank == 0:
        with open(os.environ["DP_TEST_OUT"], "w") as f:
            json.dump(losses, f)


def _single_reference(use_dist_opt_shape=False):
    init_single()
    cfg = _tiny_cfg()
    opt_cfg = OptimizerConfig(lr=1e-3, weight_decay=0.01, clip_grad=1.0)
    ddp_cfg = DDPConfig(grad_reduce_in_fp32=True, bucket_size=10_000)
    chunks, opt = setup_model_and_optimizer(_provider, cfg, opt_cfg, ddp_cfg)
    batches = _gen_batches(8)
    return _run_steps(chunks, opt, cfg, batches, 4, 2)


def test_dp2_matches_single(tmp_path, monkeypatch):
    import json

    out = tmp_path / "dp.json"
    monkeypatch.setenv("DP_TEST_OUT", str(out))
    ref = _single_reference()
    spawn_dist(_dp2_case, 2, False)
    dp_losses = json.load(open(out))
    for a, b in zip(ref, dp_losses):
        assert abs(a - b) < 1e-4, (ref, dp_losses)


def test_dp2_dist_opt_matches_single(tmp_path, monkeypatch):
    import json

    out = tmp_path / "dpo.json"
    monkeypatch.setenv("DP_TEST_OUT", str(out))
    ref = _single_reference()
    spawn_dist(_dp2_case, 2, True)
    dp_losses = json.load(open(out))
    for a, b in zip(ref, dp_losses):
        assert abs(a - b) < 1e-4, (ref, dp_losses)


def test_grad_accumulation_equals_big_batch():
    """2 microbatches of 1 == 1 microbatch of 2 (sum-loss / token-weighted)."""
    init_single()
    cfg = _tiny_cfg()
    opt_cfg = OptimizerConfig(lr=1e-3, clip_grad=0.0)
    chunks, opt = setup_model_and_optimizer(_provider, cfg, opt_cfg,
                                            DDPConfig(grad_reduce_in_fp32=True, bucket_size=10_000))
    batches = _gen_batches(2, mbs=1)
    fwd = _make_fwd_step(batches)
    train_step(fwd, None, chunks, opt, cfg, 2, 16, 1)
    g1 = {n: p.main_grad.clone() for n, p in chunks[0].module.named_parameters()}

    init_single()
    chunks2, opt2 = setup_model_and_optimizer(_provider, cfg, opt_cfg,
                                              DDPConfig(grad_reduce_in_fp32=True, bucket_size=10_000))
    big = {
        "tokens": torch.cat([batches[0]["tokens"], batches[1]["tokens"]]),
        "labels": torch.cat([batches[0]["labels"], batches[1]["labels"]]),
    }
    fwd2 = _make_fwd_step([big])
    train_step(fwd2, None, chunks2, opt2, cfg, 1, 16, 2)
    g2 = {n: p.main_grad.clone() for n, p in chunks2[0].module.named_parameters()}
    for n in g1:
        assert_close(g1[n], g2[n], rtol=1e-4, atol=1e-5, msg=n)


def test_rccl_registered_buffers_fallback_cpu():
    """The registered-pool path must degrade to plain allocation without
    CUDA (and DDP training still works)."""
    from megatron_amd.distributed.ddp import DistributedDataParallel
    from megatron_amd.distributed.rccl_allocator import RcclRegisteredPool, registered_comm_pool

    from tests.utils import init_single

    init_single()
    pool = RcclRegisteredPool()
    assert not pool.active  # no CUDA here
    with registered_comm_pool() as p:
        t = torch.zeros(16)
    assert t.sum() == 0

    cfg = _tiny_cfg()
    ddp_cfg = DDPConfig(use_rccl_registered_buffers=True)
    torch.manual_seed(0)
    model = DistributedDataParallel(cfg, ddp_cfg, _provider(cfg))
    x = torch.randint(0, 96, (2, 8))
    out = model(x, position_ids=None, attention_mask=None)
    out.float().square().mean().backward()
    model.finish_grad_sync()
    for p_ in model.parameters():
        assert p_.main_grad is not None


# --- expert-grad scaling + grad-norm group fixes (round-2 ADVICE items) -----


def _expert_scale_case(rank, world):
    """ep=world -> edp=1, dp_cp=world: expert grads must still be averaged
    over dp_cp (reference expert_gradient_scaling_factor), not left unscaled."""
    import torch.nn as nn

    G.initialize_model_parallel(expert_parallel_size=world)

    class Tiny(nn.Module):
        def __init__(self):
            super().__init__()
            self.dense = nn.Parameter(torch.ones(8))
            self.expert = nn.Parameter(torch.ones(8))
            self.expert.is_expert_parallel = True

        def forward(self, x):
            return (x * self.dense).sum() + (x * self.expert).sum()

    from megatron_amd.distributed.ddp import DistributedDataParallel

    model = Tiny()
    ddp = DistributedDataParallel(None, DDPConfig(overlap_grad_reduce=False), model)
    x = torch.ones(8)
    ddp(x).backward()
    ddp.finish_grad_sync()
    # dense: grad 1.0 on each of `world` dp ranks, averaged -> 1.0
    assert_close(model.dense.main_grad, torch.ones(8), rtol=0, atol=1e-6)
    # expert: edp group is size 1 (no collective) but the average denominator
    # must be dp_cp = world -> 1/world
    assert_close(model.expert.main_grad, torch.full((8,), 1.0 / world), rtol=0, atol=1e-6)


def test_expert_grad_scaled_by_dp_cp():
    spawn_dist(_expert_scale_case, 2)


def _cp_grad_norm_case(rank, world):
    """cp=world: grads (already dp_cp-reduced) are replicated over cp; the
    norm reduction must NOT count them cp times."""
    from megatron_amd.optimizer.clip import get_grad_norm

    G.initialize_model_parallel(context_parallel_size=world)
    g = torch.full((4,), 2.0)  # identical on every cp rank, ||g|| = 4.0
    total = get_grad_norm([g])
    assert abs(float(total) - 4.0) < 1e-5, f"cp-inflated norm: {float(total)}"


def test_grad_norm_not_inflated_by_cp():
    spawn_dist(_cp_grad_norm_case, 2)


def _ep_grad_norm_case(rank, world):
    """ep=world: each rank's local experts contribute; the norm must include
    every rank's expert span (reduced over etp x ep x pp)."""
    from megatron_amd.optimizer.clip import get_grad_norm

    G.initialize_model_parallel(expert_parallel_size=world)
    dense = torch.full((4,), 2.0)       # replicated -> ||.|| = 4
    expert = torch.full((4,), float(rank + 1))  # rank0: 2.0, rank1: 4.0 norms
    total = get_grad_norm([dense], expert_grads=[expert])
    # total^2 = 16 + (4*1 + 4*4) = 36 -> 6.0
    assert abs(float(total) - 6.0) < 1e-5, f"expert norm wrong: {float(total)}"


def test_grad_norm_includes_all_ep_ranks():
    spawn_dist(_ep_grad_norm_case, 2)


def test_check_for_nan_in_grad_names_the_bucket():
    """check_for_nan_in_grad fails fast before the reduce and names params
    in the offending bucket (reference DistributedDataParallelConfig)."""
    import pytest

    from megatron_amd.config import DDPConfig, TransformerConfig
    from megatron_amd.distributed.ddp import DistributedDataParallel
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.parallel.random import model_parallel_seed
    from tests.utils import init_single

    init_single()
    model_parallel_seed(3)
    cfg = TransformerConfig(num_layers=2, hidden_size=32, num_attention_heads=4,
                            num_query_groups=4, vocab_size=64, ffn_hidden_size=48,
                            gradient_accumulation_fusion=True)
    m = DistributedDataParallel(cfg,
                                DDPConfig(check_for_nan_in_grad=True,
                                          overlap_grad_reduce=False,
                                          bucket_size=5_000), GPTModel(cfg))
    tokens = torch.randint(0, 64, (2, 8))
    m.zero_grad_buffer()
    m(tokens, labels=tokens).sum().backward()
    m.finish_grad_sync()  # clean grads pass

    m.zero_grad_buffer()
    m(tokens, labels=tokens).sum().backward()
    p = m.module.output_layer.weight
    p.main_grad.view(-1)[0] = float("nan")
    with pytest.raises(RuntimeError, match="NaN/Inf grad .*bucket"):
        m.finish_grad_sync()


def _dp4_two_instance_case(rank, world):
    """dist-opt with 2 optimizer instances over dp=4: shards live on 2-rank
    sub-groups, grads two-level reduced (RS intra + AR across instances)."""
    import json, os

    G.initialize_model_parallel()
    model_parallel_seed(1234)
    cfg = _tiny_cfg()
    opt_cfg = OptimizerConfig(lr=1e-3, weight_decay=0.01, clip_grad=1.0,
                              use_distributed_optimizer=True)
    ddp_cfg = DDPConfig(grad_reduce_in_fp32=True, use_distributed_optimizer=True,
                        bucket_size=10_000, num_distributed_optimizer_instances=2)
    chunks, opt = setup_model_and_optimizer(_provider, cfg, opt_cfg, ddp_cfg)
    buf = chunks[0].buffers[0]
    assert buf.inter_group is not None
    assert dist.get_world_size(group=buf.dp_group) == 2  # intra-instance

    all_batches = _gen_batches(8)
    mine = [all_batches[i] for i in range(len(all_batches)) if i % world == rank]
    losses = _run_steps(chunks, opt, cfg, mine, 2, 1)
    t = torch.tensor(losses)
    dist.broadcast(t, src=0)
    assert torch.allclose(t, torch.tensor(losses), atol=1e-6), "ranks disagree"
    if rank == 0:
        with open(os.environ["DP_TEST_OUT"], "w") as f:
            json.dump(losses, f)


def test_dp4_dist_opt_two_instances_matches_single(tmp_path, monkeypatch):
    import json

    out = tmp_path / "dp4i2.json"
    monkeypatch.setenv("DP_TEST_OUT", str(out))
    init_single()
    cfg = _tiny_cfg()
    opt_cfg = OptimizerConfig(lr=1e-3, weight_decay=0.01, clip_grad=1.0)
    ddp_cfg = DDPConfig(grad_reduce_in_fp32=True, bucket_size=10_000)
    chunks, opt = setup_model_and_optimizer(_provider, cfg, opt_cfg, ddp_cfg)
    ref = _run_steps(chunks, opt, cfg, _gen_batches(8), 2, 4)
    spawn_dist(_dp4_two_instance_case, 4)
    got = json.load(open(out))
    for a, b in zip(ref, got):
        assert abs(a - b) < 1e-4, (ref, got)


def _dp4_instance_resume_case(rank, world, ckdir):
    """Save at 1 optimizer instance, resume at 2: the sharded optimizer
    atlas reshards across instance layouts."""
    import json, os

    from megatron_amd.checkpoint.checkpointing import load_checkpoint, save_checkpoint

    G.initialize_model_parallel()
    model_parallel_seed(77)
    cfg = _tiny_cfg()
    opt_cfg = OptimizerConfig(lr=1e-3, weight_decay=0.01, clip_grad=1.0,
                              use_distributed_optimizer=True)
    mk = lambda inst: DDPConfig(grad_reduce_in_fp32=True, use_distributed_optimizer=True,
                                bucket_size=10_000,
                                num_distributed_optimizer_instances=inst)
    all_batches = _gen_batches(8)
    mine = [all_batches[i] for i in range(len(all_batches)) if i % world == rank]

    # reference: 1 instance straight through
    chunks, opt = setup_model_and_optimizer(_provider, cfg, opt_cfg, mk(1))
    ref = _run_steps(chunks, opt, cfg, mine, 2, 1)

    # save at 1 instance after step 1
    G.destroy_model_parallel(); G.initialize_model_parallel()
    model_parallel_seed(77)
    chunks, opt = setup_model_and_optimizer(_provider, cfg, opt_cfg, mk(1))
    first = _run_steps(chunks, opt, cfg, mine[:1], 1, 1)
    save_checkpoint(ckdir, chunks, opt, 1)

    # resume at 2 instances
    G.destroy_model_parallel(); G.initialize_model_parallel()
    model_parallel_seed(123)
    chunks2, opt2 = setup_model_and_optimizer(_provider, cfg, opt_cfg, mk(2))
    it = load_checkpoint(ckdir, chunks2, opt2)
    assert it == 1
    second = _run_steps(chunks2, opt2, cfg, mine[1:], 1, 1)
    for a, b in zip(ref, first + second):
        assert abs(a - b) < 1e-5, (rank, ref, first + second)


def test_dp4_dist_opt_instance_reshard_resume(tmp_path):
    spawn_dist(_dp4_instance_resume_case, 4, str(tmp_path / "ck"))
