"""Combined (layer-interleaved) 1F1B: gradient equivalence with plain
whole-model forward/backward over the same microbatches."""

import torch

from megatron_amd.config import TransformerConfig
from megatron_amd.models.gpt import GPTModel
from megatron_amd.pipeline.combined_1f1b import (
    ModelChunkSchedulePlan,
    ScheduleNode,
    combined_1f1b_step,
)

from tests.utils import assert_close, init_single


def test_schedule_node_chain_matches_autograd():
    torch.manual_seed(0)
    lin1 = torch.nn.Linear(8, 8)
    lin2 = torch.nn.Linear(8, 8)
    x = torch.randn(4, 8, requires_grad=True)

    # reference
    out = lin2(torch.relu(lin1(x)))
    loss = out.square().sum()
    loss.backward()
    ref = {n: p.grad.clone() for n, p in [("l1w", lin1.weight), ("l2w", lin2.weight)]}
    ref_x = x.grad.clone()
    lin1.zero_grad(), lin2.zero_grad()

    n1 = ScheduleNode(lambda t: torch.relu(lin1(t)))
    n2 = ScheduleNode(lambda t: lin2(t).square().sum())
    x2 = x.detach().clone().requires_grad_(True)
    h = n1.forward(x2)
    n2.forward(h)
    g = n2.backward(None)
    g = n1.backward(g)
    # input grad is handed back (the pipeline would p2p it upstream)
    assert_close(g, ref_x, rtol=1e-6, atol=1e-7)
    assert_close(lin1.weight.grad, ref["l1w"], rtol=1e-6, atol=1e-7)
    assert_close(lin2.weight.grad, ref["l2w"], rtol=1e-6, atol=1e-7)


def test_combined_1f1b_matches_plain_two_microbatches():
    init_single()
    cfg = TransformerConfig(
        num_layers=3, hidden_size=32, num_attention_heads=4, num_query_groups=2,
        vocab_size=64, ffn_hidden_size=48, gradient_accumulation_fusion=False)
    torch.manual_seed(9)
    model = GPTModel(cfg)
    mb1 = torch.randint(0, 64, (2, 8))
    mb2 = torch.randint(0, 64, (2, 8))

    # reference: two plain fwd+bwd, accumulated grads
    for mb in (mb1, mb2):
        out = model(mb, position_ids=None, attention_mask=None)
        out.float().square().mean().backward()
    ref_grads = {n: p.grad.clone() for n, p in model.named_parameters() if p.grad is not None}
    model.zero_grad(set_to_none=True)

    # combined: fwd(mb2) interleaved with bwd(mb1)
    def loss_fn(logits):
        return logits.float().square().mean()

    freqs = model._rotary_freqs(8, mb1.device)

    plan1 = ModelChunkSchedulePlan.from_gpt(model, rotary_freqs=freqs, loss_fn=loss_fn)
    plan2 = ModelChunkSchedulePlan.from_gpt(model, rotary_freqs=freqs, loss_fn=loss_fn)

    combined_1f1b_step(plan1, mb1)                     # warmup fwd mb1
    combined_1f1b_step(plan2, mb2, bwd_plan=plan1)     # steady: fwd mb2 + bwd mb1
    # cooldown: drain mb2's backward
    n = len(plan2)
    g = None
    for k in range(n):
        g = plan2.backward_node(n - 1 - k, g)

    for name, p in model.named_parameters():
        if name in ref_grads:
            assert torch.allclose(p.grad, ref_grads[name], atol=1e-5), name


def test_combined_schedule_matches_standard_training():
    """The combined pp=1 schedule (overlap_moe_expert_parallel_comm) must
    produce the same losses and updated weights as the standard schedule."""
    import copy

    from megatron_amd.config import DDPConfig, OptimizerConfig, TransformerConfig
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.parallel.random import model_parallel_seed
    from megatron_amd.training.training import setup_model_and_optimizer, train_step
    from tests.utils import init_single

    def run(combined):
        init_single()
        model_parallel_seed(21)
        cfg = TransformerConfig(
            num_layers=2, hidden_size=64, num_attention_heads=4, num_query_groups=2,
            vocab_size=96, ffn_hidden_size=96, num_experts=4, moe_router_topk=2,
            moe_ffn_hidden_size=64, moe_aux_loss_coeff=0.01,
            overlap_moe_expert_parallel_comm=combined,
            gradient_accumulation_fusion=False,
        )
        opt_cfg = OptimizerConfig(lr=1e-3, clip_grad=1.0)
        chunks, opt = setup_model_and_optimizer(_seeded_provider, cfg, opt_cfg,
                                                DDPConfig(grad_reduce_in_fp32=True))
        g = torch.Generator().manual_seed(9)
        batches = []
        for _ in range(4):
            t = torch.randint(0, 96, (2, 17), generator=g)
            batches.append({"tokens": t[:, :-1], "labels": t[:, 1:]})

        def fwd(it, model):
            batch = next(it)

            def loss_func(loss_sb):
                s = loss_sb.sum()
                return s, torch.tensor(loss_sb.numel()), {"loss_sum": s.detach()}

            return model(batch["tokens"], labels=batch["labels"]), loss_func

        losses = []
        for s in range(2):
            it = iter(batches[s * 2 : (s + 1) * 2])
            r = train_step(fwd, [it], chunks, opt, cfg, 2, 16, 2)
            losses.append(r["lm_loss"])
        params = {n: p.detach().clone() for n, p in chunks[0].module.named_parameters()}
        return losses, params

    def _seeded_provider(config, pre_process=True, post_process=True, vp_stage=None):
        torch.manual_seed(42)
        return GPTModel(config, pre_process=pre_process, post_process=post_process)

    globals()["_seeded_provider"] = _seeded_provider
    l_std, p_std = run(False)
    l_cmb, p_cmb = run(True)
    for a, b in zip(l_std, l_cmb):
        assert abs(a - b) < 1e-5, (l_std, l_cmb)
    for n in p_std:
        torch.testing.assert_close(p_cmb[n], p_std[n], rtol=1e-5, atol=1e-6)


def _tp2_combined_case(rank, world, combined):
    import json
    import os

    from megatron_amd.config import DDPConfig, OptimizerConfig, TransformerConfig
    from megatron_amd.parallel import grid as G
    from megatron_amd.parallel.random import model_parallel_seed
    from megatron_amd.training.training import setup_model_and_optimizer, train_step

    G.initialize_model_parallel(tensor_parallel_size=2)
    model_parallel_seed(21)
    cfg = TransformerConfig(
        num_layers=2, hidden_size=64, num_attention_heads=4, num_query_groups=2,
        vocab_size=96, ffn_hidden_size=96, num_experts=4, moe_router_topk=2,
        moe_ffn_hidden_size=64, moe_aux_loss_coeff=0.01, tensor_parallel_size=2,
        overlap_moe_expert_parallel_comm=combined, gradient_accumulation_fusion=True)

    def provider(config, pre_process=True, post_process=True, vp_stage=None):
        torch.manual_seed(42)
        from megatron_amd.models.gpt import GPTModel

        return GPTModel(config)

    chunks, opt = setup_model_and_optimizer(provider, cfg, OptimizerConfig(lr=1e-3, clip_grad=1.0),
                                            DDPConfig(grad_reduce_in_fp32=True))
    g = torch.Generator().manual_seed(9)
    batches = []
    for _ in range(4):
        t = torch.randint(0, 96, (2, 17), generator=g)
        batches.append({"tokens": t[:, :-1], "labels": t[:, 1:]})

    def fwd(it, model):
        batch = next(it)

        def loss_func(loss_sb):
            s = loss_sb.sum()
            return s, torch.tensor(loss_sb.numel()), {"loss_sum": s.detach()}

        return model(batch["tokens"], labels=batch["labels"]), loss_func

    losses = []
    for s in range(2):
        r = train_step(fwd, [iter(batches[s * 2:(s + 1) * 2])], chunks, opt, cfg, 2, 16, 2)
        losses.append(r["lm_loss"])
    if rank == 0:
        with open(os.environ["CMB_TEST_OUT"], "w") as f:
            json.dump(losses, f)


def test_tp2_combined_matches_standard(tmp_path, monkeypatch):
    """Combined co-schedule under TP=2 (per-layer all-reduces inside plan
    nodes) trains identically to the standard schedule."""
    import json

    from tests.utils import spawn_dist

    out_s, out_c = tmp_path / "s.json", tmp_path / "c.json"
    monkeypatch.setenv("CMB_TEST_OUT", str(out_s))
    spawn_dist(_tp2_combined_case, 2, False)
    monkeypatch.setenv("CMB_TEST_OUT", str(out_c))
    spawn_dist(_tp2_combined_case, 2, True)
    std, cmb = json.load(open(out_s)), json.load(open(out_c))
    for a, b in zip(std, cmb):
        assert abs(a - b) < 1e-5, (std, cmb)


def test_combined_schedule_with_distributed_optimizer():
    """Combined co-schedule x dist-opt (the Mixtral bench configuration):
    losses and updated weights equal the standard schedule under ZeRO-1."""
    from megatron_amd.config import DDPConfig, OptimizerConfig, TransformerConfig
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.parallel.random import model_parallel_seed
    from megatron_amd.training.training import setup_model_and_optimizer, train_step
    from tests.utils import init_single

    def provider(config, pre_process=True, post_process=True, vp_stage=None):
        torch.manual_seed(42)
        return GPTModel(config)

    def run(combined):
        init_single()
        model_parallel_seed(21)
        cfg = TransformerConfig(
            num_layers=2, hidden_size=64, num_attention_heads=4, num_query_groups=2,
            vocab_size=96, ffn_hidden_size=96, num_experts=4, moe_router_topk=2,
            moe_ffn_hidden_size=64, moe_aux_loss_coeff=0.01,
            overlap_moe_expert_parallel_comm=combined,
            gradient_accumulation_fusion=True)
        opt_cfg = OptimizerConfig(lr=1e-3, clip_grad=1.0, use_distributed_optimizer=True)
        ddp_cfg = DDPConfig(grad_reduce_in_fp32=True, use_distributed_optimizer=True,
                            bucket_size=10_000)
        chunks, opt = setup_model_and_optimizer(provider, cfg, opt_cfg, ddp_cfg)
        g = torch.Generator().manual_seed(9)
        batches = []
        for _ in range(4):
            t = torch.randint(0, 96, (2, 17), generator=g)
            batches.append({"tokens": t[:, :-1], "labels": t[:, 1:]})

        def fwd(it, model):
            batch = next(it)

            def loss_func(loss_sb):
                s = loss_sb.sum()
                return s, torch.tensor(loss_sb.numel()), {"loss_sum": s.detach()}

            return model(batch["tokens"], labels=batch["labels"]), loss_func

        losses = []
        for s in range(2):
            r = train_step(fwd, [iter(batches[s * 2:(s + 1) * 2])], chunks, opt,
                           cfg, 2, 16, 2)
            losses.append(r["lm_loss"])
        params = {n: p.detach().clone() for n, p in chunks[0].module.named_parameters()}
        return losses, params

    l_std, p_std = run(False)
    l_cmb, p_cmb = run(True)
    for a, b in zip(l_std, l_cmb):
        assert abs(a - b) < 1e-5, (l_std, l_cmb)
    for n in p_std:
        torch.testing.assert_close(p_cmb[n], p_std[n], rtol=1e-5, atol=1e-6, msg=n)
