"""Pipeline-parallel schedule correctness on CPU/gloo.

Invariant: non-interleaved and interleaved 1F1B produce
the same losses as the single-process run for the same global batch and
deterministic weights.
"""

import json
import os
import zlib

import torch

from megatron_amd.config import DDPConfig, OptimizerConfig, TransformerConfig
from megatron_amd.models.gpt import GPTModel
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.random import model_parallel_seed
from megatron_amd.training.training import setup_model_and_optimizer, train_step

from tests.utils import init_single, spawn_dist

N_LAYERS = 4
VOCAB = 96
SEQ = 16


def _cfg(pp=1, vpp=None):
    return TransformerConfig(
        num_layers=N_LAYERS, hidden_size=64, num_attention_heads=4, num_query_groups=2,
        vocab_size=VOCAB, ffn_hidden_size=128, pipeline_parallel_size=pp,
        virtual_pipeline_parallel_size=vpp, gradient_accumulation_fusion=True,
    )


def _fill_deterministic(model):
    """Weights keyed by (global layer number, param name) so every PP layout
    builds identical values."""
    core = model.module if hasattr(model, "module") else model
    def fill(t, key):
        g = torch.Generator().manual_seed(zlib.crc32(key.encode()) % (2**31))
        with torch.no_grad():
            t.copy_(torch.randn(t.shape, generator=g) * 0.02)

    if core.pre_process:
        fill(core.embedding.weight, "embedding")
    for layer in core.decoder.layers:
        ln = layer.layer_number
        for name, p in layer.named_parameters():
            fill(p, f"layer{ln}.{name}")
    if core.post_process:
        fill(core.decoder.final_layernorm.weight, "final_ln")
        core.decoder.final_layernorm.weight.data.add_(1.0)
        fill(core.output_layer.weight, "output")


def _gen_batches(n, mbs=2, seq=SEQ, vocab=VOCAB, seed=7):
    g = torch.Generator().manual_seed(seed)
    out = []
    for _ in range(n):
        t = torch.randint(0, vocab, (mbs, seq + 1), generator=g)
        out.append({"tokens": t[:, :-1], "labels": t[:, 1:]})
    return out


def forward_step(data_iterator, model):
    """Uses the PASSED iterator (per-chunk for interleaved)."""
    batch = next(data_iterator)

    def loss_func(loss_sb):
        s = loss_sb.sum()
        return s, torch.tensor(loss_sb.numel()), {"loss_sum": s.detach()}

    out = model(batch["tokens"], labels=batch["labels"])
    return out, loss_func


def _provider(config, pre_process=True, post_process=True, vp_stage=None):
    torch.manual_seed(42)
    m = GPTModel(config, pre_process=pre_process, post_process=post_process, vp_stage=vp_stage)
    return m


def _run(cfg, n_steps, microbatches_per_step, world_batches):
    opt_cfg = OptimizerConfig(lr=1e-3, clip_grad=1.0)
    ddp_cfg = DDPConfig(grad_reduce_in_fp32=True, bucket_size=10_000)
    chunks, opt = setup_model_and_optimizer(_provider, cfg, opt_cfg, ddp_cfg)
    for c in chunks:
        _fill_deterministic(c)
        # refresh fp32 main params after overwriting weights
    for o in opt.chained_optimizers:
        if hasattr(o, "reload_model_params"):
            o.reload_model_params()
    losses = []
    for s in range(n_steps):
        step_batches = world_batches[s * microbatches_per_step : (s + 1) * microbatches_per_step]
        its = [iter(step_batches) for _ in chunks]  # one iterator per virtual chunk
        r = train_step(forward_step, its, chunks, opt, cfg, microbatches_per_step, SEQ, 2)
        losses.append(r["lm_loss"])
    return losses


def _single_reference():
    init_single()
    return _run(_cfg(), 2, 4, _gen_batches(8))


def _pp2_case(rank, world, vpp):
    G.initialize_model_parallel(pipeline_parallel_size=world, virtual_pipeline_parallel_size=vpp)
    model_parallel_seed(1234)
    cfg = _cfg(pp=world, vpp=vpp)
    losses = _run(cfg, 2, 4, _gen_batches(8))
    if G.get_grid().is_pipeline_last_stage(ignore_virtual=True):
        with open(os.environ["PP_TEST_OUT"], "w") as f:
            json.dump(losses, f)


def test_pp2_1f1b_matches_single(tmp_path, monkeypatch):
    out = tmp_path / "pp.json"
    monkeypatch.setenv("PP_TEST_OUT", str(out))
    ref = _single_reference()
    spawn_dist(_pp2_case, 2, None)
    pp_losses = json.load(open(out))
    for a, b in zip(ref, pp_losses):
        assert abs(a - b) < 2e-4, (ref, pp_losses)


def test_pp2_interleaved_matches_single(tmp_path, monkeypatch):
    out = tmp_path / "ppi.json"
    monkeypatch.setenv("PP_TEST_OUT", str(out))
    ref = _single_reference()
    spawn_dist(_pp2_case, 2, 2)
    pp_losses = json.load(open(out))
    for a, b in zip(ref, pp_losses):
        assert abs(a - b) < 2e-4, (ref, pp_losses)


def test_schedule_table():
    from megatron_amd.pipeline.pipelined import get_schedule_table

    t = get_schedule_table(4, 2, 2)
    assert t == [(0, 0), (1, 0), (0, 1), (1, 1), (2, 0), (3, 0), (2, 1), (3, 1)]


def _varlen_p2p_case(rank, world):
    from megatron_amd.pipeline.p2p import P2PCommunicator
    from megatron_amd.config import TransformerConfig

    G.initialize_model_parallel(pipeline_parallel_size=2)
    cfg = TransformerConfig(num_layers=2, hidden_size=8, num_attention_heads=2,
                            vocab_size=32, pipeline_parallel_size=2,
                            variable_seq_lengths=True, pipeline_dtype=torch.float32,
                            gradient_accumulation_fusion=False)
    comm = P2PCommunicator(cfg, seq_length=16, micro_batch_size=2)
    # stage 0 sends two different-length activations; stage 1 receives with
    # exact shapes via the pre-exchange
    for s in (5, 11):
        if rank == 0:
            payload = torch.full((s, 2, 8), float(s))
            comm.send_forward(payload, is_last_stage=False)
        else:
            t = comm.recv_forward(is_first_stage=False)
            assert t.shape == (s, 2, 8), t.shape
            assert torch.all(t == float(s))
    # and backward direction
    for s in (3, 7):
        if rank == 1:
            comm.send_backward(torch.full((s, 2, 8), -float(s)), is_first_stage=False)
        else:
            g = comm.recv_backward(is_last_stage=False)
            assert g.shape == (s, 2, 8)
            assert torch.all(g == -float(s))


def test_varlen_p2p_shape_exchange():
    spawn_dist(_varlen_p2p_case, 2)


def _tp2pp2_case(rank, world, ckpt_dir):
    from megatron_amd.checkpoint.checkpointing import load_checkpoint

    G.initialize_model_parallel(tensor_parallel_size=2, pipeline_parallel_size=2)
    model_parallel_seed(1234)
    cfg = _cfg(pp=2)
    cfg = cfg.replace(tensor_parallel_size=2)
    opt_cfg = OptimizerConfig(lr=1e-3, clip_grad=1.0)
    ddp_cfg = DDPConfig(grad_reduce_in_fp32=True, bucket_size=10_000)
    chunks, opt = setup_model_and_optimizer(_provider, cfg, opt_cfg, ddp_cfg)
    load_checkpoint(ckpt_dir, chunks, opt, load_rng=False)
    batches = _gen_batches(4)
    its = [iter(batches) for _ in chunks]
    r = train_step(forward_step, its, chunks, opt, cfg, 4, SEQ, 2)
    grid = G.get_grid()
    if grid.is_pipeline_last_stage(ignore_virtual=True) and grid.rank_in("tp") == 0:
        with open(os.environ["PP_TEST_OUT"], "w") as f:
            json.dump([r["lm_loss"]], f)


def test_tp2_pp2_composition_matches_single(tmp_path, monkeypatch):
    """4-rank composition (TP=2 x PP=2) resharded from a single-process
    checkpoint reproduces the single-process first-step loss."""
    from megatron_amd.checkpoint.checkpointing import save_checkpoint

    out = tmp_path / "tp2pp2.json"
    ckpt = str(tmp_path / "ckpt")
    monkeypatch.setenv("PP_TEST_OUT", str(out))

    init_single()
    cfg = _cfg()
    opt_cfg = OptimizerConfig(lr=1e-3, clip_grad=1.0)
    ddp_cfg = DDPConfig(grad_reduce_in_fp32=True, bucket_size=10_000)
    chunks, opt = setup_model_and_optimizer(_provider, cfg, opt_cfg, ddp_cfg)
    for c in chunks:
        _fill_deterministic(c)
    for o in opt.chained_optimizers:
        if hasattr(o, "reload_model_params"):
            o.reload_model_params()
    save_checkpoint(ckpt, chunks, opt, iteration=0)
    batches = _gen_batches(4)
    r = train_step(forward_step, [iter(batches)], chunks, opt, cfg, 4, SEQ, 2)
    ref_loss = r["lm_loss"]

    spawn_dist(_tp2pp2_case, 4, ckpt)
    got = json.load(open(out))[0]
    assert abs(got - ref_loss) < 5e-4, (got, ref_loss)


def _pp2_ragged_case(rank, world, vpp):
    G.initialize_model_parallel(pipeline_parallel_size=world, virtual_pipeline_parallel_size=vpp)
    model_parallel_seed(1234)
    cfg = _cfg(pp=world, vpp=vpp)
    losses = _run(cfg, 2, 3, _gen_batches(6))  # 3 microbatches % pp=2 != 0
    if G.get_grid().is_pipeline_last_stage(ignore_virtual=True):
        with open(os.environ["PP_TEST_OUT"], "w") as f:
            json.dump(losses, f)


def test_pp2_interleaved_ragged_microbatches(tmp_path, monkeypatch):
    """num_microbatches % pp != 0 (reference schedules.py:959 clamping)."""
    out = tmp_path / "ppr.json"
    monkeypatch.setenv("PP_TEST_OUT", str(out))
    init_single()
    ref = _run(_cfg(), 2, 3, _gen_batches(6))
    spawn_dist(_pp2_ragged_case, 2, 2)
    pp_losses = json.load(open(out))
    for a, b in zip(ref, pp_losses):
        assert abs(a - b) < 2e-4, (ref, pp_losses)


def test_pp2_noninterleaved_ragged_microbatches(tmp_path, monkeypatch):
    out = tmp_path / "ppr2.json"
    monkeypatch.setenv("PP_TEST_OUT", str(out))
    init_single()
    ref = _run(_cfg(), 2, 3, _gen_batches(6))
    spawn_dist(_pp2_ragged_case, 2, None)
    pp_losses = json.load(open(out))
    for a, b in zip(ref, pp_losses):
        assert abs(a - b) < 2e-4, (ref, pp_losses)


def _cfg_moe(pp=1, combined=False):
    return TransformerConfig(
        num_layers=N_LAYERS, hidden_size=64, num_attention_heads=4, num_query_groups=2,
        vocab_size=VOCAB, ffn_hidden_size=128, num_experts=4, moe_router_topk=2,
        moe_ffn_hidden_size=64, moe_aux_loss_coeff=0.01, pipeline_parallel_size=pp,
        overlap_moe_expert_parallel_comm=combined, gradient_accumulation_fusion=True,
    )


def _pp2_combined_case(rank, world):
    G.initialize_model_parallel(pipeline_parallel_size=world)
    model_parallel_seed(1234)
    losses = _run(_cfg_moe(pp=world, combined=True), 2, 4, _gen_batches(8))
    if G.get_grid().is_pipeline_last_stage(ignore_virtual=True):
        with open(os.environ["PP_TEST_OUT"], "w") as f:
            json.dump(losses, f)


def test_pp2_combined_1f1b_matches_single(tmp_path, monkeypatch):
    """MoE model under the combined (layer-interleaved fwd/bwd) pp=2 schedule
    trains identically to a single-rank standard run (reference
    combined_1f1b.py steady-state co-schedule)."""
    out = tmp_path / "ppc.json"
    monkeypatch.setenv("PP_TEST_OUT", str(out))
    init_single()
    ref = _run(_cfg_moe(), 2, 4, _gen_batches(8))
    spawn_dist(_pp2_combined_case, 2)
    pp_losses = json.load(open(out))
    for a, b in zip(ref, pp_losses):
        assert abs(a - b) < 2e-4, (ref, pp_losses)


def _pp2ep2_combined_case(rank, world, combined):
    G.initialize_model_parallel(pipeline_parallel_size=2, expert_parallel_size=2)
    model_parallel_seed(1234)
    cfg = _cfg_moe(pp=2, combined=combined)
    cfg = cfg.replace(expert_parallel_size=2)
    losses = _run(cfg, 2, 4, _gen_batches(8))
    grid = G.get_grid()
    if grid.is_pipeline_last_stage(ignore_virtual=True) and grid.rank_in("dp") == 0:
        with open(os.environ["PP_TEST_OUT"], "w") as f:
            json.dump(losses, f)


def test_pp2_ep2_combined_matches_standard(tmp_path, monkeypatch):
    """The Mixtral-style layout (PP=2 x EP=2, world 4): the combined
    co-schedule must train identically to the standard 1F1B schedule."""
    out_s = tmp_path / "std.json"
    out_c = tmp_path / "cmb.json"
    monkeypatch.setenv("PP_TEST_OUT", str(out_s))
    spawn_dist(_pp2ep2_combined_case, 4, False)
    monkeypatch.setenv("PP_TEST_OUT", str(out_c))
    spawn_dist(_pp2ep2_combined_case, 4, True)
    std = json.load(open(out_s))
    cmb = json.load(open(out_c))
    for a, b in zip(std, cmb):
        assert abs(a - b) < 2e-4, (std, cmb)


def _pp4_case(rank, world, vpp, combined=False):
    G.initialize_model_parallel(pipeline_parallel_size=world,
                                virtual_pipeline_parallel_size=vpp)
    model_parallel_seed(1234)
    cfg = _cfg(pp=world, vpp=vpp)
    if combined:
        cfg = cfg.replace(overlap_moe_expert_parallel_comm=True)
    losses = _run(cfg, 2, 4, _gen_batches(8))
    if G.get_grid().is_pipeline_last_stage(ignore_virtual=True):
        with open(os.environ["PP_TEST_OUT"], "w") as f:
            json.dump(losses, f)


def test_pp4_matches_single(tmp_path, monkeypatch):
    """Four-deep pipeline (warmup 3 on stage 0) reproduces the single run —
    the depth the 70B TP4xPP2 and deeper-PP configs rely on."""
    out = tmp_path / "pp4.json"
    monkeypatch.setenv("PP_TEST_OUT", str(out))
    ref = _single_reference()
    spawn_dist(_pp4_case, 4, None)
    got = json.load(open(out))
    for a, b in zip(ref, got):
        assert abs(a - b) < 2e-4, (ref, got)


def test_pp4_combined_matches_single(tmp_path, monkeypatch):
    out = tmp_path / "pp4c.json"
    monkeypatch.setenv("PP_TEST_OUT", str(out))
    ref = _single_reference()
    spawn_dist(_pp4_case, 4, None, True)
    got = json.load(open(out))
    for a, b in zip(ref, got):
        assert abs(a - b) < 2e-4, (ref, got)


def _pp2_uneven_case(rank, world):
    G.initialize_model_parallel(pipeline_parallel_size=world)
    model_parallel_seed(1234)
    cfg = _cfg(pp=world).replace(num_layers_in_first_pipeline_stage=1,
                                 num_layers_in_last_pipeline_stage=3)
    losses = _run(cfg, 2, 4, _gen_batches(8))
    if G.get_grid().is_pipeline_last_stage(ignore_virtual=True):
        with open(os.environ["PP_TEST_OUT"], "w") as f:
            json.dump(losses, f)


def test_pp2_uneven_stage_split_matches_single(tmp_path, monkeypatch):
    """Uneven stage layout (1 layer on the embedding stage, 3 on the loss
    stage — reference --decoder-first/last-pipeline-num-layers) trains
    identically to the single-rank run."""
    out = tmp_path / "ppu.json"
    monkeypatch.setenv("PP_TEST_OUT", str(out))
    ref = _single_reference()
    spawn_dist(_pp2_uneven_case, 2)
    got = json.load(open(out))
    for a, b in zip(ref, got):
        assert abs(a - b) < 2e-4, (ref, got)


def _pp2_combined_fwd_only_case(rank, world):
    """forward_only through the combined pp schedule (eval path) must relay
    activations downstream and produce the same losses as training-forward."""
    from megatron_amd.pipeline.schedules import get_forward_backward_func

    G.initialize_model_parallel(pipeline_parallel_size=world)
    model_parallel_seed(1234)
    cfg = _cfg_moe(pp=world, combined=True)
    opt_cfg = OptimizerConfig(lr=1e-3, clip_grad=1.0)
    ddp_cfg = DDPConfig(grad_reduce_in_fp32=True, bucket_size=10_000)
    chunks, opt = setup_model_and_optimizer(_provider, cfg, opt_cfg, ddp_cfg)
    for c in chunks:
        _fill_deterministic(c)
    fb = get_forward_backward_func(cfg)
    batches = _gen_batches(4)
    losses, ntok = fb(forward_step_func=forward_step, data_iterator=[iter(batches)],
                      model=chunks, num_microbatches=4, seq_length=SEQ,
                      micro_batch_size=2, forward_only=True)
    if G.get_grid().is_pipeline_last_stage(ignore_virtual=True):
        total = sum(float(x["loss_sum"]) for x in losses) / max(int(ntok), 1)
        with open(os.environ["PP_TEST_OUT"], "w") as f:
            json.dump([total], f)


def test_pp2_combined_forward_only(tmp_path, monkeypatch):
    out = tmp_path / "ppfo.json"
    monkeypatch.setenv("PP_TEST_OUT", str(out))
    init_single()
    cfg = _cfg_moe()
    opt_cfg = OptimizerConfig(lr=1e-3, clip_grad=1.0)
    ddp_cfg = DDPConfig(grad_reduce_in_fp32=True, bucket_size=10_000)
    chunks, opt = setup_model_and_optimizer(_provider, cfg, opt_cfg, ddp_cfg)
    for c in chunks:
        _fill_deterministic(c)
    batches = _gen_batches(4)
    total = n = 0.0
    for mb in batches:
        with torch.no_grad():
            loss = chunks[0](mb["tokens"], labels=mb["labels"])
        total += float(loss.sum()); n += loss.numel()
    ref = total / n
    spawn_dist(_pp2_combined_fwd_only_case, 2)
    got = json.load(open(out))[0]
    assert abs(got - ref) < 2e-4, (got, ref)
