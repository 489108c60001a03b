"""Distillation loss and quantization recipe selection (CPU)."""

import torch
import torch.distributed as dist

from megatron_amd.config import TransformerConfig
from megatron_amd.models.gpt import GPTModel
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.random import model_parallel_seed
from megatron_amd.post_training import (
    DistillationLoss,
    QuantRecipe,
    QuantRecipeConfig,
    resolve_layer_recipes,
)
from megatron_amd.post_training.distillation import soft_cross_entropy_vocab_parallel

from tests.utils import assert_close, init_single, spawn_dist


def test_soft_ce_matches_torch_kl():
    init_single()
    torch.manual_seed(0)
    s = torch.randn(6, 32)
    t = torch.randn(6, 32)
    T = 2.0
    got = soft_cross_entropy_vocab_parallel(s, t, temperature=T)
    # -sum p_t log p_s, T^2-scaled
    log_ps = torch.log_softmax(s / T, dim=-1)
    p_t = torch.softmax(t / T, dim=-1)
    expect = -(p_t * log_ps).sum(dim=-1) * T * T
    assert_close(got, expect, rtol=1e-5, atol=1e-6)


def _tp2_soft_ce(rank, world):
    G.initialize_model_parallel(tensor_parallel_size=2)
    model_parallel_seed(1)
    torch.manual_seed(9)
    s_full = torch.randn(5, 64)
    t_full = torch.randn(5, 64)
    dist.broadcast(s_full, src=0)
    dist.broadcast(t_full, src=0)
    shard = slice(rank * 32, (rank + 1) * 32)
    got = soft_cross_entropy_vocab_parallel(s_full[:, shard], t_full[:, shard], temperature=1.5)
    log_ps = torch.log_softmax(s_full / 1.5, dim=-1)
    p_t = torch.softmax(t_full / 1.5, dim=-1)
    expect = -(p_t * log_ps).sum(dim=-1) * 1.5 * 1.5
    assert torch.allclose(got, expect, atol=1e-5), (got, expect)


def test_soft_ce_tp2_matches_full():
    spawn_dist(_tp2_soft_ce, 2)


def test_distillation_loss_combines_and_trains():
    init_single()
    torch.manual_seed(3)
    student = torch.randn(4, 8, 32, requires_grad=True)
    teacher = torch.randn(4, 8, 32)
    lm = torch.tensor(2.0)
    loss_fn = DistillationLoss(temperature=2.0, alpha=0.5)
    loss = loss_fn(student, teacher, lm_loss=lm)
    assert loss.requires_grad
    loss.backward()
    assert student.grad is not None and torch.isfinite(student.grad).all()
    # alpha=0 -> pure LM loss
    assert float(DistillationLoss(alpha=0.0)(student.detach(), teacher, lm_loss=lm)) == 2.0
    # identical distributions -> KD part equals teacher entropy (minimal)
    same = DistillationLoss(alpha=1.0)(teacher, teacher)
    perturbed = DistillationLoss(alpha=1.0)(teacher + torch.randn_like(teacher), teacher)
    assert float(same) < float(perturbed)


def test_distillation_hidden_projection():
    loss_fn = DistillationLoss(alpha=1.0, hidden_beta=0.1, student_hidden=16, teacher_hidden=32)
    s_logits = torch.randn(2, 4, 10)
    t_logits = torch.randn(2, 4, 10)
    s_h = torch.randn(2, 4, 16)
    t_h = torch.randn(2, 4, 32)
    base = loss_fn(s_logits, t_logits)
    with_h = loss_fn(s_logits, t_logits, student_hidden=s_h, teacher_hidden=t_h)
    assert float(with_h) != float(base)


def test_quant_recipe_resolution():
    init_single()
    cfg = TransformerConfig(
        num_layers=2, hidden_size=64, num_attention_heads=4, num_query_groups=2,
        vocab_size=96, ffn_hidden_size=128, gradient_accumulation_fusion=False,
    )
    torch.manual_seed(0)
    model = GPTModel(cfg)
    qcfg = QuantRecipeConfig.from_dict({
        "rules": [
            {"match": "*output_layer*", "name": "bf16"},
            {"match": "decoder.layers.0.*", "name": "bf16"},
            {"match": "decoder.layers.*", "name": "fp8", "fmt": "hybrid"},
        ],
        "default": {"name": "bf16"},
    })
    resolved = resolve_layer_recipes(model, qcfg)
    assert any(r.is_quantized for r in resolved.values())
    for name, r in resolved.items():
        if name.startswith("decoder.layers.0.") or "output_layer" in name:
            assert not r.is_quantized, name
        elif name.startswith("decoder.layers.1."):
            assert r.is_quantized, name
    # recipes are pinned on modules
    mods = dict(model.named_modules())
    first = [m for n, m in mods.items() if n.startswith("decoder.layers.1") and hasattr(m, "quant_recipe")]
    assert first and all(m.quant_recipe.name == "fp8" for m in first)


# --- PTQ calibration / fake-quant / export flow -----------------------------


def _ptq_model_and_cfg():
    from megatron_amd.config import TransformerConfig
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.post_training.quant_config import QuantRecipeConfig
    from tests.utils import init_single

    init_single()
    torch.manual_seed(0)
    cfg = TransformerConfig(num_layers=2, hidden_size=32, num_attention_heads=4,
                            vocab_size=64, ffn_hidden_size=48,
                            gradient_accumulation_fusion=False)
    m = GPTModel(cfg)
    qc = QuantRecipeConfig.from_dict({
        "rules": [{"match": "decoder.layers.*.mlp.*", "name": "fp8"},
                  {"match": "decoder.layers.*.self_attention.*", "name": "fp8"}],
        "default": {"name": "bf16"},
    })
    return m, qc


def test_ptq_calibration_collects_amax():
    from megatron_amd.post_training.ptq import CalibrationCollector

    m, qc = _ptq_model_and_cfg()
    toks = torch.randint(0, 64, (2, 8))
    with CalibrationCollector(m, qc) as cal:
        for _ in range(3):
            m(toks)
    assert cal.act_amax, "no activations collected"
    for name, amax in cal.act_amax.items():
        assert "mlp" in name or "self_attention" in name
        assert float(amax) > 0 and cal.samples[name] == 3
    # embeddings / output excluded by the pattern rules
    assert not any("output_layer" in n for n in cal.act_amax)


def test_ptq_fake_quant_bounds_error_and_keeps_quality():
    from megatron_amd.post_training.ptq import quantize_model_weights

    m, qc = _ptq_model_and_cfg()
    toks = torch.randint(0, 64, (2, 8))
    labels = torch.randint(0, 64, (2, 8))
    with torch.no_grad():
        base = m(toks, labels=labels).sum()
        w_before = m.decoder.layers[0].mlp.linear_fc1.weight.detach().clone()
        scales = quantize_model_weights(m, qc, mode="int8")
        w_after = m.decoder.layers[0].mlp.linear_fc1.weight.detach()
        quant = m(toks, labels=labels).sum()
    assert scales
    # weights changed but by bounded quantization error
    assert not torch.equal(w_before, w_after)
    rel = (w_after - w_before).abs().max() / w_before.abs().max()
    assert float(rel) < 0.02
    assert abs(float(quant - base)) / abs(float(base)) < 0.05


def test_ptq_export_int8_payloads():
    from megatron_amd.post_training.ptq import export_quantized_state_dict

    m, qc = _ptq_model_and_cfg()
    sd = export_quantized_state_dict(m, qc, mode="int8")
    qw = [k for k, v in sd.items() if v.dtype == torch.int8]
    assert qw and all(k + "_scale" in sd for k in qw)
    # unmatched tensors exported untouched
    assert any("embedding" in k and sd[k].dtype != torch.int8 for k in sd)
