"""HF <-> megatron_amd weight conversion tests: a tiny random Llama built
locally via transformers must produce identical logits through both stacks,
and the mapping must round-trip exactly."""

import pytest
import torch

from tests.utils import init_single

transformers = pytest.importorskip("transformers")


def _tiny_llama():
    from transformers import LlamaConfig, LlamaForCausalLM

    cfg = LlamaConfig(
        hidden_size=64, intermediate_size=128, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2, vocab_size=128,
        max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
        tie_word_embeddings=False, attention_bias=False,
    )
    torch.manual_seed(5)
    return LlamaForCausalLM(cfg).eval(), cfg


def test_hf_llama_logits_match():
    from tools.checkpoint.convert_hf import config_from_hf, hf_to_mcore_state_dict

    hf, hf_cfg = _tiny_llama()
    init_single()
    from megatron_amd.models.gpt import GPTModel

    cfg = config_from_hf(hf_cfg)
    sd = hf_to_mcore_state_dict(hf.state_dict(), cfg)
    model = GPTModel(cfg).eval()
    missing, unexpected = model.load_state_dict(sd, strict=False)
    assert not unexpected and not missing

    tokens = torch.randint(0, 128, (2, 16))
    with torch.no_grad():
        ours = model(tokens)  # [s, b, V]
        theirs = hf(tokens).logits  # [b, s, V]
    torch.testing.assert_close(ours.permute(1, 0, 2), theirs, rtol=2e-2, atol=2e-2)


def test_roundtrip_exact():
    from tools.checkpoint.convert_hf import (
        config_from_hf,
        hf_to_mcore_state_dict,
        mcore_to_hf_state_dict,
    )

    hf, hf_cfg = _tiny_llama()
    cfg = config_from_hf(hf_cfg)
    sd = hf_to_mcore_state_dict(hf.state_dict(), cfg)
    back = mcore_to_hf_state_dict(sd, cfg)
    hf_sd = hf.state_dict()
    for k, v in back.items():
        assert torch.equal(v, hf_sd[k]), k


def test_mixtral_round_trip_and_logits():
    """Mixtral-style MoE HF dict <-> our MoE GPTModel: exact round trip and
    logits equality against a manual expert computation path."""
    import torch

    from megatron_amd.config import TransformerConfig
    from megatron_amd.models.gpt import GPTModel
    from tools.checkpoint.convert_hf import hf_to_mcore_state_dict, mcore_to_hf_state_dict

    from tests.utils import init_single

    init_single()
    cfg = TransformerConfig(
        num_layers=2, hidden_size=32, num_attention_heads=4, num_query_groups=2,
        vocab_size=64, ffn_hidden_size=48, num_experts=4, moe_router_topk=2,
        moe_ffn_hidden_size=40, gradient_accumulation_fusion=False)
    torch.manual_seed(0)
    model = GPTModel(cfg)
    sd = {k: v for k, v in model.state_dict().items() if "expert_bias" not in k
          and "local_tokens" not in k}
    hf = mcore_to_hf_state_dict(sd, cfg)
    assert "model.layers.0.block_sparse_moe.gate.weight" in hf
    assert "model.layers.1.block_sparse_moe.experts.3.w2.weight" in hf
    back = hf_to_mcore_state_dict(hf, cfg)
    for k, v in sd.items():
        assert k in back, k
        assert torch.allclose(back[k].to(v.dtype), v, atol=0), k
    # loading the round-tripped dict reproduces identical logits
    model2 = GPTModel(cfg)
    missing, unexpected = model2.load_state_dict(back, strict=False)
    assert not unexpected
    toks = torch.randint(0, 64, (2, 8))
    with torch.no_grad():
        a = model(toks, position_ids=None, attention_mask=None)
        b = model2(toks, position_ids=None, attention_mask=None)
    assert torch.allclose(a, b, atol=1e-6)


def test_export_hf_dir_round_trip(tmp_path):
    """Exported safetensors + config re-import to identical logits."""
    import json

    import torch
    from safetensors.torch import load_file

    from megatron_amd.config import TransformerConfig
    from megatron_amd.models.gpt import GPTModel
    from tools.checkpoint.convert_hf import hf_to_mcore_state_dict
    from tools.checkpoint.export_hf import export_hf_dir

    from tests.utils import init_single

    init_single()
    cfg = TransformerConfig(
        num_layers=2, hidden_size=32, num_attention_heads=4, num_query_groups=2,
        vocab_size=64, ffn_hidden_size=48, gradient_accumulation_fusion=False)
    torch.manual_seed(1)
    model = GPTModel(cfg)
    out = str(tmp_path / "hf")
    hf_cfg = export_hf_dir(model, cfg, out)
    assert hf_cfg["architectures"] == ["LlamaForCausalLM"]
    assert json.load(open(out + "/config.json"))["num_hidden_layers"] == 2

    hf_sd = load_file(out + "/model.safetensors")
    back = hf_to_mcore_state_dict(hf_sd, cfg)
    model2 = GPTModel(cfg)
    model2.load_state_dict(back, strict=False)
    toks = torch.randint(0, 64, (2, 8))
    with torch.no_grad():
        a = model(toks, position_ids=None, attention_mask=None)
        b = model2(toks, position_ids=None, attention_mask=None)
    assert torch.allclose(a, b, atol=1e-6)


def test_qwen2_style_qkv_bias_roundtrip():
    """Qwen2-family: QKV biases fuse into the grouped layout, load into a
    model with add_qkv_bias, and split back exactly."""
    import tools.checkpoint.convert_hf as C

    from megatron_amd.config import TransformerConfig
    from megatron_amd.models.gpt import GPTModel

    init_single()
    cfg = TransformerConfig(num_layers=2, hidden_size=32, num_attention_heads=4,
                            num_query_groups=2, ffn_hidden_size=48, vocab_size=64,
                            add_qkv_bias=True, untie_embeddings_and_output_weights=True,
                            gradient_accumulation_fusion=False)
    d = cfg.kv_channels
    torch.manual_seed(0)
    hf = {"model.embed_tokens.weight": torch.randn(64, 32),
          "model.norm.weight": torch.randn(32), "lm_head.weight": torch.randn(64, 32)}
    for i in range(2):
        p = f"model.layers.{i}."
        hf[p + "self_attn.q_proj.weight"] = torch.randn(4 * d, 32)
        hf[p + "self_attn.k_proj.weight"] = torch.randn(2 * d, 32)
        hf[p + "self_attn.v_proj.weight"] = torch.randn(2 * d, 32)
        hf[p + "self_attn.q_proj.bias"] = torch.randn(4 * d)
        hf[p + "self_attn.k_proj.bias"] = torch.randn(2 * d)
        hf[p + "self_attn.v_proj.bias"] = torch.randn(2 * d)
        hf[p + "self_attn.o_proj.weight"] = torch.randn(32, 4 * d)
        hf[p + "mlp.gate_proj.weight"] = torch.randn(48, 32)
        hf[p + "mlp.up_proj.weight"] = torch.randn(48, 32)
        hf[p + "mlp.down_proj.weight"] = torch.randn(32, 48)
        hf[p + "input_layernorm.weight"] = torch.randn(32)
        hf[p + "post_attention_layernorm.weight"] = torch.randn(32)
    sd = C.hf_to_mcore_state_dict(hf, cfg)
    assert "decoder.layers.0.self_attention.linear_qkv.bias" in sd
    model = GPTModel(cfg)
    missing, unexpected = model.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rotary" in m or "freqs" in m for m in missing), missing
    back = C.mcore_to_hf_state_dict(sd, cfg)
    for k in hf:
        if "bias" in k:
            torch.testing.assert_close(back[k], hf[k])


def test_deepseek_mla_moe_round_trip():
    """MLA + routed-MoE + shared-expert model exports to DeepSeek-style HF
    names and converts back bit-identically; converted dict loads strict."""
    from tools.checkpoint.convert_hf import (
        deepseek_hf_to_mcore_state_dict,
        mcore_to_deepseek_hf_state_dict,
    )

    from megatron_amd.config import TransformerConfig
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.parallel.random import model_parallel_seed

    init_single()
    model_parallel_seed(19)
    cfg = TransformerConfig(
        num_layers=2, hidden_size=64, num_attention_heads=4, num_query_groups=4,
        ffn_hidden_size=128, vocab_size=128, max_position_embeddings=64,
        multi_latent_attention=True, q_lora_rank=48, kv_lora_rank=32,
        qk_nope_head_dim=16, qk_rope_head_dim=16, v_head_dim=16,
        num_experts=4, moe_router_topk=2, moe_ffn_hidden_size=32,
        moe_shared_expert_intermediate_size=32,
        moe_router_enable_expert_bias=True,
        untie_embeddings_and_output_weights=True)
    m = GPTModel(cfg)
    sd = {k: v for k, v in m.state_dict().items()}
    hf = mcore_to_deepseek_hf_state_dict(sd, cfg)
    assert "model.layers.0.self_attn.kv_a_proj_with_mqa.weight" in hf
    assert "model.layers.1.mlp.experts.3.down_proj.weight" in hf
    assert "model.layers.0.mlp.shared_experts.up_proj.weight" in hf
    assert "model.layers.0.mlp.gate.e_score_correction_bias" in hf
    back = deepseek_hf_to_mcore_state_dict(hf, cfg)
    for k, v in back.items():
        assert k in sd, k
        torch.testing.assert_close(v.float(), sd[k].float(), rtol=0, atol=0, msg=k)
    missing, unexpected = m.load_state_dict(back, strict=False)
    assert not unexpected, unexpected[:5]
    # only buffers/aux state may be missing, no weight tensors
    assert all("weight" not in k or "router" in k for k in missing), missing


def test_export_hf_dir_deepseek(tmp_path):
    """MLA models export as deepseek_v2-style HF dirs (weights + config)."""
    import json as _json

    from megatron_amd.config import TransformerConfig
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.parallel.random import model_parallel_seed
    from tools.checkpoint.export_hf import export_hf_dir

    init_single()
    model_parallel_seed(5)
    cfg = TransformerConfig(
        num_layers=2, hidden_size=64, num_attention_heads=4, num_query_groups=4,
        ffn_hidden_size=128, vocab_size=128, max_position_embeddings=64,
        multi_latent_attention=True, q_lora_rank=48, kv_lora_rank=32,
        qk_nope_head_dim=16, qk_rope_head_dim=16, v_head_dim=16,
        untie_embeddings_and_output_weights=True)
    m = GPTModel(cfg)
    out = str(tmp_path / "hf")
    hf_cfg = export_hf_dir(m, cfg, out)
    assert hf_cfg["model_type"] == "deepseek_v2"
    assert hf_cfg["kv_lora_rank"] == 32
    from safetensors.torch import load_file

    sd = load_file(out + "/model.safetensors")
    assert "model.layers.0.self_attn.kv_a_proj_with_mqa.weight" in sd
    assert _json.load(open(out + "/config.json"))["q_lora_rank"] == 48
