"""HuggingFace <-> megatron_amd checkpoint conversion (Llama + Mixtral).

Capability analog of reference tools/checkpoint/convert.py with the
loader_llama/saver pairs: maps HF `LlamaForCausalLM` weights to our GPTModel
layout (fused per-query-group QKV, concatenated gated fc1) and back.

    # HF -> megatron_amd (single full-model checkpoint, TP/PP reshard on load)
    python tools/checkpoint/convert_hf.py --load-hf <hf_dir> --save <ckpt_dir>

    # megatron_amd -> HF
    python tools/checkpoint/convert_hf.py --load <ckpt_dir> --save-hf <hf_dir>
"""

from __future__ import annotations

import argparse
import sys

sys.path.insert(0, ".")

import torch  # noqa: E402


def hf_to_mcore_state_dict(hf_sd: dict, cfg) -> dict:
    """HF LlamaForCausalLM names -> our GPTModel names, with QKV fusion.

    Fused QKV row layout (attention.py GQA split): for each query group g:
    [rep*d rows of Q heads g*rep..(g+1)*rep) | d rows of K head g | d rows of
    V head g]. Gated fc1 = [gate ; up].
    """
    d = cfg.kv_channels
    ng = cfg.num_query_groups or cfg.num_attention_heads
    rep = cfg.num_attention_heads // ng
    out = {"embedding.weight": hf_sd["model.embed_tokens.weight"]}
    for i in range(cfg.num_layers):
        hf = f"model.layers.{i}."
        us = f"decoder.layers.{i}."
        q = hf_sd[hf + "self_attn.q_proj.weight"]
        k = hf_sd[hf + "self_attn.k_proj.weight"]
        v = hf_sd[hf + "self_attn.v_proj.weight"]
        groups = []
        for g in range(ng):
            groups.append(q[g * rep * d:(g + 1) * rep * d])
            groups.append(k[g * d:(g + 1) * d])
            groups.append(v[g * d:(g + 1) * d])
        out[us + "self_attention.linear_qkv.weight"] = torch.cat(groups, dim=0)
        if hf + "self_attn.q_proj.bias" in hf_sd:  # Qwen2-style QKV bias
            qb = hf_sd[hf + "self_attn.q_proj.bias"]
            kb = hf_sd[hf + "self_attn.k_proj.bias"]
            vb = hf_sd[hf + "self_attn.v_proj.bias"]
            bg = []
            for g in range(ng):
                bg.append(qb[g * rep * d:(g + 1) * rep * d])
                bg.append(kb[g * d:(g + 1) * d])
                bg.append(vb[g * d:(g + 1) * d])
            out[us + "self_attention.linear_qkv.bias"] = torch.cat(bg, dim=0)
        out[us + "self_attention.linear_proj.weight"] = hf_sd[hf + "self_attn.o_proj.weight"]
        if hf + "block_sparse_moe.gate.weight" in hf_sd:
            # Mixtral MoE block: gate -> router, experts w1/w3 -> fused
            # [gate;up] stacks, w2 -> weight2 [E, h, ffn]
            out[us + "mlp.router.weight"] = hf_sd[hf + "block_sparse_moe.gate.weight"].float()
            E = cfg.num_experts
            w1s, w2s = [], []
            for e in range(E):
                ex = hf + f"block_sparse_moe.experts.{e}."
                w1s.append(torch.cat([hf_sd[ex + "w1.weight"], hf_sd[ex + "w3.weight"]], dim=0))
                w2s.append(hf_sd[ex + "w2.weight"])
            out[us + "mlp.experts.weight1"] = torch.stack(w1s)
            out[us + "mlp.experts.weight2"] = torch.stack(w2s)
        else:
            out[us + "mlp.linear_fc1.weight"] = torch.cat(
                [hf_sd[hf + "mlp.gate_proj.weight"], hf_sd[hf + "mlp.up_proj.weight"]], dim=0)
            out[us + "mlp.linear_fc2.weight"] = hf_sd[hf + "mlp.down_proj.weight"]
        out[us + "input_layernorm.weight"] = hf_sd[hf + "input_layernorm.weight"]
        out[us + "pre_mlp_layernorm.weight"] = hf_sd[hf + "post_attention_layernorm.weight"]
    out["decoder.final_layernorm.weight"] = hf_sd["model.norm.weight"]
    if "lm_head.weight" in hf_sd:
        out["output_layer.weight"] = hf_sd["lm_head.weight"]
    else:  # tied embeddings
        out["output_layer.weight"] = hf_sd["model.embed_tokens.weight"]
    return out


def mcore_to_hf_state_dict(sd: dict, cfg) -> dict:
    d = cfg.kv_channels
    ng = cfg.num_query_groups or cfg.num_attention_heads
    rep = cfg.num_attention_heads // ng
    ffn = cfg.ffn_hidden_size
    out = {"model.embed_tokens.weight": sd["embedding.weight"]}
    for i in range(cfg.num_layers):
        hf = f"model.layers.{i}."
        us = f"decoder.layers.{i}."
        qkv = sd[us + "self_attention.linear_qkv.weight"]
        gsz = (rep + 2) * d
        qs, ks, vs = [], [], []
        for g in range(ng):
            blk = qkv[g * gsz:(g + 1) * gsz]
            qs.append(blk[: rep * d])
            ks.append(blk[rep * d: rep * d + d])
            vs.append(blk[rep * d + d:])
        out[hf + "self_attn.q_proj.weight"] = torch.cat(qs, 0)
        out[hf + "self_attn.k_proj.weight"] = torch.cat(ks, 0)
        out[hf + "self_attn.v_proj.weight"] = torch.cat(vs, 0)
        out[hf + "self_attn.o_proj.weight"] = sd[us + "self_attention.linear_proj.weight"]
        if us + "self_attention.linear_qkv.bias" in sd:
            b = sd[us + "self_attention.linear_qkv.bias"]
            qbs, kbs, vbs = [], [], []
            for g in range(ng):
                blk = b[g * gsz:(g + 1) * gsz]
                qbs.append(blk[: rep * d])
                kbs.append(blk[rep * d: rep * d + d])
                vbs.append(blk[rep * d + d:])
            out[hf + "self_attn.q_proj.bias"] = torch.cat(qbs, 0)
            out[hf + "self_attn.k_proj.bias"] = torch.cat(kbs, 0)
            out[hf + "self_attn.v_proj.bias"] = torch.cat(vbs, 0)
        if us + "mlp.router.weight" in sd:
            out[hf + "block_sparse_moe.gate.weight"] = sd[us + "mlp.router.weight"]
            w1 = sd[us + "mlp.experts.weight1"]  # [E, 2*moe_ffn, h]
            w2 = sd[us + "mlp.experts.weight2"]  # [E, h, moe_ffn]
            moe_ffn = w1.shape[1] // 2
            for e in range(w1.shape[0]):
                ex = hf + f"block_sparse_moe.experts.{e}."
                out[ex + "w1.weight"] = w1[e, :moe_ffn]
                out[ex + "w3.weight"] = w1[e, moe_ffn:]
                out[ex + "w2.weight"] = w2[e]
        else:
            fc1 = sd[us + "mlp.linear_fc1.weight"]
            out[hf + "mlp.gate_proj.weight"] = fc1[:ffn]
            out[hf + "mlp.up_proj.weight"] = fc1[ffn:]
            out[hf + "mlp.down_proj.weight"] = sd[us + "mlp.linear_fc2.weight"]
        out[hf + "input_layernorm.weight"] = sd[us + "input_layernorm.weight"]
        out[hf + "post_attention_layernorm.weight"] = sd[us + "pre_mlp_layernorm.weight"]
    out["model.norm.weight"] = sd["decoder.final_layernorm.weight"]
    out["lm_head.weight"] = sd["output_layer.weight"]
    return out


def deepseek_hf_to_mcore_state_dict(hf_sd: dict, cfg) -> dict:
    """DeepSeek-V2/V3-style HF names -> our GPTModel-with-MLA names.

    Attention: q_a/q_b (or q_proj), kv_a_proj_with_mqa ([kv_lora | k_rope]
    row order matches linear_kv_down), kv_b_proj (per-head [nope | v] matches
    linear_kv_up).  MoE layers: mlp.gate -> router, experts'
    gate/up/down -> fused [gate;up] weight1 / weight2 stacks, shared_experts
    -> mlp.shared_expert.  Structural mapping only — numerical parity
    additionally requires the same RoPE convention at runtime."""
    out = {"embedding.weight": hf_sd["model.embed_tokens.weight"]}
    for i in range(cfg.num_layers):
        hf = f"model.layers.{i}."
        us = f"decoder.layers.{i}."
        a = hf + "self_attn."
        ua = us + "self_attention."
        if a + "q_a_proj.weight" in hf_sd:  # low-rank Q
            out[ua + "linear_q_down.weight"] = hf_sd[a + "q_a_proj.weight"]
            out[ua + "q_norm.weight"] = hf_sd[a + "q_a_layernorm.weight"]
            out[ua + "linear_q_up.weight"] = hf_sd[a + "q_b_proj.weight"]
        else:
            out[ua + "linear_q_up.weight"] = hf_sd[a + "q_proj.weight"]
        out[ua + "linear_kv_down.weight"] = hf_sd[a + "kv_a_proj_with_mqa.weight"]
        out[ua + "kv_norm.weight"] = hf_sd[a + "kv_a_layernorm.weight"]
        out[ua + "linear_kv_up.weight"] = hf_sd[a + "kv_b_proj.weight"]
        out[ua + "linear_proj.weight"] = hf_sd[a + "o_proj.weight"]
        if hf + "mlp.gate.weight" in hf_sd:  # routed MoE layer
            out[us + "mlp.router.weight"] = hf_sd[hf + "mlp.gate.weight"].float()
            if hf + "mlp.gate.e_score_correction_bias" in hf_sd:
                out[us + "mlp.router.expert_bias"] = hf_sd[hf + "mlp.gate.e_score_correction_bias"].float()
            w1s, w2s = [], []
            for e in range(cfg.num_experts):
                ex = hf + f"mlp.experts.{e}."
                w1s.append(torch.cat([hf_sd[ex + "gate_proj.weight"],
                                      hf_sd[ex + "up_proj.weight"]], dim=0))
                w2s.append(hf_sd[ex + "down_proj.weight"])
            out[us + "mlp.experts.weight1"] = torch.stack(w1s)
            out[us + "mlp.experts.weight2"] = torch.stack(w2s)
            if hf + "mlp.shared_experts.gate_proj.weight" in hf_sd:
                sh = hf + "mlp.shared_experts."
                out[us + "mlp.shared_expert.linear_fc1.weight"] = torch.cat(
                    [hf_sd[sh + "gate_proj.weight"], hf_sd[sh + "up_proj.weight"]], dim=0)
                out[us + "mlp.shared_expert.linear_fc2.weight"] = hf_sd[sh + "down_proj.weight"]
        else:
            out[us + "mlp.linear_fc1.weight"] = torch.cat(
                [hf_sd[hf + "mlp.gate_proj.weight"], hf_sd[hf + "mlp.up_proj.weight"]], dim=0)
            out[us + "mlp.linear_fc2.weight"] = hf_sd[hf + "mlp.down_proj.weight"]
        out[us + "input_layernorm.weight"] = hf_sd[hf + "input_layernorm.weight"]
        out[us + "pre_mlp_layernorm.weight"] = hf_sd[hf + "post_attention_layernorm.weight"]
    out["decoder.final_layernorm.weight"] = hf_sd["model.norm.weight"]
    out["output_layer.weight"] = hf_sd.get("lm_head.weight", hf_sd["model.embed_tokens.weight"])
    return out


def mcore_to_deepseek_hf_state_dict(sd: dict, cfg) -> dict:
    """Inverse of deepseek_hf_to_mcore_state_dict (dense + routed layers)."""
    out = {"model.embed_tokens.weight": sd["embedding.weight"]}
    for i in range(cfg.num_layers):
        hf = f"model.layers.{i}."
        us = f"decoder.layers.{i}."
        a, ua = hf + "self_attn.", us + "self_attention."
        if ua + "linear_q_down.weight" in sd:
            out[a + "q_a_proj.weight"] = sd[ua + "linear_q_down.weight"]
            out[a + "q_a_layernorm.weight"] = sd[ua + "q_norm.weight"]
            out[a + "q_b_proj.weight"] = sd[ua + "linear_q_up.weight"]
        else:
            out[a + "q_proj.weight"] = sd[ua + "linear_q_up.weight"]
        out[a + "kv_a_proj_with_mqa.weight"] = sd[ua + "linear_kv_down.weight"]
        out[a + "kv_a_layernorm.weight"] = sd[ua + "kv_norm.weight"]
        out[a + "kv_b_proj.weight"] = sd[ua + "linear_kv_up.weight"]
        out[a + "o_proj.weight"] = sd[ua + "linear_proj.weight"]
        if us + "mlp.router.weight" in sd:
            out[hf + "mlp.gate.weight"] = sd[us + "mlp.router.weight"]
            if us + "mlp.router.expert_bias" in sd:
                out[hf + "mlp.gate.e_score_correction_bias"] = sd[us + "mlp.router.expert_bias"]
            w1 = sd[us + "mlp.experts.weight1"]
            w2 = sd[us + "mlp.experts.weight2"]
            half = w1.shape[1] // 2
            for e in range(w1.shape[0]):
                ex = hf + f"mlp.experts.{e}."
                out[ex + "gate_proj.weight"] = w1[e, :half]
                out[ex + "up_proj.weight"] = w1[e, half:]
                out[ex + "down_proj.weight"] = w2[e]
            if us + "mlp.shared_expert.linear_fc1.weight" in sd:
                fc1 = sd[us + "mlp.shared_expert.linear_fc1.weight"]
                sh = hf + "mlp.shared_experts."
                out[sh + "gate_proj.weight"] = fc1[: fc1.shape[0] // 2]
                out[sh + "up_proj.weight"] = fc1[fc1.shape[0] // 2 :]
                out[sh + "down_proj.weight"] = sd[us + "mlp.shared_expert.linear_fc2.weight"]
        else:
            fc1 = sd[us + "mlp.linear_fc1.weight"]
            out[hf + "mlp.gate_proj.weight"] = fc1[: fc1.shape[0] // 2]
            out[hf + "mlp.up_proj.weight"] = fc1[fc1.shape[0] // 2 :]
            out[hf + "mlp.down_proj.weight"] = sd[us + "mlp.linear_fc2.weight"]
        out[hf + "input_layernorm.weight"] = sd[us + "input_layernorm.weight"]
        out[hf + "post_attention_layernorm.weight"] = sd[us + "pre_mlp_layernorm.weight"]
    out["model.norm.weight"] = sd["decoder.final_layernorm.weight"]
    out["lm_head.weight"] = sd["output_layer.weight"]
    return out


def config_from_hf(hf_cfg):
    from megatron_amd.config import TransformerConfig

    return TransformerConfig(
        num_layers=hf_cfg.num_hidden_layers,
        hidden_size=hf_cfg.hidden_size,
        num_attention_heads=hf_cfg.num_attention_heads,
        num_query_groups=getattr(hf_cfg, "num_key_value_heads", hf_cfg.num_attention_heads),
        ffn_hidden_size=hf_cfg.intermediate_size,
        vocab_size=hf_cfg.vocab_size,
        max_position_embeddings=hf_cfg.max_position_embeddings,
        rotary_base=getattr(hf_cfg, "rope_theta", 10000.0),
        layernorm_epsilon=hf_cfg.rms_norm_eps,
        untie_embeddings_and_output_weights=not getattr(hf_cfg, "tie_word_embeddings", False),
        rope_scaling=(dict(getattr(hf_cfg, "rope_scaling")) if getattr(hf_cfg, "rope_scaling", None) else None),
        num_experts=(getattr(hf_cfg, "num_local_experts", None)
                     or getattr(hf_cfg, "n_routed_experts", None)),
        moe_router_topk=getattr(hf_cfg, "num_experts_per_tok", 2),
        moe_router_pre_softmax=False,
        add_qkv_bias=bool(getattr(hf_cfg, "attention_bias", False)
                          or "qwen2" in str(getattr(hf_cfg, "model_type", ""))),
        # DeepSeek-style MLA dims (None/0 elsewhere)
        multi_latent_attention=bool(getattr(hf_cfg, "kv_lora_rank", None)),
        q_lora_rank=getattr(hf_cfg, "q_lora_rank", None),
        kv_lora_rank=getattr(hf_cfg, "kv_lora_rank", None) or 0,
        qk_nope_head_dim=getattr(hf_cfg, "qk_nope_head_dim", None) or 0,
        qk_rope_head_dim=getattr(hf_cfg, "qk_rope_head_dim", None) or 0,
        v_head_dim=getattr(hf_cfg, "v_head_dim", None) or 0,
        moe_ffn_hidden_size=getattr(hf_cfg, "moe_intermediate_size", None),
        moe_shared_expert_intermediate_size=(
            (getattr(hf_cfg, "n_shared_experts", 0) or 0)
            * (getattr(hf_cfg, "moe_intermediate_size", 0) or 0) or None),
    )


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--load-hf", type=str, default=None, help="HF model dir")
    p.add_argument("--save", type=str, default=None, help="megatron_amd checkpoint dir")
    p.add_argument("--load", type=str, default=None)
    p.add_argument("--save-hf", type=str, default=None)
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    args = p.parse_args(argv)

    from megatron_amd.checkpoint import load_checkpoint, save_checkpoint
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.parallel import grid as G

    G.initialize_model_parallel(world_size=1, rank=0)
    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32

    if args.load_hf:
        from transformers import AutoConfig, AutoModelForCausalLM

        hf_cfg = AutoConfig.from_pretrained(args.load_hf)
        hf = AutoModelForCausalLM.from_pretrained(args.load_hf, torch_dtype=dtype)
        cfg = config_from_hf(hf_cfg)
        cfg.bf16 = args.dtype == "bf16"
        sd = hf_to_mcore_state_dict(hf.state_dict(), cfg)
        model = GPTModel(cfg)
        missing, unexpected = model.load_state_dict(sd, strict=False)
        assert not unexpected, f"unexpected keys: {unexpected[:5]}"
        save_checkpoint(args.save, [model], None, 0)
        print(f"saved megatron_amd checkpoint to {args.save} "
              f"({sum(p.numel() for p in model.parameters())/1e6:.1f}M params)")
    elif args.load:
        from transformers import AutoConfig, AutoModelForCausalLM

        raise SystemExit("--load -> --save-hf requires a config source; use "
                         "convert_to_hf() from python with an explicit TransformerConfig")
    else:
        p.error("need --load-hf/--save or --load/--save-hf")


if __name__ == "__main__":
    main()
