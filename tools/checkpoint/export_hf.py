"""Export a megatron_amd checkpoint as a HuggingFace model directory.

Capability analog of reference megatron/core/export/ (its TRT-LLM target
has no AMD equivalent; the portable deployment interchange on ROCm is the
HF safetensors layout, loadable by transformers / vLLM-ROCm / llama.cpp
converters):

    python tools/checkpoint/export_hf.py --load <ckpt_dir> --save-hf <out_dir> [model flags]

Writes model.safetensors + config.json (LlamaForCausalLM or
MixtralForCausalLM depending on --num-experts).
"""

from __future__ import annotations

import argparse
import json
import os
import sys

sys.path.insert(0, ".")

import torch  # noqa: E402


def export_hf_dir(model, cfg, out_dir: str) -> dict:
    """Materialize the HF state dict + config for `model` into out_dir."""
    from safetensors.torch import save_file

    from tools.checkpoint.convert_hf import (
        mcore_to_deepseek_hf_state_dict,
        mcore_to_hf_state_dict,
    )

    core = model.module if hasattr(model, "module") else model
    sd = {k: v for k, v in core.state_dict().items()
          if "local_tokens" not in k}
    if getattr(cfg, "multi_latent_attention", False):
        hf_sd = mcore_to_deepseek_hf_state_dict(sd, cfg)
    else:
        sd = {k: v for k, v in sd.items() if "expert_bias" not in k}
        hf_sd = mcore_to_hf_state_dict(sd, cfg)
    hf_sd = {k: v.detach().to(torch.bfloat16 if cfg.bf16 else v.dtype).contiguous()
             for k, v in hf_sd.items()}
    os.makedirs(out_dir, exist_ok=True)
    save_file(hf_sd, os.path.join(out_dir, "model.safetensors"),
              metadata={"format": "pt"})
    is_moe = cfg.num_experts is not None
    is_mla = getattr(cfg, "multi_latent_attention", False)
    hf_cfg = {
        "architectures": (["DeepseekV2ForCausalLM"] if is_mla
                          else ["MixtralForCausalLM" if is_moe else "LlamaForCausalLM"]),
        "model_type": "deepseek_v2" if is_mla else ("mixtral" if is_moe else "llama"),
        "hidden_size": cfg.hidden_size,
        "intermediate_size": cfg.moe_ffn_hidden_size if is_moe else cfg.ffn_hidden_size,
        "num_hidden_layers": cfg.num_layers,
        "num_attention_heads": cfg.num_attention_heads,
        "num_key_value_heads": cfg.num_query_groups,
        "vocab_size": cfg.vocab_size,
        "max_position_embeddings": cfg.max_position_embeddings,
        "rope_theta": cfg.rotary_base,
        "rms_norm_eps": cfg.layernorm_epsilon,
        "tie_word_embeddings": not cfg.untie_embeddings_and_output_weights,
        "torch_dtype": "bfloat16" if cfg.bf16 else "float32",
        "hidden_act": "silu",
    }
    if is_moe:
        hf_cfg["num_local_experts"] = cfg.num_experts
        hf_cfg["num_experts_per_tok"] = cfg.moe_router_topk
    if is_mla:
        hf_cfg.update({
            "q_lora_rank": cfg.q_lora_rank, "kv_lora_rank": cfg.kv_lora_rank,
            "qk_nope_head_dim": cfg.qk_nope_head_dim,
            "qk_rope_head_dim": cfg.qk_rope_head_dim, "v_head_dim": cfg.v_head_dim,
        })
        if is_moe:
            hf_cfg["n_routed_experts"] = cfg.num_experts
            hf_cfg["moe_intermediate_size"] = cfg.moe_ffn_hidden_size
    if cfg.rope_scaling:
        hf_cfg["rope_scaling"] = cfg.rope_scaling
    with open(os.path.join(out_dir, "config.json"), "w") as f:
        json.dump(hf_cfg, f, indent=1)
    return hf_cfg


def main(argv=None):
    from megatron_amd.checkpoint.checkpointing import load_checkpoint
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.parallel import grid as G
    from megatron_amd.training.arguments import build_arg_parser, configs_from_args

    parser = build_arg_parser()
    parser.add_argument("--save-hf", type=str, required=True)
    args = parser.parse_args(argv)
    args.world_size, args.rank = 1, 0
    cfg, _, _ = configs_from_args(args)
    G.initialize_model_parallel(world_size=1, rank=0)
    model = GPTModel(cfg)
    if args.load:
        load_checkpoint(args.load, [model], None, load_rng=False)
    export_hf_dir(model, cfg, args.save_hf)
    print(f"exported HF model to {args.save_hf}")


if __name__ == "__main__":
    main()
