"""TransformerLayer + TransformerBlock.

Capability analog of reference megatron/core/transformer/transformer_layer.py
(:302) and transformer_block.py (:486): pre-norm residual layers
(input_layernorm -> attention -> residual -> pre_mlp_layernorm -> MLP/MoE ->
residual), per-pipeline-stage layer construction, final norm, activation
recompute dispatch.
"""

from __future__ import annotations

from typing import Optional

import contextlib

import torch
import torch.nn as nn

from megatron_amd import ops
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.random import checkpoint as rng_checkpoint
from megatron_amd.transformer.attention import SelfAttention
from megatron_amd.transformer.mlp import MLP


class Norm(nn.Module):
    """RMSNorm (default, llama) or LayerNorm via the fused HIP kernel (K3)."""

    def __init__(self, config, hidden_size: Optional[int] = None):
        super().__init__()
        self.config = config
        h = hidden_size or config.hidden_size
        self.kind = config.normalization
        self.eps = config.layernorm_epsilon
        self.weight = nn.Parameter(torch.ones(h, dtype=config.params_dtype))
        self.weight.sequence_parallel_dup = True  # replicated over TP; grads all-reduced
        if self.kind == "layernorm":
            self.bias = nn.Parameter(torch.zeros(h, dtype=config.params_dtype))
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        if self.kind == "rmsnorm":
            return ops.rms_norm(x, self.weight, self.eps)
        return ops.reference.layer_norm(x, self.weight, self.bias, self.eps)


def _make_mixer(config, layer_number: int):
    from megatron_amd.transformer.layer_specs import moe_layer_pattern

    if moe_layer_pattern(config, layer_number):
        from megatron_amd.moe.moe_layer import MoELayer

        return MoELayer(config, layer_number=layer_number)
    return MLP(config)


class TransformerLayer(nn.Module):
    """One decoder layer.  `submodules` (a TransformerLayerSubmodules spec,
    reference transformer_layer.py:243) overrides any slot declaratively;
    unset slots fall back to the config-driven defaults."""

    def __init__(self, config, layer_number: int = 0, submodules=None):
        super().__init__()
        from megatron_amd.transformer.spec_utils import build_module

        self.config = config
        self.layer_number = layer_number
        sub = submodules
        if sub is not None and sub.input_layernorm is not None:
            self.input_layernorm = build_module(sub.input_layernorm, config)
        else:
            self.input_layernorm = Norm(config)
        if sub is not None and sub.self_attention is not None:
            self.self_attention = build_module(sub.self_attention, config, layer_number=layer_number)
        elif config.multi_latent_attention:
            from megatron_amd.transformer.multi_latent_attention import MLASelfAttention

            self.self_attention = MLASelfAttention(config, layer_number=layer_number)
        else:
            self.self_attention = SelfAttention(config, layer_number=layer_number)
        if sub is not None and sub.pre_mlp_layernorm is not None:
            self.pre_mlp_layernorm = build_module(sub.pre_mlp_layernorm, config)
        else:
            self.pre_mlp_layernorm = Norm(config)
        if sub is not None and sub.mlp is not None:
            mlp_cls = sub.mlp
            from megatron_amd.moe.moe_layer import MoELayer
            from megatron_amd.transformer.spec_utils import get_module

            if get_module(mlp_cls) is MoELayer:
                self.mlp = build_module(mlp_cls, config, layer_number=layer_number)
            else:
                self.mlp = build_module(mlp_cls, config)
        else:
            self.mlp = _make_mixer(config, layer_number)
        self.hidden_dropout = config.hidden_dropout

    def forward(self, hidden_states, rotary_freqs=None, attention_mask=None, inference_context=None, packed_seq_params=None, attention_bias=None):
        residual = hidden_states
        x = self.input_layernorm(hidden_states)
        kw = {"attention_bias": attention_bias} if attention_bias is not None else {}
        x = self.self_attention(x, rotary_freqs=rotary_freqs, attention_mask=attention_mask,
                                inference_context=inference_context, packed_seq_params=packed_seq_params,
                                **kw)
        x = ops.bias_dropout_add(x, None, residual, self.hidden_dropout, self.training)
        residual = x
        y = self.pre_mlp_layernorm(x)
        if (self.config.recompute_granularity == "selective" and self.training
                and any(m in ("mlp", "moe") for m in (self.config.recompute_modules or []))):
            # module-level recompute (reference --recompute-modules): the
            # MLP/MoE block (incl. EP a2a) reruns in backward under RNG replay
            from megatron_amd.parallel.random import checkpoint as rng_checkpoint

            y = rng_checkpoint(self.mlp, False, y)
        else:
            y = self.mlp(y)
        y = ops.bias_dropout_add(y, None, residual, self.hidden_dropout, self.training)
        return y


def _uneven_split(config):
    """Per-stage layer counts for uneven first/last splits, or None."""
    first = config.num_layers_in_first_pipeline_stage
    last = config.num_layers_in_last_pipeline_stage
    if first is None and last is None:
        return None
    pp = config.pipeline_parallel_size
    assert config.virtual_pipeline_parallel_size in (None, 1), (
        "uneven first/last stage splits are non-interleaved (vpp=1) in v1")
    L = config.num_layers
    mid_stages = pp - (first is not None) - (last is not None)
    mid_layers = L - (first or 0) - (last or 0)
    assert mid_stages >= 0 and mid_layers >= 0
    assert mid_stages == 0 or mid_layers % mid_stages == 0, (
        f"{mid_layers} middle layers must divide {mid_stages} middle stages")
    per_mid = mid_layers // mid_stages if mid_stages else 0
    counts = []
    for r in range(pp):
        if r == 0 and first is not None:
            counts.append(first)
        elif r == pp - 1 and last is not None:
            counts.append(last)
        else:
            counts.append(per_mid)
    assert sum(counts) == L, (counts, L)
    return counts


def get_num_layers_to_build(config) -> int:
    """Per-pipeline-stage layer count (reference transformer_block.py:71;
    uneven first/last splits via num_layers_in_first/last_pipeline_stage)."""
    pp = config.pipeline_parallel_size
    vpp = config.virtual_pipeline_parallel_size
    counts = _uneven_split(config)
    if counts is not None:
        r = G.get_pipeline_model_parallel_rank() if G.grid_initialized() else 0
        return counts[r]
    chunks = pp * (vpp or 1)
    assert config.num_layers % chunks == 0, (
        f"num_layers {config.num_layers} must divide pp*vpp = {chunks}"
    )
    return config.num_layers // chunks


def get_layer_offset(config, vp_stage: Optional[int] = None) -> int:
    """Global index of this stage's first layer (reference
    transformer_layer.py:57 get_transformer_layer_offset)."""
    pp_rank = G.get_pipeline_model_parallel_rank() if G.grid_initialized() else 0
    pp = config.pipeline_parallel_size
    vpp = config.virtual_pipeline_parallel_size
    counts = _uneven_split(config)
    if counts is not None:
        return sum(counts[:pp_rank])
    per_chunk = get_num_layers_to_build(config)
    if vpp is not None and vp_stage is not None:
        # interleaved: chunk c on pp rank r holds layers [ (c*pp + r) * per_chunk , ... )
        return (vp_stage * pp + pp_rank) * per_chunk
    return pp_rank * per_chunk


class TransformerBlock(nn.Module):
    """Layer stack for this pipeline stage.  `layer_specs` (optional list of
    ModuleSpec, one per GLOBAL layer — reference get_gpt_decoder_block_spec)
    switches construction to the declarative path."""

    def __init__(self, config, pre_process: bool = True, post_process: bool = True,
                 vp_stage: Optional[int] = None, layer_specs=None):
        super().__init__()
        self.config = config
        self.pre_process = pre_process
        self.post_process = post_process
        num_layers = get_num_layers_to_build(config)
        offset = get_layer_offset(config, vp_stage)
        if layer_specs is not None:
            from megatron_amd.transformer.spec_utils import build_module

            assert len(layer_specs) == config.num_layers
            self.layers = nn.ModuleList(
                [build_module(layer_specs[offset + i], config, layer_number=offset + i)
                 for i in range(num_layers)]
            )
        else:
            self.layers = nn.ModuleList(
                [TransformerLayer(config, layer_number=offset + i) for i in range(num_layers)]
            )
        self.final_layernorm = Norm(config) if post_process else None

    def _checkpointed(self, layer, *args):
        def run(*a):
            return layer(*a)

        return rng_checkpoint(run, self.config.distribute_saved_activations, *args)

    def forward(self, hidden_states, rotary_freqs=None, attention_mask=None, inference_context=None, packed_seq_params=None, attention_bias=None):
        recompute = (
            self.config.recompute_granularity == "full" and self.training and inference_context is None
        )
        num_ckpt = (
            self.config.recompute_num_layers
            if self.config.recompute_num_layers is not None
            else len(self.layers)
        )
        offload_n = 0
        if self.config.activation_cpu_offload and self.training and inference_context is None:
            offload_n = (self.config.activation_offload_layers
                         if self.config.activation_offload_layers is not None
                         else len(self.layers))
        for i, layer in enumerate(self.layers):
            if offload_n and i < offload_n and not (recompute and i < num_ckpt):
                # saved tensors of this layer go to pinned host memory and
                # come back at backward (reference cpu_offloading); pinning
                # only matters on GPU, CPU runs degenerate to plain saves
                ctx = torch.autograd.graph.save_on_cpu(pin_memory=hidden_states.is_cuda)
            else:
                ctx = contextlib.nullcontext()
            with ctx:
                if recompute and i < num_ckpt:
                    hidden_states = self._checkpointed(layer, hidden_states, rotary_freqs)
                    continue
                if getattr(self, "_graphed_layers", None) is not None and not recompute:
                    from megatron_amd.transformer.hip_graphs import graphed_layer_or_none

                    g = graphed_layer_or_none(self, i, hidden_states, inference_context)
                    if g is not None:
                        hidden_states = g(hidden_states)
                        continue
                kw = {"attention_bias": attention_bias} if attention_bias is not None else {}
                hidden_states = layer(
                    hidden_states, rotary_freqs=rotary_freqs, attention_mask=attention_mask,
                    inference_context=inference_context, packed_seq_params=packed_seq_params, **kw,
                )
        if self.final_layernorm is not None:
            hidden_states = self.final_layernorm(hidden_states)
        return hidden_states
