"""Layer-spec providers: declarative composition of transformer layers.

Capability analog of reference megatron/core/models/gpt/gpt_layer_specs.py
(:179 TE spec, :359 local spec, :676 get_gpt_decoder_block_spec — per-layer
dense/MoE mix).  One backend here (the CDNA4 kernel path), so the providers
express MODEL variants: dense GPT, MLA, MoE, and the per-layer mixed block.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Union

from megatron_amd.transformer.spec_utils import ModuleSpec


@dataclass
class TransformerLayerSubmodules:
    """Spec slots for one layer (reference transformer_layer.py:243)."""

    input_layernorm: Union[ModuleSpec, type, None] = None
    self_attention: Union[ModuleSpec, type, None] = None
    pre_mlp_layernorm: Union[ModuleSpec, type, None] = None
    mlp: Union[ModuleSpec, type, None] = None


def get_gpt_layer_spec(moe: bool = False, mla: bool = False) -> ModuleSpec:
    """The standard decoder layer: norm -> (self|MLA) attention -> norm ->
    (MLP|MoE)."""
    from megatron_amd.moe.moe_layer import MoELayer
    from megatron_amd.transformer.attention import SelfAttention
    from megatron_amd.transformer.block import Norm, TransformerLayer
    from megatron_amd.transformer.mlp import MLP
    from megatron_amd.transformer.multi_latent_attention import MLASelfAttention

    return ModuleSpec(
        module=TransformerLayer,
        submodules=TransformerLayerSubmodules(
            input_layernorm=Norm,
            self_attention=MLASelfAttention if mla else SelfAttention,
            pre_mlp_layernorm=Norm,
            mlp=MoELayer if moe else MLP,
        ),
    )


def moe_layer_pattern(config, layer_index: int) -> bool:
    """Whether layer ``layer_index`` (0-based global) is a MoE layer.
    config.moe_layer_freq: int N = every Nth layer (reference), or a 0/1
    list per layer (DeepSeek "first k dense" patterns)."""
    if config.num_experts is None:
        return False
    freq = config.moe_layer_freq
    if isinstance(freq, (list, tuple)):
        return bool(freq[layer_index % len(freq)])
    return layer_index % freq == freq - 1


def get_gpt_decoder_block_spec(config) -> List[ModuleSpec]:
    """Per-layer spec list mixing dense and MoE layers according to
    config.moe_layer_freq (reference gpt_layer_specs.py:676)."""
    specs = []
    for i in range(config.num_layers):
        specs.append(get_gpt_layer_spec(moe=moe_layer_pattern(config, i),
                                        mla=config.multi_latent_attention))
    return specs
