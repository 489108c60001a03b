"""Hybrid Mamba/attention/MLP layer stack.

Capability analog of reference megatron/core/ssm/mamba_block.py
(MambaStack) + mamba_layer.py / mlp_layer.py: per-layer-type modules
(Mamba mixer layer, attention-only layer, MLP-only layer) assembled from
the hybrid allocation pattern, pipeline-stage aware, final norm on the
last stage.
"""

from __future__ import annotations

from typing import Optional

import torch.nn as nn

from megatron_amd import ops
from megatron_amd.ssm.hybrid_allocation import Symbols, allocate_layers
from megatron_amd.ssm.mamba_mixer import MambaMixer
from megatron_amd.transformer.attention import SelfAttention
from megatron_amd.transformer.block import Norm, get_layer_offset, get_num_layers_to_build
from megatron_amd.transformer.mlp import MLP


class MambaLayer(nn.Module):
    """norm -> MambaMixer -> residual (reference mamba_layer.py)."""

    def __init__(self, config, layer_number: int = 0):
        super().__init__()
        self.norm = Norm(config)
        self.mixer = MambaMixer(config, layer_number=layer_number)
        self.hidden_dropout = config.hidden_dropout

    def forward(self, hidden_states, rotary_freqs=None, attention_mask=None,
                inference_context=None, inference_state=None, packed_seq_params=None):
        residual = hidden_states
        x = self.norm(hidden_states)
        x = self.mixer(x, inference_state=inference_state)
        return ops.bias_dropout_add(x, None, residual, self.hidden_dropout, self.training)


class AttentionLayer(nn.Module):
    """norm -> self-attention -> residual (the '*' layer of a hybrid stack)."""

    def __init__(self, config, layer_number: int = 0):
        super().__init__()
        self.norm = Norm(config)
        self.self_attention = SelfAttention(config, layer_number=layer_number)
        self.hidden_dropout = config.hidden_dropout

    def forward(self, hidden_states, rotary_freqs=None, attention_mask=None,
                inference_context=None, inference_state=None, packed_seq_params=None):
        residual = hidden_states
        x = self.norm(hidden_states)
        x = self.self_attention(x, rotary_freqs=rotary_freqs, attention_mask=attention_mask,
                                inference_context=inference_context)
        return ops.bias_dropout_add(x, None, residual, self.hidden_dropout, self.training)


class MLPLayer(nn.Module):
    """norm -> MLP -> residual (the '-' layer; reference mlp_layer.py)."""

    def __init__(self, config, layer_number: int = 0):
        super().__init__()
        self.norm = Norm(config)
        self.mlp = MLP(config)
        self.hidden_dropout = config.hidden_dropout

    def forward(self, hidden_states, rotary_freqs=None, attention_mask=None,
                inference_context=None, inference_state=None, packed_seq_params=None):
        residual = hidden_states
        x = self.norm(hidden_states)
        x = self.mlp(x)
        return ops.bias_dropout_add(x, None, residual, self.hidden_dropout, self.training)


_LAYER_CLASSES = {
    Symbols.MAMBA: MambaLayer,
    Symbols.ATTENTION: AttentionLayer,
    Symbols.MLP: MLPLayer,
}


class MambaStack(nn.Module):
    def __init__(self, config, pre_process: bool = True, post_process: bool = True,
                 vp_stage: Optional[int] = None):
        super().__init__()
        self.config = config
        self.pre_process = pre_process
        self.post_process = post_process
        pattern = allocate_layers(
            config.num_layers,
            override_pattern=config.hybrid_override_pattern,
            attention_ratio=config.hybrid_attention_ratio,
            mlp_ratio=config.hybrid_mlp_ratio,
        )
        n_local = get_num_layers_to_build(config)
        offset = get_layer_offset(config, vp_stage)
        self.layer_types = pattern[offset : offset + n_local]
        self.layers = nn.ModuleList(
            [_LAYER_CLASSES[t](config, layer_number=offset + i) for i, t in enumerate(self.layer_types)]
        )
        self.final_norm = Norm(config) if post_process else None

    def allocate_inference_states(self, batch: int, device, dtype):
        """Per-layer Mamba decode state (None for non-Mamba layers)."""
        return [
            layer.mixer.allocate_inference_state(batch, device, dtype)
            if isinstance(layer, MambaLayer) else None
            for layer in self.layers
        ]

    def forward(self, hidden_states, rotary_freqs=None, attention_mask=None,
                inference_context=None, inference_states=None):
        for i, layer in enumerate(self.layers):
            hidden_states = layer(
                hidden_states, rotary_freqs=rotary_freqs, attention_mask=attention_mask,
                inference_context=inference_context,
                inference_state=None if inference_states is None else inference_states[i],
            )
        if self.final_norm is not None:
            hidden_states = self.final_norm(hidden_states)
        return hidden_states
