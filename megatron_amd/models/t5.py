"""T5-family encoder-decoder model.

Capability analog of reference megatron/core/models/T5/t5_model.py:
shared embedding -> bidirectional encoder -> causal decoder with per-layer
cross-attention over the encoder memory -> tied vocab projection + CE loss.
Positions are learned-absolute in v1 (the classic relative-position-bias form
needs an additive-bias attention path; tracked for the bias kernel pass)."""

from __future__ import annotations

import dataclasses
from typing import Optional

import torch
import torch.nn as nn

from megatron_amd import ops
from megatron_amd.parallel.cross_entropy import vocab_parallel_cross_entropy
from megatron_amd.parallel.layers import ColumnParallelLinear, VocabParallelEmbedding
from megatron_amd.transformer.attention import SelfAttention
from megatron_amd.transformer.block import Norm, TransformerBlock
from megatron_amd.transformer.cross_attention import CrossAttention
from megatron_amd.transformer.mlp import MLP


class T5DecoderLayer(nn.Module):
    """self-attn (causal) -> cross-attn (encoder memory) -> MLP, pre-norm."""

    def __init__(self, config, layer_number: int = 0):
        super().__init__()
        self.input_layernorm = Norm(config)
        self.self_attention = SelfAttention(config, layer_number=layer_number)
        self.pre_cross_layernorm = Norm(config)
        self.cross_attention = CrossAttention(config, layer_number=layer_number)
        self.pre_mlp_layernorm = Norm(config)
        self.mlp = MLP(config)

    def forward(self, hidden, memory, rotary_freqs=None):
        h = hidden + self.self_attention(self.input_layernorm(hidden), rotary_freqs=rotary_freqs)
        h = h + self.cross_attention(self.pre_cross_layernorm(h), memory)
        return h + self.mlp(self.pre_mlp_layernorm(h))


class T5Model(nn.Module):
    def __init__(self, config, pre_process: bool = True, post_process: bool = True,
                 vp_stage=None):
        super().__init__()
        if config.position_embedding_type == "rope":
            config.position_embedding_type = "learned"
        config.untie_embeddings_and_output_weights = False
        self.config = config
        self.pre_process, self.post_process = pre_process, post_process

        enc_cfg = dataclasses.replace(config, causal_attention=False)
        self.embedding = VocabParallelEmbedding(config.vocab_size, config.hidden_size, config=config)
        self.position_embedding = nn.Embedding(
            config.max_position_embeddings, config.hidden_size, dtype=config.params_dtype)
        self.encoder = TransformerBlock(enc_cfg, pre_process=True, post_process=True)
        dec_cfg = dataclasses.replace(config, causal_attention=True)
        self.decoder_layers = nn.ModuleList(
            [T5DecoderLayer(dec_cfg, layer_number=i) for i in range(config.num_layers)])
        self.final_layernorm = Norm(config)
        self.output_layer = ColumnParallelLinear(
            config.hidden_size, config.vocab_size, config=config, bias=False,
            gather_output=False, skip_bias_add=True)
        self.output_layer.weight = self.embedding.weight

    def _embed(self, ids: torch.Tensor) -> torch.Tensor:
        h = self.embedding(ids)  # [s, b, h]
        pos = torch.arange(ids.size(1), device=ids.device)
        return h + self.position_embedding(pos).unsqueeze(1).to(h.dtype)

    def forward(self, encoder_input_ids=None, decoder_input_ids=None, labels=None,
                loss_mask=None, **_):
        """encoder/decoder ids: [b, s_enc]/[b, s_dec]; returns loss [s_dec, b]
        (labels given) or decoder logits."""
        memory = self.encoder(self._embed(encoder_input_ids))
        h = self._embed(decoder_input_ids)
        for layer in self.decoder_layers:
            h = layer(h, memory)
        h = self.final_layernorm(h)
        logits, _ = self.output_layer(h)
        if labels is None:
            return logits
        loss = vocab_parallel_cross_entropy(logits, labels.transpose(0, 1).contiguous())
        if loss_mask is not None:
            loss = loss * loss_mask.transpose(0, 1).to(loss.dtype)
        return loss
