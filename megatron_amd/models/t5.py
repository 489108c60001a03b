"""T5-family encoder-decoder model.

Capability analog of reference megatron/core/models/T5/t5_model.py:
shared embedding -> bidirectional encoder -> causal decoder with per-layer
cross-attention over the encoder memory -> tied vocab projection + CE loss.
Positions: learned-absolute, or the classic T5 bucketed relative-position
bias (position_embedding_type="relative"): one bias table per stack,
shared by every layer's self-attention, added to the scores pre-softmax;
cross-attention carries no bias (T5 convention)."""

from __future__ import annotations

import dataclasses
import math
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


class T5RelativePositionBias(nn.Module):
    """Bucketed relative-position bias (T5): log-spaced buckets beyond
    max_exact, one learned scalar per (bucket, head); heads sliced to this
    TP partition."""

    def __init__(self, config, bidirectional: bool):
        super().__init__()
        from megatron_amd.parallel import grid as G

        self.num_buckets = config.relative_attention_num_buckets
        self.max_distance = config.relative_attention_max_distance
        self.bidirectional = bidirectional
        tp = G.get_tensor_model_parallel_world_size() if G.grid_initialized() else 1
        self.heads_per_partition = config.num_attention_heads // tp
        self.tp_rank = G.get_tensor_model_parallel_rank() if G.grid_initialized() else 0
        self.embedding = nn.Embedding(self.num_buckets, config.num_attention_heads,
                                      dtype=config.params_dtype)

    def _bucket(self, rel: torch.Tensor) -> torch.Tensor:
        # rel[i, j] = j - i (key pos minus query pos)
        n = self.num_buckets
        if self.bidirectional:
            n //= 2
            base = (rel > 0).long() * n
            rel = rel.abs()
        else:
            base = torch.zeros_like(rel)
            rel = (-rel).clamp(min=0)  # causal: only the past gets buckets
        max_exact = n // 2
        is_small = rel < max_exact
        # log-spaced buckets for distances in [max_exact, max_distance)
        log_pos = max_exact + (
            torch.log(rel.float().clamp(min=1) / max_exact)
            / math.log(self.max_distance / max_exact) * (n - max_exact)
        ).long()
        log_pos = log_pos.clamp(max=n - 1)
        return base + torch.where(is_small, rel, log_pos)

    def forward(self, sq: int, sk: int, device) -> torch.Tensor:
        """-> [heads/tp, sq, sk] additive bias."""
        q_pos = torch.arange(sq, device=device)[:, None]
        k_pos = torch.arange(sk, device=device)[None, :]
        buckets = self._bucket(k_pos - q_pos)  # [sq, sk]
        bias = self.embedding(buckets)  # [sq, sk, H]
        h0 = self.tp_rank * self.heads_per_partition
        return bias[..., h0 : h0 + self.heads_per_partition].permute(2, 0, 1)


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

    def forward(self, hidden, memory, rotary_freqs=None, attention_bias=None):
        h = hidden + self.self_attention(self.input_layernorm(hidden), rotary_freqs=rotary_freqs,
                                         attention_bias=attention_bias)
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
        self.relative_bias = config.position_embedding_type == "relative"

        enc_cfg = dataclasses.replace(config, causal_attention=False)
        self.embedding = VocabParallelEmbedding(config.vocab_size, config.hidden_size, config=config)
        if self.relative_bias:
            self.position_embedding = None
            self.encoder_rel_bias = T5RelativePositionBias(config, bidirectional=True)
            self.decoder_rel_bias = T5RelativePositionBias(config, bidirectional=False)
        else:
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
        if self.position_embedding is None:
            return h
        pos = torch.arange(ids.size(1), device=ids.device)
        return h + self.position_embedding(pos).unsqueeze(1).to(h.dtype)

    def forward(self, encoder_input_ids=None, decoder_input_ids=None, labels=None,
                loss_mask=None, encoder_attention_mask=None, **_):
        """encoder/decoder ids: [b, s_enc]/[b, s_dec]; returns loss [s_dec, b]
        (labels given) or decoder logits."""
        enc_bias = dec_bias = None
        if self.relative_bias:
            se, sd = encoder_input_ids.size(1), decoder_input_ids.size(1)
            enc_bias = self.encoder_rel_bias(se, se, encoder_input_ids.device)
            dec_bias = self.decoder_rel_bias(sd, sd, decoder_input_ids.device)
        memory = self.encoder(self._embed(encoder_input_ids), attention_bias=enc_bias,
                              attention_mask=encoder_attention_mask)
        h = self._embed(decoder_input_ids)
        for layer in self.decoder_layers:
            h = layer(h, memory, attention_bias=dec_bias)
        h = self.final_layernorm(h)
        logits, _ = self.output_layer(h)
        if labels is None:
            return logits
        loss = vocab_parallel_cross_entropy(logits, labels.transpose(0, 1).contiguous())
        if loss_mask is not None:
            loss = loss * loss_mask.transpose(0, 1).to(loss.dtype)
        return loss
