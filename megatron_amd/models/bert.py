"""BERT-family bidirectional encoder.

Capability analog of reference megatron/core/models/bert/bert_model.py:
word + learned-position (+ tokentype) embeddings -> bidirectional
TransformerBlock -> MLM head (dense + gelu + norm -> tied vocab-parallel
output) with masked-LM loss; optional sequence-level binary head."""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from megatron_amd import ops
from megatron_amd.parallel.cross_entropy import vocab_parallel_cross_entropy
from megatron_amd.parallel.layers import ColumnParallelLinear, VocabParallelEmbedding
from megatron_amd.transformer.block import Norm, TransformerBlock


class BertLMHead(nn.Module):
    """dense -> gelu -> norm before the (tied) vocab projection."""

    def __init__(self, config):
        super().__init__()
        h = config.hidden_size
        self.dense = nn.Linear(h, h, dtype=config.params_dtype)
        self.norm = Norm(config)

    def forward(self, hidden):
        return self.norm(torch.nn.functional.gelu(self.dense(hidden)))


class BertModel(nn.Module):
    def __init__(self, config, num_tokentypes: int = 2, add_binary_head: bool = False,
                 pre_process: bool = True, post_process: bool = True, vp_stage=None):
        super().__init__()
        config.causal_attention = False
        if config.position_embedding_type == "rope":
            config.position_embedding_type = "learned"
        config.untie_embeddings_and_output_weights = False
        self.config = config
        self.pre_process, self.post_process = pre_process, post_process
        if pre_process:
            self.embedding = VocabParallelEmbedding(config.vocab_size, config.hidden_size, config=config)
            self.position_embedding = nn.Embedding(
                config.max_position_embeddings, config.hidden_size, dtype=config.params_dtype)
            self.tokentype_embedding = (
                nn.Embedding(num_tokentypes, config.hidden_size, dtype=config.params_dtype)
                if num_tokentypes else None)
        self.encoder = TransformerBlock(config, pre_process=pre_process,
                                        post_process=post_process, vp_stage=vp_stage)
        if post_process:
            self.lm_head = BertLMHead(config)
            self.output_layer = ColumnParallelLinear(
                config.hidden_size, config.vocab_size, config=config, bias=False,
                gather_output=False, skip_bias_add=True)
            if pre_process:
                self.output_layer.weight = self.embedding.weight
            self.pooler = (nn.Linear(config.hidden_size, config.hidden_size,
                                     dtype=config.params_dtype) if add_binary_head else None)
            self.binary_head = (nn.Linear(config.hidden_size, 2, dtype=config.params_dtype)
                                if add_binary_head else None)
        self.input_tensor: Optional[torch.Tensor] = None

    def set_input_tensor(self, t):
        self.input_tensor = t

    def forward(self, input_ids=None, tokentype_ids=None, labels=None, loss_mask=None,
                position_ids=None, attention_mask=None, inference_context=None):
        """input_ids/labels [b, s]. Returns masked-LM loss [s, b] when labels
        given (positions with loss_mask==0 contribute 0), else logits."""
        if self.pre_process:
            hidden = self.embedding(input_ids)  # [s, b, h]
            s = input_ids.size(1)
            if position_ids is None:
                position_ids = torch.arange(s, device=input_ids.device)
            hidden = hidden + self.position_embedding(position_ids).unsqueeze(1).to(hidden.dtype)
            if self.tokentype_embedding is not None and tokentype_ids is not None:
                hidden = hidden + self.tokentype_embedding(tokentype_ids).transpose(0, 1).to(hidden.dtype)
        else:
            hidden = self.input_tensor
        hidden = self.encoder(hidden, attention_mask=attention_mask)
        if not self.post_process:
            return hidden
        logits, _ = self.output_layer(self.lm_head(hidden))  # [s, b, V/tp]
        self.binary_logits = None
        if self.binary_head is not None:
            # reference Pooler: tanh(dense(first-token hidden)) -> NSP head
            pooled = torch.tanh(self.pooler(hidden[0]))
            self.binary_logits = self.binary_head(pooled)
        if labels is None:
            return logits
        loss = vocab_parallel_cross_entropy(logits, labels.transpose(0, 1).contiguous())
        if loss_mask is not None:
            loss = loss * loss_mask.transpose(0, 1).to(loss.dtype)
        return loss
