"""Mamba / hybrid Mamba-attention causal-LM.

Capability analog of reference megatron/core/models/mamba/mamba_model.py
(MambaModel): vocab-parallel embedding -> hybrid MambaStack (pattern of
Mamba / attention / MLP layers) -> shared or separate output layer ->
vocab-parallel cross-entropy.  Same forward contract as GPTModel
([b, s] ids in, [s, b] loss or [s, b, V/tp] logits out) so the training
app, schedules, and checkpointing treat both uniformly.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from megatron_amd.parallel import grid as G
from megatron_amd.parallel.cross_entropy import vocab_parallel_cross_entropy
from megatron_amd.parallel.layers import ColumnParallelLinear, VocabParallelEmbedding
from megatron_amd.ssm.hybrid_allocation import Symbols
from megatron_amd.ssm.mamba_block import MambaStack


class MambaModel(nn.Module):
    def __init__(self, config, pre_process: bool = True, post_process: bool = True,
                 vp_stage: Optional[int] = None):
        super().__init__()
        self.config = config
        self.pre_process = pre_process
        self.post_process = post_process
        self.share_embeddings_and_output_weights = not config.untie_embeddings_and_output_weights
        if pre_process:
            self.embedding = VocabParallelEmbedding(config.vocab_size, config.hidden_size, config=config)
        self.decoder = MambaStack(config, pre_process=pre_process, post_process=post_process,
                                  vp_stage=vp_stage)
        self.input_tensor = None
        if post_process:
            self.output_layer = ColumnParallelLinear(
                config.hidden_size, config.vocab_size, config=config, bias=False,
                gather_output=False, skip_bias_add=True,
            )
            if self.share_embeddings_and_output_weights and pre_process:
                self.output_layer.weight = self.embedding.weight
        self._rotary_cache = {}

    def set_input_tensor(self, tensor):
        self.input_tensor = tensor

    def _rotary_freqs(self, seq_len: int, device):
        if self.config.position_embedding_type != "rope":
            return None
        if not any(t == Symbols.ATTENTION for t in self.decoder.layer_types):
            return None
        from megatron_amd.ops import reference as ref

        key = (seq_len, str(device))
        if key not in self._rotary_cache:
            self._rotary_cache.clear()
            self._rotary_cache[key] = ref.rope_freqs(
                seq_len, self.config.kv_channels, base=self.config.rotary_base,
                device=device, rotary_percent=self.config.rotary_percent,
            )
        return self._rotary_cache[key]

    def forward(
        self,
        input_ids: Optional[torch.Tensor] = None,
        position_ids: Optional[torch.Tensor] = None,
        attention_mask: Optional[torch.Tensor] = None,
        labels: Optional[torch.Tensor] = None,
        loss_mask: Optional[torch.Tensor] = None,
        inference_context=None,
        inference_states=None,
    ):
        if self.pre_process:
            hidden = self.embedding(input_ids)  # [s, b, h]
            seq_len = input_ids.size(1)
        else:
            hidden = self.input_tensor
            assert hidden is not None
            seq_len = hidden.size(0)
        rotary = self._rotary_freqs(
            seq_len if inference_context is None else self.config.max_position_embeddings,
            hidden.device,
        )
        if rotary is not None and inference_context is not None:
            pos = inference_context.rope_positions(seq_len)
            rotary = rotary[pos]
        hidden = self.decoder(hidden, rotary_freqs=rotary, attention_mask=attention_mask,
                              inference_context=inference_context, inference_states=inference_states)
        if not self.post_process:
            return hidden
        logits, _ = self.output_layer(hidden)
        if labels is None:
            return logits
        labels_sb = labels.transpose(0, 1).contiguous()
        return vocab_parallel_cross_entropy(logits, labels_sb)
