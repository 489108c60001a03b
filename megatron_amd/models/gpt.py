"""GPT / Llama-family causal-LM model.

Capability analog of reference megatron/core/models/gpt/gpt_model.py:51
(GPTModel): vocab-parallel embedding -> TransformerBlock -> column-parallel
output layer -> vocab-parallel cross-entropy.  Pipeline-stage aware via
pre_process/post_process; rotary table computed once and cached on device.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from megatron_amd.ops import reference as ref
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.cross_entropy import vocab_parallel_cross_entropy
from megatron_amd.parallel.layers import ColumnParallelLinear, VocabParallelEmbedding
from megatron_amd.parallel.mappings import (
    gather_from_tensor_model_parallel_region,
    scatter_to_sequence_parallel_region,
)
from megatron_amd.transformer.block import TransformerBlock


class GPTModel(nn.Module):
    def __init__(
        self,
        config,
        pre_process: bool = True,
        post_process: bool = True,
        vp_stage: Optional[int] = None,
    ):
        super().__init__()
        self.config = config
        self.pre_process = pre_process
        self.post_process = post_process
        self.vp_stage = vp_stage
        self.share_embeddings_and_output_weights = not config.untie_embeddings_and_output_weights

        if pre_process:
            self.embedding = VocabParallelEmbedding(config.vocab_size, config.hidden_size, config=config)
            # decoupled-lr group + Muon exclusion (reference is_embedding_or_output_parameter)
            self.embedding.weight.is_embedding_or_output_parameter = True
            self.embedding.weight.muon_exclude = True
            if config.position_embedding_type == "learned":
                self.position_embedding = nn.Embedding(
                    config.max_position_embeddings, config.hidden_size, dtype=config.params_dtype
                )
            else:
                self.position_embedding = None
        self.decoder = TransformerBlock(config, pre_process=pre_process, post_process=post_process, vp_stage=vp_stage)
        if post_process:
            self.output_layer = ColumnParallelLinear(
                config.hidden_size, config.vocab_size, config=config, bias=False, gather_output=False,
                skip_bias_add=True,
            )
            self.output_layer.weight.is_embedding_or_output_parameter = True
            self.output_layer.weight.muon_exclude = True
            if self.share_embeddings_and_output_weights and pre_process:
                self.output_layer.weight = self.embedding.weight
            self.mtp = None
            self.mtp_embedding = None
            if config.mtp_num_layers:
                from megatron_amd.transformer.multi_token_prediction import (
                    MultiTokenPredictionBlock,
                )

                self.mtp = MultiTokenPredictionBlock(config)
                if not pre_process:
                    # PP>1: the last stage needs a word-embedding replica for
                    # the MTP token re-embed; its grads are summed over the
                    # embd group and its value is broadcast-synced at setup
                    # (reference multi_token_prediction.py + finalize:164)
                    self.mtp_embedding = VocabParallelEmbedding(
                        config.vocab_size, config.hidden_size, config=config)
                    self.mtp_embedding.weight.is_embedding_or_output_parameter = True
                    self.mtp_embedding.weight.muon_exclude = True
                    if self.share_embeddings_and_output_weights:
                        # tied: the output weight on this stage IS the replica
                        self.mtp_embedding.weight = self.output_layer.weight
        self._rope_cache = {}
        # set by pipeline runner between stages
        self.input_tensor: Optional[torch.Tensor] = None

    # -- pipeline plumbing ---------------------------------------------------

    def set_input_tensor(self, input_tensor: Optional[torch.Tensor]):
        self.input_tensor = input_tensor

    def shared_embedding_or_output_weight(self):
        if self.pre_process:
            return self.embedding.weight
        if self.post_process:
            return self.output_layer.weight
        return None

    # -- rotary table ----------------------------------------------------------

    def _rotary_freqs(self, seq_len: int, device) -> Optional[torch.Tensor]:
        if self.config.position_embedding_type != "rope":
            return None
        key = (seq_len, str(device))
        if key not in self._rope_cache:
            self._rope_cache.clear()
            self._rope_cache[key] = ref.rope_freqs(
                seq_len,
                self.config.kv_channels,
                base=self.config.rotary_base,
                device=device,
                rotary_percent=self.config.rotary_percent,
                rope_scaling=getattr(self.config, "rope_scaling", None),
            )
        return self._rope_cache[key]

    # -- forward ---------------------------------------------------------------

    def forward(
        self,
        input_ids: Optional[torch.Tensor] = None,
        position_ids: Optional[torch.Tensor] = None,
        attention_mask: Optional[torch.Tensor] = None,
        labels: Optional[torch.Tensor] = None,
        loss_mask: Optional[torch.Tensor] = None,
        inference_context=None,
        packed_seq_params=None,
    ):
        """input_ids/labels: [b, s].  Returns loss [s, b] (labels given) or
        logits [s, b, V/tp]."""
        if self.pre_process:
            hidden = self.embedding(input_ids)  # [s(/tp if SP), b, h]
            if self.position_embedding is not None:
                if position_ids is None:
                    position_ids = torch.arange(input_ids.size(1), device=input_ids.device)
                pos = self.position_embedding(position_ids).unsqueeze(1)
                hidden = hidden + pos.to(hidden.dtype)
            seq_len = input_ids.size(1)
        else:
            hidden = self.input_tensor
            assert hidden is not None, "intermediate stage requires set_input_tensor"
            seq_len = hidden.size(0) * (
                G.get_tensor_model_parallel_world_size() if self.config.sequence_parallel else 1
            )

        cp = G.get_context_parallel_world_size()
        if cp > 1 and self.config.position_embedding_type == "rope":
            # tokens are CP-sharded: index the full-sequence freq table at this
            # rank's global positions (reference rope_utils.py:48 analog)
            from megatron_amd.parallel.context_parallel import cp_rope_positions

            s_global = seq_len * cp
            table = self._rotary_freqs(s_global, hidden.device)
            pos = cp_rope_positions(s_global, G.get_context_parallel_rank(), cp,
                                    hidden.device, mode=self.config.cp_comm_type)
            rotary = table[pos]
        elif inference_context is not None and self.config.position_embedding_type == "rope":
            # positions come from the KV context (cache offset / per-request
            # lengths under continuous batching): index the full freq table
            table = self._rotary_freqs(self.config.max_position_embeddings, hidden.device)
            pos = inference_context.rope_positions(seq_len)  # [s] or [s, b]
            rotary = table[pos]
        elif packed_seq_params is not None and self.config.position_embedding_type == "rope":
            # THD pack: positions restart at each document boundary
            table = self._rotary_freqs(packed_seq_params.max_seqlen, hidden.device)
            rotary = table[packed_seq_params.positions().to(hidden.device)]
        else:
            rotary = self._rotary_freqs(seq_len, hidden.device)
        hidden = self.decoder(hidden, rotary_freqs=rotary, attention_mask=attention_mask,
                              inference_context=inference_context,
                              packed_seq_params=packed_seq_params)

        if not self.post_process:
            return hidden

        if self.config.sequence_parallel:
            from megatron_amd.parallel.mappings import gather_from_sequence_parallel_region

            hidden = gather_from_sequence_parallel_region(hidden)
        logits, _ = self.output_layer(hidden)  # [s, b, V/tp]

        if labels is None:
            return logits
        labels_sb = labels.transpose(0, 1).contiguous()  # [s, b]
        loss = vocab_parallel_cross_entropy(logits, labels_sb,
                                            label_smoothing=self.config.label_smoothing)
        if self.mtp is not None and input_ids is not None:
            emb = self.embedding if self.pre_process else self.mtp_embedding
            loss = loss + self.mtp(
                hidden, input_ids, labels, emb, self.output_layer,
                rotary, vocab_parallel_cross_entropy)
        return loss
