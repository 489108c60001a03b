"""Self-attention with GQA, RoPE, TP/SP sharding.

Capability analog of reference megatron/core/transformer/attention.py
(Attention :289, SelfAttention :1645): fused QKV column-parallel linear ->
GQA split -> RoPE -> fused (flash-style) core attention -> row-parallel proj.
The core attention is the hand-written CDNA4 MFMA kernel (ops.flash_attention,
K1 in SURVEY.md §2.3); the torch reference path runs on CPU.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from megatron_amd import ops
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.layers import ColumnParallelLinear, RowParallelLinear


class SelfAttention(nn.Module):
    def __init__(self, config, layer_number: int = 0):
        super().__init__()
        self.config = config
        self.layer_number = layer_number
        tp = G.get_tensor_model_parallel_world_size()
        self.hidden_size = config.hidden_size
        self.kv_channels = config.kv_channels
        self.num_heads = config.num_attention_heads
        self.num_query_groups = config.num_query_groups
        assert self.num_heads % tp == 0
        self.num_heads_per_partition = self.num_heads // tp
        if self.num_query_groups >= tp:
            assert self.num_query_groups % tp == 0
            self.num_query_groups_per_partition = self.num_query_groups // tp
        else:
            raise NotImplementedError("kv replication for tp > num_query_groups not yet supported")
        q_size = self.num_heads * self.kv_channels
        kv_size = self.num_query_groups * self.kv_channels
        self.qkv_size = q_size + 2 * kv_size
        self.linear_qkv = ColumnParallelLinear(
            self.hidden_size, self.qkv_size, config=config,
            bias=config.add_linear_bias or getattr(config, "add_qkv_bias", False)
        )
        self.linear_proj = RowParallelLinear(
            q_size, self.hidden_size, config=config, bias=config.add_linear_bias
        )
        self.softmax_scale = config.softmax_scale or (1.0 / (self.kv_channels**0.5))
        rs = getattr(config, "rope_scaling", None)
        if rs and rs.get("type") == "yarn":
            from megatron_amd.ops.reference import yarn_mscale

            # YaRN temperature: q and k tables each carry mscale -> scores
            # carry its square; equivalent to scaling softmax_scale here
            self.softmax_scale *= yarn_mscale(float(rs.get("factor", 8.0))) ** 2
        self.window = config.window_size
        skip = getattr(config, "window_attn_skip_freq", None)
        if self.window is not None and skip and (layer_number % skip == skip - 1):
            self.window = None  # this layer is a global-attention layer
        if config.qk_layernorm:
            self.q_layernorm = nn.Parameter(torch.ones(self.kv_channels, dtype=config.params_dtype))
            self.k_layernorm = nn.Parameter(torch.ones(self.kv_channels, dtype=config.params_dtype))
        else:
            self.q_layernorm = None
            self.k_layernorm = None

    def forward(self, hidden_states: torch.Tensor, rotary_freqs: Optional[torch.Tensor] = None,
                attention_mask=None, inference_context=None, packed_seq_params=None,
                attention_bias=None) -> torch.Tensor:
        # hidden_states: [s(/tp if SP), b, h]
        qkv, _ = self.linear_qkv(hidden_states)  # [s, b, qkv_size/tp]
        s, b = qkv.shape[0], qkv.shape[1]
        ng = self.num_query_groups_per_partition
        rep = self.num_heads_per_partition // ng
        d = self.kv_channels
        # fused-QKV layout: per query group [rep*d q | d k | d v] (reference attention.py GQA split)
        qkv = qkv.view(s, b, ng, (rep + 2) * d)
        q, k, v = torch.split(qkv, [rep * d, d, d], dim=3)
        q = q.reshape(s, b, ng * rep, d)
        k = k.reshape(s, b, ng, d)
        v = v.reshape(s, b, ng, d)

        if self.q_layernorm is not None:
            q = ops.rms_norm(q, self.q_layernorm, self.config.layernorm_epsilon)
            k = ops.rms_norm(k, self.k_layernorm, self.config.layernorm_epsilon)

        if rotary_freqs is not None:
            q = ops.rope_apply(q, rotary_freqs)
            k = ops.rope_apply(k, rotary_freqs)

        if getattr(self.config, "qk_clip_threshold", None) and self.training:
            from megatron_amd.optimizer.qk_clip import max_logits_per_group

            self.last_max_logit = max_logits_per_group(q, k, self.softmax_scale)

        if packed_seq_params is not None and inference_context is None:
            from megatron_amd.ops import reference as _ref

            core_out = _ref.attention_varlen(
                q, k, v, packed_seq_params.cu_seqlens.to(q.device),
                causal=self.config.causal_attention, scale=self.softmax_scale)
        elif inference_context is not None:
            # the context owns the KV cache and the attention kernel choice
            # (contiguous flash for prefill, paged masked decode for dynamic)
            core_out = inference_context.attend(
                self.layer_number, q, k, v, self.softmax_scale, self.window
            )
        elif attention_bias is not None:
            # additive score bias [hq_local, sq, sk] (T5 relative position
            # bias): unfused path — the bias re-materializes per step, so
            # the flash kernel's fused softmax does not apply.  Composes
            # with a key-padding mask (padded T5/BERT batches).
            from megatron_amd.ops import reference as _ref

            if attention_mask is not None:
                core_out = _ref.attention_padded(
                    q, k, v, attention_mask, causal=self.config.causal_attention,
                    scale=self.softmax_scale, bias=attention_bias)
            else:
                core_out = _ref.attention(
                    q, k, v, causal=self.config.causal_attention,
                    scale=self.softmax_scale, bias=attention_bias)
        elif attention_mask is not None:
            # key-padding mask ([b, s] bool, True = valid): the arbitrary-
            # mask path (BERT-style padded batches) — torch composition,
            # not the flash kernel (reference's unfused masked-softmax path)
            from megatron_amd.ops import reference as _ref

            core_out = _ref.attention_padded(
                q, k, v, attention_mask, causal=self.config.causal_attention,
                scale=self.softmax_scale)
        elif G.get_context_parallel_world_size() > 1:
            from megatron_amd.parallel.context_parallel import ring_attention, ulysses_attention

            if self.config.cp_comm_type == "a2a":
                core_out = ulysses_attention(q, k, v, scale=self.softmax_scale,
                                             window=self.window)
            else:
                core_out = ring_attention(q, k, v, scale=self.softmax_scale,
                                          window=self.window)
        elif (self.config.recompute_granularity == "selective" and self.training
              and "core_attn" in (self.config.recompute_modules or ["core_attn"])):
            # selective recompute: checkpoint only the core-attention region
            # (reference transformer_config 'selective' — the s^2-shaped
            # softmax state is recomputed in backward, everything else saved)
            from megatron_amd.parallel.random import checkpoint as rng_checkpoint

            def _core(q_, k_, v_):
                return ops.flash_attention(q_, k_, v_, causal=self.config.causal_attention,
                                           scale=self.softmax_scale, window=self.window)

            core_out = rng_checkpoint(_core, False, q, k, v)
        elif self.config.attention_dropout > 0.0 and self.training:
            # attention-prob dropout is not in the flash kernel; the torch
            # composition applies it exactly (RNG-tracker forked so TP ranks
            # draw independent masks for their own heads)
            from megatron_amd.ops import reference as _ref
            from megatron_amd.parallel.random import get_rng_tracker

            with get_rng_tracker().fork():
                core_out = _ref.attention(
                    q, k, v, causal=self.config.causal_attention,
                    scale=self.softmax_scale, window=self.window,
                    dropout_p=self.config.attention_dropout, training=True)
        else:
            core_out = ops.flash_attention(q, k, v, causal=self.config.causal_attention,
                                           scale=self.softmax_scale, window=self.window)
        core_out = core_out.reshape(s, b, ng * rep * d)
        out, _ = self.linear_proj(core_out)
        return out
