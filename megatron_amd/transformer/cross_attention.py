"""Cross-attention (encoder-decoder): q from the decoder stream, k/v from the
encoder memory. Capability analog of the reference's CrossAttention
(megatron/core/transformer/attention.py, T5/retro path) on our flash kernel
(non-causal, sq != skv supported natively)."""

from __future__ import annotations

import torch
import torch.nn as nn

from megatron_amd import ops
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.layers import ColumnParallelLinear, RowParallelLinear


class CrossAttention(nn.Module):
    def __init__(self, config, layer_number: int = 0):
        super().__init__()
        self.config = config
        tp = G.get_tensor_model_parallel_world_size()
        d = config.kv_channels
        self.num_heads_per_partition = config.num_attention_heads // tp
        ng = config.num_query_groups or config.num_attention_heads
        assert ng % tp == 0
        self.num_query_groups_per_partition = ng // tp
        self.kv_channels = d
        self.linear_q = ColumnParallelLinear(
            config.hidden_size, config.num_attention_heads * d, config=config,
            bias=config.add_linear_bias)
        self.linear_kv = ColumnParallelLinear(
            config.hidden_size, 2 * ng * d, config=config, bias=config.add_linear_bias)
        self.linear_proj = RowParallelLinear(
            config.num_attention_heads * d, config.hidden_size, config=config,
            bias=config.add_linear_bias)
        self.softmax_scale = config.softmax_scale or (1.0 / (d ** 0.5))

    def forward(self, hidden_states: torch.Tensor, memory: torch.Tensor) -> torch.Tensor:
        """hidden_states [sq, b, h] (decoder), memory [skv, b, h] (encoder)."""
        sq, b = hidden_states.shape[0], hidden_states.shape[1]
        skv = memory.shape[0]
        d = self.kv_channels
        nh, ng = self.num_heads_per_partition, self.num_query_groups_per_partition
        q, _ = self.linear_q(hidden_states)
        q = q.view(sq, b, nh, d)
        kv, _ = self.linear_kv(memory)
        kv = kv.view(skv, b, ng, 2 * d)
        k, v = torch.split(kv, [d, d], dim=3)
        out = ops.flash_attention(q, k.contiguous(), v.contiguous(),
                                  causal=False, scale=self.softmax_scale)
        out, _ = self.linear_proj(out.reshape(sq, b, nh * d))
        return out
