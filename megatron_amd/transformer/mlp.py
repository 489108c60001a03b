"""MLP (dense feed-forward) with fused gated activation.

Capability analog of reference megatron/core/transformer/mlp.py:
fc1 column-parallel (doubled width for gated activations) -> fused
SwiGLU/GeGLU HIP kernel (K5) -> fc2 row-parallel.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from megatron_amd import ops
from megatron_amd.parallel.layers import ColumnParallelLinear, RowParallelLinear


class MLP(nn.Module):
    def __init__(self, config, is_expert: bool = False, ffn_hidden_size: int = None):
        super().__init__()
        self.config = config
        ffn = ffn_hidden_size if ffn_hidden_size is not None else config.ffn_hidden_size
        self.gated = config.activation in ("swiglu", "geglu")
        fc1_out = 2 * ffn if self.gated else ffn
        self.linear_fc1 = ColumnParallelLinear(
            config.hidden_size, fc1_out, config=config, bias=config.add_linear_bias, is_expert=is_expert
        )
        self.linear_fc2 = RowParallelLinear(
            ffn, config.hidden_size, config=config, bias=config.add_linear_bias, is_expert=is_expert
        )
        if self.gated:
            # per-TP-rank layout is [gate_shard ; up_shard]: checkpointing must
            # split this weight into two global tensors to be reshard-safe
            self.linear_fc1.weight.is_gated_fc1 = True
        self.activation = config.activation

    def _act(self, x: torch.Tensor) -> torch.Tensor:
        if self.activation == "swiglu":
            return ops.swiglu(x)
        if self.activation == "geglu":
            return ops.geglu(x)
        if self.activation == "gelu":
            return torch.nn.functional.gelu(x)
        if self.activation == "squared_relu":
            return ops.squared_relu(x)
        raise ValueError(self.activation)

    def forward(self, hidden_states: torch.Tensor) -> torch.Tensor:
        h, _ = self.linear_fc1(hidden_states)
        if self.gated:
            # column-parallel sharding keeps [x1_shard | x2_shard] adjacency per rank:
            # fc1 weight rows are [ffn/tp gate rows ; ffn/tp up rows] interleaved per rank
            h = self._act(h)
        else:
            h = self._act(h)
        out, _ = self.linear_fc2(h)
        return out
