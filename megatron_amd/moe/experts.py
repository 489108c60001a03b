"""Grouped expert MLPs.

Capability analog of reference megatron/core/transformer/moe/experts.py
(TEGroupedMLP :182 / SequentialMLP :1263).  Weights live as single stacked
tensors [E_local, ...] so a grouped GEMM (hipBLASLt grouped API, K11) can
replace the per-expert loop without a layout change; expert ffn dim is
sharded over the (expert-)TP group like the dense MLP.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from megatron_amd import ops
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.mappings import _reduce
from megatron_amd.parallel.random import get_rng_tracker


class GroupedMLP(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.config = config
        ep = G.get_expert_model_parallel_world_size() if G.grid_initialized() else 1
        tp = G.get_tensor_model_parallel_world_size()
        self.num_local_experts = config.num_experts // max(ep, 1)
        ffn = config.moe_ffn_hidden_size
        assert ffn % tp == 0
        self.ffn_per_partition = ffn // tp
        self.gated = config.activation in ("swiglu", "geglu")
        fc1_out = 2 * self.ffn_per_partition if self.gated else self.ffn_per_partition
        self.weight1 = nn.Parameter(
            torch.empty(self.num_local_experts, fc1_out, config.hidden_size, dtype=config.params_dtype)
        )
        self.weight2 = nn.Parameter(
            torch.empty(self.num_local_experts, config.hidden_size, self.ffn_per_partition, dtype=config.params_dtype)
        )
        if self.gated:
            self.weight1.is_gated_fc1 = True
        for w in (self.weight1, self.weight2):
            w.is_expert_parallel = True
            w.tensor_parallel = tp > 1
            with get_rng_tracker().fork("expert-parallel-rng"):
                with torch.no_grad():
                    w.normal_(0.0, config.init_method_std)

    def _act(self, x):
        if self.config.activation == "swiglu":
            return ops.swiglu(x)
        if self.config.activation == "geglu":
            return ops.geglu(x)
        if self.config.activation == "squared_relu":
            return ops.squared_relu(x)
        return torch.nn.functional.gelu(x)

    def forward(self, tokens: torch.Tensor, tokens_per_expert: torch.Tensor) -> torch.Tensor:
        """tokens [Tlocal, h] sorted by local expert.

        GPU: ONE hipBLASLt grouped GEMM per linear (K11, ops.grouped_linear)
        — fc1 for all experts in one launch, fused activation, fc2 in one
        launch; the per-expert sizes sync below is the layer's single chosen
        host sync point (reference token_dispatcher.py:453-460).  CPU /
        no-native: per-expert matmul loop."""
        if tokens_per_expert.is_cuda:
            splits = tokens_per_expert.cpu().tolist()  # single sync per layer
        else:
            splits = [int(x) for x in tokens_per_expert.tolist()]
        if tokens.is_cuda and tokens.dtype == torch.bfloat16 and ops.has_native():
            h = ops.grouped_linear(tokens, self.weight1, splits)
            h = self._act(h)
            out = ops.grouped_linear(h, self.weight2, splits)
        else:
            outs = []
            start = 0
            for e, n in enumerate(splits):
                # n == 0 still goes through the matmuls: an empty [0, h] slice
                # keeps the autograd graph connected, so the EP a2a backward
                # runs on EVERY rank even when this rank's experts got no
                # tokens this microbatch (skipping it desyncs the per-pair
                # message order of the EP communicator).
                x = tokens[start : start + n]
                start += n
                h = torch.matmul(x, self.weight1[e].t())
                h = self._act(h)
                y = torch.matmul(h, self.weight2[e].t())
                outs.append(y)
            out = torch.cat(outs, dim=0) if len(outs) > 1 else outs[0]
        tp_group = G.get_tensor_model_parallel_group() if G.grid_initialized() else None
        if tp_group is not None and G.get_tensor_model_parallel_world_size() > 1:
            out = _ReduceExpertOutput.apply(out)
        return out


class _ReduceExpertOutput(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        return _reduce(x, G.get_tensor_model_parallel_group())

    @staticmethod
    def backward(ctx, g):
        return g
