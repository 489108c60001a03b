"""MoE layer: router -> dispatch -> experts (+ shared expert) -> combine.

Capability analog of reference megatron/core/transformer/moe/moe_layer.py:213.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from megatron_amd.moe.experts import GroupedMLP
from megatron_amd.moe.router import AuxLossScaler, TopKRouter
from megatron_amd.moe.token_dispatcher import MoEAllGatherTokenDispatcher, MoEAlltoAllTokenDispatcher
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.mappings import (
    gather_from_sequence_parallel_region,
    scatter_to_sequence_parallel_region,
)


class MoELayer(nn.Module):
    def __init__(self, config, layer_number: int = 0):
        super().__init__()
        self.config = config
        self.layer_number = layer_number
        self.router = TopKRouter(config)
        if config.moe_token_dispatcher_type == "alltoall":
            self.dispatcher = MoEAlltoAllTokenDispatcher(config)
        else:
            self.dispatcher = MoEAllGatherTokenDispatcher(config)
        self.experts = GroupedMLP(config)
        self.shared_expert = None
        self._comm_stream = None
        if config.moe_shared_expert_intermediate_size:
            from megatron_amd.transformer.mlp import MLP

            self.shared_expert = MLP(config, ffn_hidden_size=config.moe_shared_expert_intermediate_size)

    def forward(self, hidden_states: torch.Tensor) -> torch.Tensor:
        # [s(/tp if SP), b, h]
        s, b, h = hidden_states.shape
        if self.config.sequence_parallel and self.config.tensor_parallel_size > 1:
            full = gather_from_sequence_parallel_region(hidden_states)
        else:
            full = hidden_states
        tokens = full.reshape(-1, h)

        self.router.seq_len = full.shape[0]
        probs, top_idx = self.router(tokens)

        # shared-expert / dispatch-a2a overlap (reference shared_experts.py +
        # combined_1f1b.py's a2a-hiding intent): the EP all-to-all runs on a
        # dedicated HIP stream while the shared expert's GEMMs fill the
        # compute units; autograd replays each op on its recording stream so
        # the backward overlaps the same way.  On CPU streams are no-ops.
        shared_out = None
        if self.shared_expert is not None and torch.cuda.is_available() and tokens.is_cuda:
            if self._comm_stream is None:
                self._comm_stream = torch.cuda.Stream()
            self._comm_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self._comm_stream):
                dispatched, tokens_per_expert = self.dispatcher.dispatch(tokens, probs, top_idx)
            shared_out = self.shared_expert(full.reshape(-1, h))
            torch.cuda.current_stream().wait_stream(self._comm_stream)
            # comm-stream allocations consumed on the compute stream: pin
            # their lifetime to it so the caching allocator doesn't reuse
            # the blocks while the expert GEMMs still read them
            dispatched.record_stream(torch.cuda.current_stream())
            if tokens_per_expert.is_cuda:
                tokens_per_expert.record_stream(torch.cuda.current_stream())
        else:
            dispatched, tokens_per_expert = self.dispatcher.dispatch(tokens, probs, top_idx)
            if self.shared_expert is not None:
                shared_out = self.shared_expert(full.reshape(-1, h))

        expert_out = self.experts(dispatched, tokens_per_expert)
        out = self.dispatcher.combine(expert_out)

        if shared_out is not None:
            out = out + shared_out

        for aux in self.router.aux_losses.values():
            out = AuxLossScaler.apply(out, aux)

        out = out.view(full.shape).to(hidden_states.dtype)
        if self.config.sequence_parallel and self.config.tensor_parallel_size > 1:
            out = scatter_to_sequence_parallel_region(out)
        return out
