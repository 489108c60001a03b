"""Multi-token prediction (MTP).

Capability analog of reference megatron/core/transformer/
multi_token_prediction.py (1,884 LoC; DeepSeek-V3-style speculative heads):
depth-k head re-embeds the k-step-ahead token, fuses it with the running
hidden state through a projection + one extra transformer layer, and predicts
token i+k+1 through the shared output head. Losses are scaled by
mtp_loss_scaling_factor / num_depths and added to the main LM loss.

v1 scope: pipeline-last-stage only with pp=1 (the shared embedding lives on
the first stage; cross-stage embedding exchange is the reference's embd-group
all-reduce, planned for the PP integration pass).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from megatron_amd import ops
from megatron_amd.transformer.block import Norm, TransformerLayer


class MTPHead(nn.Module):
    def __init__(self, config, depth: int):
        super().__init__()
        h = config.hidden_size
        self.norm_hidden = Norm(config)
        self.norm_embed = Norm(config)
        self.proj = nn.Linear(2 * h, h, bias=False, dtype=config.params_dtype)
        self.layer = TransformerLayer(config, layer_number=config.num_layers + depth)


class MultiTokenPredictionBlock(nn.Module):
    """Chained MTP heads over the final decoder hidden state."""

    def __init__(self, config):
        super().__init__()
        self.config = config
        self.num_depths = config.mtp_num_layers
        self.heads = nn.ModuleList([MTPHead(config, k) for k in range(self.num_depths)])

    def forward(self, hidden, tokens, labels, embedding, output_layer, rotary,
                compute_loss) -> torch.Tensor:
        """hidden [s,b,h]; tokens/labels [b,s]. Returns the summed scaled MTP
        loss [s,b] (zero at positions without a k-ahead target)."""
        s = hidden.shape[0]
        scale = self.config.mtp_loss_scaling_factor / self.num_depths
        total = torch.zeros_like(hidden[..., 0])  # [s, b]
        h_k = hidden
        for k, head in enumerate(self.heads, start=1):
            # token stream shifted k ahead; tail positions have no target
            fut = torch.roll(tokens, shifts=-k, dims=1)
            fut[:, -k:] = 0
            emb = embedding(fut)  # [s, b, h]
            fused = torch.cat([head.norm_hidden(h_k), head.norm_embed(emb)], dim=-1)
            h_k = head.layer(self.proj_apply(head, fused), rotary_freqs=rotary)
            tgt = torch.roll(labels, shifts=-k, dims=1)
            tgt[:, -k:] = 0
            loss_k = compute_loss(output_layer(h_k)[0], tgt.transpose(0, 1).contiguous())
            mask = torch.ones_like(loss_k)
            mask[s - k:, :] = 0.0
            total = total + scale * loss_k * mask
        return total

    @staticmethod
    def proj_apply(head: MTPHead, fused: torch.Tensor) -> torch.Tensor:
        return head.proj(fused)
