"""Top-K MoE router.

Capability analog of reference megatron/core/transformer/moe/router.py:144
(TopKRouter): fp32 gating, softmax/sigmoid score functions, pre/post-softmax
top-k, switch load-balancing aux loss (moe_utils.py:63), z-loss, and the
grad-injection scaler for aux losses (MoEAuxLossAutoScaler analog).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


class AuxLossScaler(torch.autograd.Function):
    """Pass activations through; inject d(aux_loss) with the main-loss scale
    in backward so router gradients flow without touching the loss plumbing."""

    main_loss_backward_scale: float = 1.0

    @staticmethod
    def forward(ctx, output, aux_loss):
        ctx.save_for_backward(aux_loss)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        (aux,) = ctx.saved_tensors
        scale = AuxLossScaler.main_loss_backward_scale
        return grad_output, torch.full_like(aux, scale)


class TopKRouter(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.config = config
        self.num_experts = config.num_experts
        self.topk = config.moe_router_topk
        self.score_function = config.moe_router_score_function
        self.pre_softmax = config.moe_router_pre_softmax
        # gating in fp32 (reference router.py gate fp32 option)
        self.weight = nn.Parameter(torch.empty(self.num_experts, config.hidden_size, dtype=torch.float32))
        with torch.no_grad():
            self.weight.normal_(0.0, config.init_method_std)
        # aux-free balancing bias (updated outside autograd)
        self.register_buffer("expert_bias", torch.zeros(self.num_experts), persistent=True)
        self.aux_losses = {}

    def forward(self, hidden: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """hidden [T, h] -> (probs [T, topk], indices [T, topk]); also stashes
        aux losses for injection by the MoE layer."""
        logits = F.linear(hidden.float(), self.weight)  # [T, E]
        self.aux_losses = {}

        if self.score_function == "sigmoid":
            scores = torch.sigmoid(logits)
            scores_for_topk = scores + self.expert_bias
            top_vals, top_idx = torch.topk(scores_for_topk, self.topk, dim=-1)
            probs = scores.gather(-1, top_idx)
            probs = probs / probs.sum(dim=-1, keepdim=True).clamp(min=1e-20)
            full_probs = scores / scores.sum(dim=-1, keepdim=True).clamp(min=1e-20)
        elif self.pre_softmax:
            full_probs = torch.softmax(logits, dim=-1)
            probs, top_idx = torch.topk(full_probs + self.expert_bias, self.topk, dim=-1)
            probs = full_probs.gather(-1, top_idx)
        else:
            top_logits, top_idx = torch.topk(logits + self.expert_bias, self.topk, dim=-1)
            top_logits = logits.gather(-1, top_idx)
            probs = torch.softmax(top_logits, dim=-1)
            full_probs = torch.softmax(logits, dim=-1)

        # switch load-balancing aux loss (reference moe_utils.py:63)
        if self.config.moe_aux_loss_coeff > 0:
            T = logits.shape[0]
            routing_map = torch.zeros_like(logits).scatter_(1, top_idx, 1.0)
            f = routing_map.mean(dim=0) * self.num_experts / self.topk  # fraction per expert
            P = full_probs.mean(dim=0)
            aux = (f * P).sum() * self.num_experts * self.config.moe_aux_loss_coeff
            self.aux_losses["load_balancing_loss"] = aux
        if self.config.moe_z_loss_coeff > 0:
            z = torch.logsumexp(logits, dim=-1)
            self.aux_losses["z_loss"] = (z.square()).mean() * self.config.moe_z_loss_coeff

        return probs, top_idx
