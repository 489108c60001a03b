"""Top-K MoE router.

Capability analog of reference megatron/core/transformer/moe/router.py:144
(TopKRouter): fp32 gating, softmax/sigmoid score functions, pre/post-softmax
top-k, group-limited (node-limited) top-k (moe_utils.py:673), input jitter,
switch/sequence load-balancing aux losses (moe_utils.py:63), z-loss,
aux-loss-free expert-bias balancing (finalize_model_grads.py:334 update rule),
and the grad-injection scaler for aux losses (MoEAuxLossAutoScaler analog).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


class AuxLossScaler(torch.autograd.Function):
    """Pass activations through; inject d(aux_loss) with the main-loss scale
    in backward so router gradients flow without touching the loss plumbing.

    The scale (1/(ntok*num_microbatches)) is only known after the microbatch's
    loss_func has run, i.e. after this Function's forward.  Under 1F1B the
    forward of microbatch i+k runs before the backward of microbatch i, so a
    plain class attribute would hand backward a *later* microbatch's scale
    (wrong with variable per-microbatch token counts).  Instead forward parks
    its ctx on a pending list and ``bind_scale`` — called by the scheduler
    right after loss_func — stamps the current scale onto exactly the ctxs of
    the microbatch that just finished its forward."""

    main_loss_backward_scale: float = 1.0
    _pending: list = []

    @staticmethod
    def forward(ctx, output, aux_loss):
        ctx.save_for_backward(aux_loss)
        ctx.scale = None
        if torch.is_grad_enabled():
            AuxLossScaler._pending.append(ctx)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        (aux,) = ctx.saved_tensors
        scale = ctx.scale if ctx.scale is not None else AuxLossScaler.main_loss_backward_scale
        return grad_output, torch.full_like(aux, scale)

    @staticmethod
    def bind_scale(scale: float):
        """Bind `scale` to every ctx created since the last call (i.e. this
        microbatch's MoE layers) and remember it as the fallback."""
        AuxLossScaler.main_loss_backward_scale = scale
        for ctx in AuxLossScaler._pending:
            ctx.scale = scale
        AuxLossScaler._pending.clear()


def group_limited_topk(
    scores: torch.Tensor, topk: int, num_groups: int, group_topk: int
) -> Tuple[torch.Tensor, torch.Tensor]:
    """DeepSeek node-limited routing (reference moe_utils.py:673).

    Experts are partitioned into `num_groups` contiguous groups (nodes); each
    token may only route to experts inside its best `group_topk` groups, where
    a group's score is the sum of its top-2 expert scores.  Returns
    (values, indices) of the final per-token topk over the masked scores.
    """
    T, E = scores.shape
    gsz = E // num_groups
    grouped = scores.view(T, num_groups, gsz)
    group_scores = grouped.topk(min(2, gsz), dim=-1).values.sum(dim=-1)  # [T, G]
    top_groups = group_scores.topk(group_topk, dim=-1).indices  # [T, group_topk]
    group_mask = torch.zeros_like(group_scores).scatter_(1, top_groups, 1.0)
    score_mask = group_mask.unsqueeze(-1).expand(T, num_groups, gsz).reshape(T, E)
    masked = scores.masked_fill(score_mask == 0, float("-inf"))
    return torch.topk(masked, topk, dim=-1)


class TopKRouter(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.config = config
        self.num_experts = config.num_experts
        self.topk = config.moe_router_topk
        self.score_function = config.moe_router_score_function
        self.pre_softmax = config.moe_router_pre_softmax
        self.num_groups = getattr(config, "moe_router_num_groups", None)
        self.group_topk = getattr(config, "moe_router_group_topk", None)
        self.jitter_eps = getattr(config, "moe_input_jitter_eps", None)
        # gating in fp32 (reference router.py gate fp32 option)
        self.weight = nn.Parameter(torch.empty(self.num_experts, config.hidden_size, dtype=torch.float32))
        with torch.no_grad():
            self.weight.normal_(0.0, config.init_method_std)
        # aux-free balancing bias (updated outside autograd by
        # megatron_amd.distributed.finalize.update_router_expert_bias)
        self.register_buffer("expert_bias", torch.zeros(self.num_experts), persistent=True)
        # per-step local routed-token counts, consumed by the bias update
        self.register_buffer("local_tokens_per_expert", torch.zeros(self.num_experts), persistent=False)
        self.seq_len: Optional[int] = None  # set by MoELayer for seq-aux loss
        self.aux_losses = {}

    def _topk(self, scores: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        if self.num_groups is not None and self.num_groups > 1:
            return group_limited_topk(scores, self.topk, self.num_groups, self.group_topk or 1)
        return torch.topk(scores, self.topk, dim=-1)

    def forward(self, hidden: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """hidden [T, h] -> (probs [T, topk], indices [T, topk]); also stashes
        aux losses for injection by the MoE layer."""
        if self.jitter_eps and self.training:
            # multiplicative input jitter (reference router.py apply_input_jitter)
            noise = torch.empty_like(hidden).uniform_(1.0 - self.jitter_eps, 1.0 + self.jitter_eps)
            hidden = hidden * noise
        # gate math in fp32 even if a module-wide .bfloat16() cast hit the weight
        logits = F.linear(hidden.float(), self.weight.float())  # [T, E]
        self.aux_losses = {}

        if self.score_function == "sigmoid":
            scores = torch.sigmoid(logits)
            _, top_idx = self._topk(scores + self.expert_bias)
            probs = scores.gather(-1, top_idx)
            probs = probs / probs.sum(dim=-1, keepdim=True).clamp(min=1e-20)
            full_probs = scores / scores.sum(dim=-1, keepdim=True).clamp(min=1e-20)
        elif self.pre_softmax:
            full_probs = torch.softmax(logits, dim=-1)
            _, top_idx = self._topk(full_probs + self.expert_bias)
            probs = full_probs.gather(-1, top_idx)
        else:
            _, top_idx = self._topk(logits + self.expert_bias)
            top_logits = logits.gather(-1, top_idx)
            probs = torch.softmax(top_logits, dim=-1)
            full_probs = torch.softmax(logits, dim=-1)

        if getattr(self.config, "moe_router_force_load_balancing", False):
            # benchmark/debug mode (reference force_load_balancing): override
            # the selection with a perfectly balanced round-robin assignment;
            # probs still come from the real gate so the numerics pipeline
            # (weighting, grads through probs) stays representative.
            T = logits.shape[0]
            k = self.topk
            base = torch.arange(T * k, device=logits.device) % self.num_experts
            top_idx = base.view(T, k)
            dup = top_idx[:, :1] == top_idx[:, 1:]  # avoid same expert twice
            if dup.any():
                top_idx = top_idx.clone()
                top_idx[:, 1:][dup] = (top_idx[:, 1:][dup] + 1) % self.num_experts
            probs = full_probs.gather(-1, top_idx)
            probs = probs / probs.sum(dim=-1, keepdim=True).clamp(min=1e-20)

        routing_map = torch.zeros_like(logits).scatter_(1, top_idx, 1.0)
        with torch.no_grad():
            self.local_tokens_per_expert = routing_map.sum(dim=0)

        # load-balancing aux loss: 'aux' = switch-style over the whole batch
        # (reference moe_utils.py:63); 'seq_aux' averages the loss per sequence
        # (DeepSeek-V2 style) using config.seq_length-sized slices.
        if self.config.moe_aux_loss_coeff > 0:
            if getattr(self.config, "moe_aux_loss_type", "aux") == "seq_aux" and self.seq_len:
                s = self.seq_len  # set by MoELayer from the activation shape
                T = logits.shape[0]
                if T % s == 0 and T >= s:
                    # tokens were flattened from [s, b, h]: dim 0 is sequence pos
                    rm = routing_map.view(s, -1, self.num_experts)
                    fp = full_probs.view(s, -1, self.num_experts)
                    f = rm.mean(dim=0) * self.num_experts / self.topk  # [B, E]
                    P = fp.mean(dim=0)
                    aux = (f * P).sum(dim=-1).mean() * self.num_experts * self.config.moe_aux_loss_coeff
                else:  # fall back to batch-level when tokens aren't seq-shaped
                    f = routing_map.mean(dim=0) * self.num_experts / self.topk
                    P = full_probs.mean(dim=0)
                    aux = (f * P).sum() * self.num_experts * self.config.moe_aux_loss_coeff
            else:
                f = routing_map.mean(dim=0) * self.num_experts / self.topk  # fraction per expert
                P = full_probs.mean(dim=0)
                aux = (f * P).sum() * self.num_experts * self.config.moe_aux_loss_coeff
            self.aux_losses["load_balancing_loss"] = aux
        if self.config.moe_z_loss_coeff > 0:
            z = torch.logsumexp(logits, dim=-1)
            self.aux_losses["z_loss"] = (z.square()).mean() * self.config.moe_z_loss_coeff

        return probs, top_idx
