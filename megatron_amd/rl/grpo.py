"""GRPO objective: group-relative advantages + clipped policy-gradient loss.

The reference's RL loop (megatron/rl/) implements GRPO-style training:
rewards are normalized within each prompt's group of G samples (no value
network), and the policy loss is the PPO clipped surrogate with an optional
KL penalty against a frozen reference policy.
"""

from __future__ import annotations

from typing import List, Optional

import torch


def group_relative_advantages(rewards: torch.Tensor, groups: torch.Tensor,
                              eps: float = 1e-6) -> torch.Tensor:
    """A_i = (r_i - mean(group)) / (std(group) + eps), per prompt group.

    rewards [N] fp32, groups [N] int64.  Single-sample groups get A=0.
    """
    adv = torch.zeros_like(rewards)
    for g in groups.unique():
        m = groups == g
        r = rewards[m]
        if r.numel() <= 1:
            continue
        adv[m] = (r - r.mean()) / (r.std(unbiased=False) + eps)
    return adv


def grpo_loss(
    logprobs: torch.Tensor,
    behavior_logprobs: torch.Tensor,
    advantages: torch.Tensor,
    response_mask: torch.Tensor,
    clip_ratio: float = 0.2,
    ref_logprobs: Optional[torch.Tensor] = None,
    kl_coeff: float = 0.0,
    advantages_per_token: bool = False,
) -> torch.Tensor:
    """Clipped surrogate averaged over response tokens.

    logprobs/behavior_logprobs/ref_logprobs: [N, T] per-token logprob of the
    chosen token; advantages [N] (or [N, T] with advantages_per_token, the
    THD-packed layout where every rollout's advantage is pre-spread along
    the pack); response_mask [N, T] (1 on response tokens).
    """
    ratio = torch.exp(logprobs - behavior_logprobs)
    adv = advantages if advantages_per_token else advantages.unsqueeze(-1)
    unclipped = ratio * adv
    clipped = torch.clamp(ratio, 1.0 - clip_ratio, 1.0 + clip_ratio) * adv
    obj = torch.minimum(unclipped, clipped)
    if kl_coeff > 0.0 and ref_logprobs is not None:
        # k3 estimator: exp(ref-pi) - (ref-pi) - 1 >= 0
        d = ref_logprobs - logprobs
        obj = obj - kl_coeff * (torch.exp(d) - d - 1.0)
    denom = response_mask.sum().clamp(min=1)
    return -(obj * response_mask).sum() / denom
