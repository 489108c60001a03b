"""The RL training step: rollouts -> advantages -> GRPO loss -> optimizer.

Capability analog of reference `perform_rl_step` (training.py:3428) +
sequence packing.  Logprobs under the current policy are recomputed with
gradients via vocab-parallel cross-entropy (logprob = -CE), so the step is
TP/SP-correct without gathering full-vocab logits.
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import torch

from megatron_amd.parallel.cross_entropy import vocab_parallel_cross_entropy
from megatron_amd.rl.grpo import group_relative_advantages, grpo_loss
from megatron_amd.rl.rollout import Rollout, generate_rollouts


def pack_rollouts(rollouts: Sequence[Rollout], device, pad_id: int = 0):
    """Right-pad full sequences (prompt+response) to a common length.

    Returns (input_ids [N, L], chosen [N, L] next-token ids, response_mask
    [N, L] 1.0 where position PREDICTS a response token, behavior_logprobs
    [N, L]).
    """
    n = len(rollouts)
    lens = [len(r.prompt_tokens) + len(r.response_tokens) for r in rollouts]
    L = max(lens)
    ids = torch.full((n, L), pad_id, dtype=torch.long, device=device)
    chosen = torch.full((n, L), pad_id, dtype=torch.long, device=device)
    mask = torch.zeros(n, L, dtype=torch.float32, device=device)
    beh = torch.zeros(n, L, dtype=torch.float32, device=device)
    for i, r in enumerate(rollouts):
        seq = r.prompt_tokens + r.response_tokens
        ids[i, : len(seq)] = torch.tensor(seq, dtype=torch.long, device=device)
        p = len(r.prompt_tokens)
        # position t predicts token t+1: positions p-1 .. p+len(resp)-2
        for j, (tok, lp) in enumerate(zip(r.response_tokens, r.behavior_logprobs)):
            pos = p - 1 + j
            chosen[i, pos] = tok
            mask[i, pos] = 1.0
            beh[i, pos] = lp
    return ids, chosen, mask, beh


def policy_logprobs(model, input_ids: torch.Tensor, chosen: torch.Tensor) -> torch.Tensor:
    """Per-position logprob of `chosen` under the model: [N, L]."""
    ce = vocab_parallel_cross_entropy(
        model(input_ids=input_ids), chosen.transpose(0, 1).contiguous()
    )  # [L, N] = -logprob
    return -ce.transpose(0, 1)


def rl_step(
    model,
    optimizer,
    prompts: Sequence[List[int]],
    env,
    group_size: int = 4,
    max_tokens: int = 16,
    temperature: float = 1.0,
    clip_ratio: float = 0.2,
    kl_coeff: float = 0.0,
    ref_model=None,
    seed: int = 0,
    grad_clip: Optional[float] = 1.0,
):
    """One GRPO iteration.  Returns (loss, mean_reward, rollouts)."""
    rollouts = generate_rollouts(model, prompts, env, group_size=group_size,
                                 max_tokens=max_tokens, temperature=temperature, seed=seed)
    device = next(model.parameters()).device
    ids, chosen, mask, beh = pack_rollouts(rollouts, device)
    rewards = torch.tensor([r.reward for r in rollouts], dtype=torch.float32, device=device)
    groups = torch.tensor([r.group for r in rollouts], dtype=torch.long, device=device)
    adv = group_relative_advantages(rewards, groups)

    lp = policy_logprobs(model, ids, chosen)
    ref_lp = None
    if ref_model is not None and kl_coeff > 0.0:
        with torch.no_grad():
            ref_lp = policy_logprobs(ref_model, ids, chosen)
    loss = grpo_loss(lp, beh, adv, mask, clip_ratio=clip_ratio,
                     ref_logprobs=ref_lp, kl_coeff=kl_coeff)
    optimizer.zero_grad()
    loss.backward()
    if grad_clip is not None:
        torch.nn.utils.clip_grad_norm_([p for p in model.parameters() if p.grad is not None],
                                       grad_clip)
    optimizer.step()
    return float(loss.detach()), float(rewards.mean()), rollouts
