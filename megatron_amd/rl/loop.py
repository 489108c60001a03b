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


def pack_rollouts_thd(rollouts: Sequence[Rollout], device):
    """THD packing (reference rl sequence packing): all rollouts concatenated
    into ONE [1, T] token stream with cu_seqlens — no pad tokens, so the
    policy-recompute forward does sum(len) work instead of N*max(len).

    Returns (ids [1, T], chosen [1, T], mask [1, T], behavior_lp [1, T],
    PackedSeqParams)."""
    from megatron_amd.transformer.packed_seq import PackedSeqParams

    seqs, lens = [], []
    for r in rollouts:
        seqs.append(r.prompt_tokens + r.response_tokens)
        lens.append(len(seqs[-1]))
    T = sum(lens)
    ids = torch.zeros(1, T, dtype=torch.long, device=device)
    chosen = torch.zeros(1, T, dtype=torch.long, device=device)
    mask = torch.zeros(1, T, dtype=torch.float32, device=device)
    beh = torch.zeros(1, T, dtype=torch.float32, device=device)
    off = 0
    for r, seq in zip(rollouts, seqs):
        ids[0, off : off + len(seq)] = torch.tensor(seq, dtype=torch.long, device=device)
        p = len(r.prompt_tokens)
        for j, (tok, lp) in enumerate(zip(r.response_tokens, r.behavior_logprobs)):
            pos = off + p - 1 + j
            chosen[0, pos] = tok
            mask[0, pos] = 1.0
            beh[0, pos] = lp
        off += len(seq)
    return ids, chosen, mask, beh, PackedSeqParams.from_lengths(lens, device=device)


def policy_logprobs_packed(model, ids, chosen, psp) -> torch.Tensor:
    """Per-position logprob of `chosen` under the model for a THD pack: [1, T]."""
    ce = vocab_parallel_cross_entropy(
        model(input_ids=ids, packed_seq_params=psp), chosen.transpose(0, 1).contiguous()
    )  # [T, 1]
    return -ce.transpose(0, 1)


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
    packed: bool = False,
):
    """One GRPO iteration.  Returns (loss, mean_reward, rollouts)."""
    rollouts = generate_rollouts(model, prompts, env, group_size=group_size,
                                 max_tokens=max_tokens, temperature=temperature, seed=seed)
    device = next(model.parameters()).device
    rewards = torch.tensor([r.reward for r in rollouts], dtype=torch.float32, device=device)
    groups = torch.tensor([r.group for r in rollouts], dtype=torch.long, device=device)
    adv_per_rollout = group_relative_advantages(rewards, groups)

    if packed:
        ids, chosen, mask, beh, psp = pack_rollouts_thd(rollouts, device)
        # spread per-rollout advantages along the pack
        adv = torch.zeros_like(mask)
        off = 0
        for i, r in enumerate(rollouts):
            n = len(r.prompt_tokens) + len(r.response_tokens)
            adv[0, off : off + n] = adv_per_rollout[i]
            off += n
        lp = policy_logprobs_packed(model, ids, chosen, psp)
        ref_lp = None
        if ref_model is not None and kl_coeff > 0.0:
            with torch.no_grad():
                ref_lp = policy_logprobs_packed(ref_model, ids, chosen, psp)
        loss = grpo_loss(lp, beh, adv, mask, clip_ratio=clip_ratio,
                         ref_logprobs=ref_lp, kl_coeff=kl_coeff, advantages_per_token=True)
        optimizer.zero_grad()
        loss.backward()
        if grad_clip is not None:
            torch.nn.utils.clip_grad_norm_(
                [p for p in model.parameters() if p.grad is not None], grad_clip)
        optimizer.step()
        return float(loss.detach()), float(rewards.mean()), rollouts

    adv = adv_per_rollout
    ids, chosen, mask, beh = pack_rollouts(rollouts, device)
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
