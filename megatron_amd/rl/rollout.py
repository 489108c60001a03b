"""Rollout generation: sample G responses per prompt from the policy.

Capability analog of the reference's agent/env abstraction (megatron/rl/
agent/api.py — agents return rollouts with token ids, logprobs, rewards).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Callable, List, Sequence

import torch

from megatron_amd.inference.engine import StaticInferenceEngine
from megatron_amd.inference.sampling import SamplingParams


class Environment:
    """Scores a (prompt, response) pair.  Subclass or pass a callable."""

    def score(self, prompt_tokens: List[int], response_tokens: List[int]) -> float:
        raise NotImplementedError


@dataclass
class Rollout:
    prompt_tokens: List[int]
    response_tokens: List[int]
    behavior_logprobs: List[float]  # logprob of each response token under the sampling policy
    reward: float = 0.0
    group: int = 0  # prompts index — rollouts with the same group share a baseline


def generate_rollouts(
    model,
    prompts: Sequence[List[int]],
    env,
    group_size: int = 4,
    max_tokens: int = 32,
    temperature: float = 1.0,
    seed: int = 0,
    max_seq: int = 2048,
) -> List[Rollout]:
    """Sample `group_size` responses per prompt with the in-process engine,
    score each with the environment."""
    score = env.score if isinstance(env, Environment) else env
    engine = StaticInferenceEngine(model, max_batch=len(prompts) * group_size,
                                   max_seq=max_seq)
    params = SamplingParams(max_tokens=max_tokens, temperature=temperature,
                            return_log_probs=True, seed=seed, stop_on_eod=False)
    batch = [list(p) for p in prompts for _ in range(group_size)]
    was_training = model.training
    model.eval()
    with torch.no_grad():
        results = engine.generate(batch, params)
    if was_training:
        model.train()
    rollouts = []
    for i, r in enumerate(results):
        g = i // group_size
        rollouts.append(
            Rollout(
                prompt_tokens=list(r.prompt_tokens),
                response_tokens=list(r.output_tokens),
                behavior_logprobs=list(r.log_probs),
                reward=float(score(list(r.prompt_tokens), list(r.output_tokens))),
                group=g,
            )
        )
    return rollouts
