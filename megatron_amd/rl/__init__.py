"""Reinforcement-learning fine-tuning (GRPO).

Capability analog of reference megatron/rl/ (~6.6k LoC: agent/env rollout
abstraction, inference-server integration, `perform_rl_step`
training.py:3428).  MI355X-native shape: the policy, the reference policy
snapshot and the rollout engine live on the same 8-GPU node, so "refit" is
a direct weight reuse — rollouts run through the in-process inference
engine against the live training weights (no NVSHMEM copy service).
"""

from megatron_amd.rl.grpo import grpo_loss, group_relative_advantages
from megatron_amd.rl.rollout import Environment, Rollout, generate_rollouts
from megatron_amd.rl.loop import rl_step

__all__ = [
    "Environment",
    "Rollout",
    "generate_rollouts",
    "grpo_loss",
    "group_relative_advantages",
    "rl_step",
]
