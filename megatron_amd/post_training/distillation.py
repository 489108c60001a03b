"""Knowledge-distillation losses for post-training.

Capability analog of reference megatron/post_training/ (ModelOpt
distillation integration): temperature-softened logit KL against a frozen
teacher plus optional intermediate hidden-state matching, combined with the
standard LM loss.  TP-safe: when logits are vocab-parallel shards the KL is
computed with the same max/sum-reduce pattern as the vocab-parallel cross
entropy (two small all-reduces over the TP group).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from megatron_amd.parallel import grid as G


def _tp_group():
    if G.grid_initialized() and dist.is_initialized():
        g = G.get_grid()
        if g.tp > 1:
            return g.group("tp")
    return None


def soft_cross_entropy_vocab_parallel(student_logits: torch.Tensor,
                                      teacher_logits: torch.Tensor,
                                      temperature: float = 1.0) -> torch.Tensor:
    """KL(teacher || student) per token over (possibly TP-sharded) logits
    [*, V/tp], up to the teacher-entropy constant.  Returns [*] losses."""
    group = _tp_group()
    t = temperature

    def log_softmax_tp(x):
        m = x.max(dim=-1, keepdim=True).values
        if group is not None:
            dist.all_reduce(m, op=dist.ReduceOp.MAX, group=group)
        z = x - m
        se = z.exp().sum(dim=-1, keepdim=True)
        if group is not None:
            dist.all_reduce(se, group=group)
        return z - se.log()

    log_p_s = log_softmax_tp(student_logits.float() / t)
    log_p_t = log_softmax_tp(teacher_logits.float() / t)
    p_t = log_p_t.exp()
    # -sum_v p_t log p_s  (local vocab shard; sum over TP gives the total)
    loss = -(p_t * log_p_s).sum(dim=-1)
    if group is not None:
        dist.all_reduce(loss, group=group)
    return loss * (t * t)  # standard T^2 gradient scale


class DistillationLoss(nn.Module):
    """loss = alpha * KD(logits) + beta * MSE(hidden) + (1-alpha) * lm_loss."""

    def __init__(self, temperature: float = 2.0, alpha: float = 0.5,
                 hidden_beta: float = 0.0, student_hidden: Optional[int] = None,
                 teacher_hidden: Optional[int] = None):
        super().__init__()
        self.temperature = temperature
        self.alpha = alpha
        self.hidden_beta = hidden_beta
        self.proj = None
        if hidden_beta > 0 and student_hidden is not None and teacher_hidden is not None \
                and student_hidden != teacher_hidden:
            self.proj = nn.Linear(student_hidden, teacher_hidden, bias=False)

    def forward(self, student_logits: torch.Tensor, teacher_logits: torch.Tensor,
                lm_loss: Optional[torch.Tensor] = None,
                loss_mask: Optional[torch.Tensor] = None,
                student_hidden: Optional[torch.Tensor] = None,
                teacher_hidden: Optional[torch.Tensor] = None) -> torch.Tensor:
        kd = soft_cross_entropy_vocab_parallel(student_logits, teacher_logits.detach(),
                                               self.temperature)
        if loss_mask is not None:
            kd = (kd * loss_mask).sum() / loss_mask.sum().clamp(min=1)
        else:
            kd = kd.mean()
        total = self.alpha * kd
        if lm_loss is not None:
            total = total + (1.0 - self.alpha) * lm_loss
        if self.hidden_beta > 0 and student_hidden is not None and teacher_hidden is not None:
            h = student_hidden if self.proj is None else self.proj(student_hidden)
            total = total + self.hidden_beta * torch.nn.functional.mse_loss(
                h.float(), teacher_hidden.detach().float())
        return total
