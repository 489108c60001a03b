"""LR / weight-decay scheduler.

Capability analog of reference megatron/core/optimizer/optimizer_param_scheduler.py:
warmup + {constant, linear, cosine, WSD} decay, wd schedule.
"""

from __future__ import annotations

import math


class OptimizerParamScheduler:
    def __init__(self, optimizer, config, train_iters: int):
        self.optimizer = optimizer
        self.cfg = config
        self.max_lr = config.lr
        self.min_lr = config.min_lr
        self.warmup = config.lr_warmup_iters
        self.decay_iters = config.lr_decay_iters or train_iters
        self.style = config.lr_decay_style
        self.num_steps = 0

    def _lr(self, step: int) -> float:
        if self.warmup > 0 and step <= self.warmup:
            return self.max_lr * step / self.warmup
        if self.style == "constant":
            return self.max_lr
        if step > self.decay_iters:
            return self.min_lr
        decay_ratio = (step - self.warmup) / max(1, self.decay_iters - self.warmup)
        if self.style == "linear":
            coeff = 1.0 - decay_ratio
        elif self.style == "cosine":
            coeff = 0.5 * (math.cos(math.pi * decay_ratio) + 1.0)
        elif self.style == "wsd":
            wsd_start = self.decay_iters - (self.cfg.lr_wsd_decay_iters or self.decay_iters // 10)
            if step < wsd_start:
                return self.max_lr
            coeff = 1.0 - (step - wsd_start) / max(1, self.decay_iters - wsd_start)
        else:
            raise ValueError(self.style)
        return self.min_lr + coeff * (self.max_lr - self.min_lr)

    def step(self, increment: int = 1):
        self.num_steps += increment
        lr = self._lr(self.num_steps)
        self.optimizer.set_lr(lr)
        return lr

    def state_dict(self):
        return {"num_steps": self.num_steps}

    def load_state_dict(self, sd):
        self.num_steps = sd["num_steps"]
        self.optimizer.set_lr(self._lr(self.num_steps))
