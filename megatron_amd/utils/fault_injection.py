"""Fault injection for resilience testing.

Capability analog of reference megatron/core/fault_injector.py:48
(`FaultInjectorConfig` delegating to nvidia_resiliency_ext, hooked into
train() at training.py:3404-3425): deterministically or stochastically
inject hangs, crashes, or NaN'd losses on chosen ranks/iterations so the
rerun state machine, signal handlers, and checkpoint-resume paths can be
exercised end to end without real hardware faults.
"""

from __future__ import annotations

import random
import time
from dataclasses import dataclass, field
from typing import List, Optional


class InjectedFault(RuntimeError):
    pass


@dataclass
class FaultInjectorConfig:
    enabled: bool = False
    ranks: List[int] = field(default_factory=list)      # empty -> all ranks
    fault_type: str = "crash"                           # 'crash' | 'hang' | 'nan_loss'
    # deterministic trigger: fire exactly at this iteration (takes precedence)
    at_iteration: Optional[int] = None
    # stochastic trigger: mean time (iterations) to injection, exponential
    mtti_iterations: Optional[float] = None
    hang_seconds: float = 30.0
    seed: int = 0


class FaultInjector:
    def __init__(self, config: FaultInjectorConfig, rank: int = 0):
        self.config = config
        self.rank = rank
        self._rng = random.Random(config.seed + rank)
        self._next_stochastic: Optional[int] = None
        if config.mtti_iterations and config.at_iteration is None:
            self._next_stochastic = 1 + int(self._rng.expovariate(1.0 / config.mtti_iterations))
        self.fired = False

    def _applies(self) -> bool:
        return self.config.enabled and (not self.config.ranks or self.rank in self.config.ranks)

    def should_fire(self, iteration: int) -> bool:
        if not self._applies() or self.fired:
            return False
        if self.config.at_iteration is not None:
            return iteration == self.config.at_iteration
        if self._next_stochastic is not None:
            return iteration >= self._next_stochastic
        return False

    def maybe_inject(self, iteration: int, loss=None):
        """Call once per train iteration.  Returns a possibly-poisoned loss."""
        if not self.should_fire(iteration):
            return loss
        self.fired = True
        kind = self.config.fault_type
        if kind == "crash":
            raise InjectedFault(f"injected crash on rank {self.rank} at iteration {iteration}")
        if kind == "hang":
            time.sleep(self.config.hang_seconds)
            return loss
        if kind == "nan_loss":
            import torch

            if loss is not None:
                return loss * torch.nan
            return loss
        raise ValueError(f"unknown fault_type {kind!r}")
