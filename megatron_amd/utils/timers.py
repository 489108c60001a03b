"""Named timers with cross-rank max reporting.

Capability analog of reference megatron/core/timers.py:109 (Timers with
log_level and barriered elapsed, max/minmax over ranks).
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional

import torch
import torch.distributed as dist


class _Timer:
    def __init__(self, name: str):
        self.name = name
        self._start: Optional[float] = None
        self.elapsed_total = 0.0
        self.count = 0

    def start(self, barrier: bool = False):
        if barrier and dist.is_initialized():
            dist.barrier()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        self._start = time.perf_counter()

    def stop(self, barrier: bool = False):
        if barrier and dist.is_initialized():
            dist.barrier()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        if self._start is not None:
            self.elapsed_total += time.perf_counter() - self._start
            self.count += 1
            self._start = None

    def elapsed(self, reset: bool = True) -> float:
        e = self.elapsed_total
        if reset:
            self.elapsed_total = 0.0
            self.count = 0
        return e


class Timers:
    def __init__(self, log_level: int = 0):
        self.timers: Dict[str, _Timer] = {}
        self.log_level = log_level

    def __call__(self, name: str, log_level: int = 0) -> _Timer:
        if name not in self.timers:
            self.timers[name] = _Timer(name)
        return self.timers[name]

    def log(self, names: Optional[List[str]] = None, reset: bool = True, normalizer: float = 1.0) -> str:
        names = names or list(self.timers)
        parts = []
        for n in names:
            if n in self.timers:
                e = self.timers[n].elapsed(reset=reset) / normalizer
                if dist.is_initialized():
                    t = torch.tensor([e])
                    dist.all_reduce(t, op=dist.ReduceOp.MAX)
                    e = float(t.item())
                parts.append(f"{n}: {e*1000:.1f}ms")
        return " | ".join(parts)
