"""Per-layer MoE routing statistics.

Capability analog of reference megatron/core/transformer/moe/moe_logging.py:
accumulates each router's per-step token counts and aux-loss values, and
reports per-layer load-balance metrics (max-violation = max_load/mean_load,
expert utilization, aux/z losses) for the training log."""

from __future__ import annotations

from typing import Dict, List

import torch


class MoEStatsTracker:
    def __init__(self, model: torch.nn.Module):
        from megatron_amd.moe.router import TopKRouter

        self.routers: List[TopKRouter] = [m for m in model.modules() if isinstance(m, TopKRouter)]
        self._counts = [torch.zeros_like(r.local_tokens_per_expert) for r in self.routers]
        self._aux_sums: List[Dict[str, float]] = [dict() for _ in self.routers]
        self._steps = 0

    def collect(self):
        """Call once per step, after forward."""
        for i, r in enumerate(self.routers):
            self._counts[i] += r.local_tokens_per_expert
            for k, v in r.aux_losses.items():
                self._aux_sums[i][k] = self._aux_sums[i].get(k, 0.0) + float(v.detach() if hasattr(v, 'detach') else v)
        self._steps += 1

    def report(self, reset: bool = True) -> Dict[str, dict]:
        out: Dict[str, dict] = {}
        for i, counts in enumerate(self._counts):
            total = float(counts.sum())
            mean = total / counts.numel() if counts.numel() else 0.0
            layer = {
                "tokens_routed": total,
                "max_violation": float(counts.max()) / mean if mean > 0 else 0.0,
                "expert_utilization": float((counts > 0).float().mean()),
            }
            for k, v in self._aux_sums[i].items():
                layer[k] = v / max(self._steps, 1)
            out[f"moe_layer_{i}"] = layer
        if reset:
            for c in self._counts:
                c.zero_()
            self._aux_sums = [dict() for _ in self.routers]
            self._steps = 0
        return out
