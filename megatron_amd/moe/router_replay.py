"""Router trace / replay (MoE determinism debugging).

Capability analog of the reference's router replay & trace tooling
(moe/ per-layer logging + router replay determinism tools): record every
router's top-k decisions (and probs) for a window of steps; in replay mode
the recorded indices are forced back onto the router, pinning the routing
while weights/numerics vary — the standard way to bisect "did routing or
math diverge" between two runs or two machines.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch


class RouterTraceRecorder:
    """Wraps every TopKRouter in `model`; call step() once per iteration."""

    def __init__(self, model: torch.nn.Module):
        from megatron_amd.moe.router import TopKRouter

        self.routers: List[TopKRouter] = [m for m in model.modules() if isinstance(m, TopKRouter)]
        self.trace: List[List[dict]] = []  # [step][layer] -> {indices, probs}
        self._current: List[Optional[dict]] = [None] * len(self.routers)
        self._hooks = []
        for i, r in enumerate(self.routers):
            self._hooks.append(r.register_forward_hook(self._make_hook(i)))

    def _make_hook(self, i):
        def hook(module, inputs, output):
            probs, idx = output
            self._current[i] = {"indices": idx.detach().cpu().clone(),
                                "probs": probs.detach().cpu().clone()}
        return hook

    def step(self):
        self.trace.append([c for c in self._current])
        self._current = [None] * len(self.routers)

    def save(self, path: str):
        torch.save(self.trace, path)

    def close(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []


class RouterReplayer:
    """Forces recorded routing decisions: router outputs (probs from the
    live forward, indices from the trace) so routing is pinned while the
    math under test still runs."""

    def __init__(self, model: torch.nn.Module, trace: List[List[dict]]):
        from megatron_amd.moe.router import TopKRouter

        self.routers = [m for m in model.modules() if isinstance(m, TopKRouter)]
        self.trace = trace
        self.step_idx = 0
        self._hooks = []
        for i, r in enumerate(self.routers):
            self._hooks.append(r.register_forward_hook(self._make_hook(i)))

    def _make_hook(self, i):
        def hook(module, inputs, output):
            probs, _ = output
            rec = self.trace[self.step_idx][i]
            idx = rec["indices"].to(probs.device)
            # re-gather the live probs at the recorded indices is not
            # possible post-topk; use recorded probs for full determinism
            return rec["probs"].to(probs.device).to(probs.dtype), idx
        return hook

    def step(self):
        self.step_idx += 1

    def close(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []


def routing_divergence(trace_a: List[List[dict]], trace_b: List[List[dict]]) -> Dict[str, float]:
    """Fraction of (step, layer, token, slot) routing decisions that differ."""
    total, diff = 0, 0
    first = None
    for s, (sa, sb) in enumerate(zip(trace_a, trace_b)):
        for l, (ra, rb) in enumerate(zip(sa, sb)):
            ia, ib = ra["indices"], rb["indices"]
            total += ia.numel()
            d = int((ia != ib).sum())
            diff += d
            if d and first is None:
                first = (s, l)
    return {"fraction_diverged": diff / max(total, 1),
            "first_divergence": first, "total_decisions": total}
