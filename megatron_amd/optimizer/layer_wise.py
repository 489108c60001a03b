"""Layer-wise (update-in-backward) optimizer.

Capability analog of reference megatron/core/optimizer/layer_wise_optimizer.py:
each parameter's AdamW update runs the moment its gradient is accumulated
(post-accumulate-grad hook), and the gradient is freed immediately — peak
grad memory is one layer instead of the whole model.  Trade-offs match the
reference: no global grad-norm clipping (per-tensor clip instead) and no
cross-microbatch gradient accumulation (the update consumes the grad).
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from megatron_amd import ops
from megatron_amd.config import OptimizerConfig
from megatron_amd.optimizer.optimizer import _BaseOptimizer, _model_chunks_params, _wd_group


class LayerWiseOptimizer(_BaseOptimizer):
    def __init__(self, config: OptimizerConfig, model_chunks: List):
        super().__init__(config, model_chunks)
        self.params = _model_chunks_params(model_chunks)
        self.main_params = {p: p.detach().clone().float() for p in self.params}
        self.exp_avg = {p: torch.zeros_like(m) for p, m in self.main_params.items()}
        self.exp_avg_sq = {p: torch.zeros_like(m) for p, m in self.main_params.items()}
        self._armed = False
        self._sq_norm_acc = 0.0
        self._hooks = [p.register_post_accumulate_grad_hook(self._update_param)
                       for p in self.params]

    def zero_grad(self):
        super().zero_grad()
        self._armed = True  # updates fire during the upcoming backward
        self._sq_norm_acc = 0.0
        self._pending_step = self.step_count + 1

    @torch.no_grad()
    def _update_param(self, p: torch.nn.Parameter):
        if not self._armed:
            return
        g = getattr(p, "main_grad", None)
        if g is None:
            g = p.grad
        if g is None:
            return
        g = g.float()
        self._sq_norm_acc += float(g.pow(2).sum())
        if self.config.clip_grad > 0:  # per-tensor clip (no global norm exists yet)
            n = g.norm()
            if float(n) > self.config.clip_grad:
                g = g * (self.config.clip_grad / (n + 1e-6))
        mp = self.main_params[p]
        ops.fused_adamw(
            [mp], [g], [self.exp_avg[p]], [self.exp_avg_sq[p]],
            self._lr, self.config.adam_beta1, self.config.adam_beta2,
            self.config.adam_eps,
            self._wd if _wd_group(p) else 0.0, self._pending_step,
        )
        p.data.copy_(mp.to(p.dtype))
        p.grad = None  # freed immediately - the point of layer-wise updates

    def step(self) -> Tuple[bool, Optional[torch.Tensor], Optional[int]]:
        """Updates already ran inside backward; finalize bookkeeping."""
        self.finish_grad_sync()
        self.step_count += 1
        self._armed = False
        return True, torch.tensor(self._sq_norm_acc).sqrt(), None

    def reload_model_params(self):
        for p, mp in self.main_params.items():
            mp.copy_(p.detach().float())

    def state_dict(self):
        return {
            "step": self.step_count,
            "main_params": [self.main_params[p] for p in self.params],
            "exp_avg": [self.exp_avg[p] for p in self.params],
            "exp_avg_sq": [self.exp_avg_sq[p] for p in self.params],
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        for p, m, a, s in zip(self.params, sd["main_params"], sd["exp_avg"], sd["exp_avg_sq"]):
            self.main_params[p].copy_(m)
            self.exp_avg[p].copy_(a)
            self.exp_avg_sq[p].copy_(s)
            p.data.copy_(self.main_params[p].to(p.dtype))

    def close(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []
