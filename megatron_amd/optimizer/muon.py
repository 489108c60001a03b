"""Muon optimizer: orthogonalized-momentum updates for 2-D weight matrices.

Capability analog of reference megatron/core/optimizer/muon.py (+
emerging_optimizers.py): momentum SGD whose update direction is
orthogonalized with a quintic Newton-Schulz iteration, applied to 2-D
parameters (attention / MLP weight matrices); embeddings, output layer,
norms and biases fall back to AdamW.  Distributed semantics match the
non-sharded optimizers here: each rank runs Muon on its own (TP-sharded)
main_grad, which is the reference's behavior as well.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from megatron_amd import ops
from megatron_amd.config import OptimizerConfig
from megatron_amd.optimizer.clip import (
    split_grads_for_norm,
    clip_grads_by_total_norm,
    get_grad_norm,
    param_is_not_tensor_parallel_duplicate,
)
from megatron_amd.optimizer.optimizer import _BaseOptimizer, _model_chunks_params, _wd_group


@torch.no_grad()
def newton_schulz_orthogonalize(g: torch.Tensor, steps: int = 5, eps: float = 1e-7) -> torch.Tensor:
    """Quintic Newton-Schulz iteration approximating UV^T of the SVD of `g`.

    Coefficients (3.4445, -4.7750, 2.0315) are the standard Muon quintic with
    spectral radius tuned for 5 iterations.  Runs in fp32 (bf16 is fine on GPU
    but tests compare on CPU).
    """
    assert g.dim() == 2
    a, b, c = 3.4445, -4.7750, 2.0315
    X = g.float()
    transposed = X.shape[0] > X.shape[1]
    if transposed:
        X = X.T
    X = X / (X.norm() + eps)
    for _ in range(steps):
        A = X @ X.T
        B = b * A + c * (A @ A)
        X = a * X + B @ X
    if transposed:
        X = X.T
    return X


def muon_param(p: torch.nn.Parameter) -> bool:
    """Muon applies to genuine weight matrices only: 2-D, not embeddings or
    the output projection (flagged `muon_exclude` by the model, or vocab-sized)."""
    if p.dim() != 2:
        return False
    if getattr(p, "muon_exclude", False):
        return False
    return True


class MuonOptimizer(_BaseOptimizer):
    """Muon for 2-D weights + AdamW for everything else, over main_grad
    buffers with fp32 main params (mixed-precision-safe)."""

    def __init__(self, config: OptimizerConfig, model_chunks: List):
        super().__init__(config, model_chunks)
        self.params = _model_chunks_params(model_chunks)
        self.main_params = [p.detach().clone().float() for p in self.params]
        self.is_muon = [muon_param(p) for p in self.params]
        self.momentum = config.muon_momentum
        self.ns_steps = config.muon_ns_steps
        # muon params get one momentum buffer; adamw params get exp_avg(_sq)
        self.muon_buf = [torch.zeros_like(mp) if m else None for mp, m in zip(self.main_params, self.is_muon)]
        self.exp_avg = [None if m else torch.zeros_like(mp) for mp, m in zip(self.main_params, self.is_muon)]
        self.exp_avg_sq = [None if m else torch.zeros_like(mp) for mp, m in zip(self.main_params, self.is_muon)]

    def _grads(self):
        gs = []
        for p in self.params:
            g = getattr(p, "main_grad", None)
            if g is None:
                g = p.grad if p.grad is not None else torch.zeros_like(p)
            gs.append(g.float())
        return gs

    @torch.no_grad()
    def step(self) -> Tuple[bool, Optional[torch.Tensor], Optional[int]]:
        self.finish_grad_sync()
        grads = self._grads()
        dense_g, expert_g = split_grads_for_norm(self.params, grads)
        total_norm = get_grad_norm(dense_g, expert_grads=expert_g)
        if self.config.clip_grad > 0:
            clip_grads_by_total_norm(grads, self.config.clip_grad, total_norm)
        self.step_count += 1

        # --- Muon branch ---
        for i, is_m in enumerate(self.is_muon):
            if not is_m:
                continue
            buf = self.muon_buf[i]
            buf.mul_(self.momentum).add_(grads[i])
            update = grads[i].add(buf, alpha=self.momentum)  # nesterov
            O = newton_schulz_orthogonalize(update, steps=self.ns_steps)
            n, m = self.main_params[i].shape
            scale = max(1.0, n / m) ** 0.5
            mp = self.main_params[i]
            if self._wd > 0 and _wd_group(self.params[i]):
                mp.mul_(1.0 - self._lr * self._wd)
            mp.add_(O, alpha=-self._lr * scale)
            self.params[i].data.copy_(mp.to(self.params[i].dtype))

        # --- AdamW branch ---
        decay_mask = [_wd_group(p) for p in self.params]
        for apply_wd in (True, False):
            idx = [i for i, m in enumerate(self.is_muon) if not m and decay_mask[i] == apply_wd]
            if not idx:
                continue
            ops.fused_adamw(
                [self.main_params[i] for i in idx],
                [grads[i] for i in idx],
                [self.exp_avg[i] for i in idx],
                [self.exp_avg_sq[i] for i in idx],
                self._lr,
                self.config.adam_beta1,
                self.config.adam_beta2,
                self.config.adam_eps,
                self._wd if apply_wd else 0.0,
                self.step_count,
                model_params_bf16=[self.params[i].data for i in idx],
            )
        return True, total_norm, None

    def reload_model_params(self):
        for mp, p in zip(self.main_params, self.params):
            mp.copy_(p.detach().float())

    def state_dict(self):
        return {
            "step": self.step_count,
            "main_params": self.main_params,
            "muon_buf": self.muon_buf,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        for dst, src in zip(self.main_params, sd["main_params"]):
            dst.copy_(src)
        for dst, src in zip(self.muon_buf, sd["muon_buf"]):
            if dst is not None:
                dst.copy_(src)
        for dst, src in zip(self.exp_avg, sd["exp_avg"]):
            if dst is not None:
                dst.copy_(src)
        for dst, src in zip(self.exp_avg_sq, sd["exp_avg_sq"]):
            if dst is not None:
                dst.copy_(src)
        for p, mp in zip(self.params, self.main_params):
            p.data.copy_(mp.to(p.dtype))
