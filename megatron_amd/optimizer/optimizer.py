"""Non-distributed optimizers + the chained wrapper.

Capability analog of reference megatron/core/optimizer/optimizer.py
(MegatronOptimizer :187, MixedPrecisionOptimizer :654,
Float16OptimizerWithFloat16Params :964, FP32Optimizer :1232,
ChainedOptimizer :1419).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

from megatron_amd import ops
from megatron_amd.config import OptimizerConfig
from megatron_amd.optimizer.clip import (
    split_grads_for_norm,
    clip_grads_by_total_norm,
    get_grad_norm,
    param_is_not_tensor_parallel_duplicate,
)


def _model_chunks_params(model_chunks, param_filter=None) -> List[torch.nn.Parameter]:
    params = []
    for chunk in model_chunks:
        for p in chunk.parameters():
            if p.requires_grad and (param_filter is None or param_filter(p)):
                params.append(p)
    return params


def _wd_group(param, name_hint: str = "") -> bool:
    """True -> apply weight decay.  Convention: no decay for 1-D params
    (norm weights, biases)."""
    return param.dim() > 1


def apply_param_update(config, mains, grads, exp_avg, exp_avg_sq, lr, wd, step,
                       model_params_bf16=None):
    """One param-group update: fused AdamW (K10) or SGD-with-momentum,
    chosen by config.optimizer (reference supports both)."""
    if config.optimizer == "sgd":
        if wd:
            torch._foreach_add_(grads, mains, alpha=wd)
        torch._foreach_mul_(exp_avg, config.sgd_momentum)
        torch._foreach_add_(exp_avg, grads)
        torch._foreach_add_(mains, exp_avg, alpha=-lr)
        if model_params_bf16 is not None:
            for mp, p in zip(mains, model_params_bf16):
                p.copy_(mp.to(p.dtype))
        return
    ops.fused_adamw(
        mains, grads, exp_avg, exp_avg_sq,
        lr, config.adam_beta1, config.adam_beta2, config.adam_eps, wd, step,
        **({"model_params_bf16": model_params_bf16} if model_params_bf16 is not None else {}),
    )


class _BaseOptimizer:
    # scheduler lr is multiplied by this (decoupled-lr groups, reference
    # optimizer/__init__.py lr_mult for embedding/output params)
    lr_ratio: float = 1.0
    param_filter = None

    def __init__(self, config: OptimizerConfig, model_chunks: List):
        self.config = config
        self.model_chunks = model_chunks
        self.step_count = 0
        self._lr = config.lr * self.lr_ratio
        self._wd = config.weight_decay

    # scheduler interface
    def set_lr(self, lr: float):
        self._lr = lr * self.lr_ratio

    def set_wd(self, wd: float):
        self._wd = wd

    def get_lr(self) -> float:
        return self._lr

    # chained interface
    def zero_grad(self):
        for chunk in self.model_chunks:
            if hasattr(chunk, "zero_grad_buffer"):
                chunk.zero_grad_buffer()
            else:
                chunk.zero_grad(set_to_none=True)

    def finish_grad_sync(self):
        for chunk in self.model_chunks:
            if hasattr(chunk, "finish_grad_sync"):
                chunk.finish_grad_sync()

    def step(self) -> Tuple[bool, Optional[torch.Tensor], Optional[int]]:
        raise NotImplementedError

    def state_dict(self) -> dict:
        raise NotImplementedError

    def load_state_dict(self, sd: dict):
        raise NotImplementedError


class FP32Optimizer(_BaseOptimizer):
    """Plain fp32 AdamW over main_grad buffers (CPU tests + fp32 runs)."""

    def __init__(self, config: OptimizerConfig, model_chunks: List):
        super().__init__(config, model_chunks)
        self.params = _model_chunks_params(model_chunks, self.param_filter)
        self.exp_avg = [torch.zeros_like(p, dtype=torch.float32) for p in self.params]
        self.exp_avg_sq = [torch.zeros_like(p, dtype=torch.float32) for p in self.params]

    def _grads(self):
        gs = []
        for p in self.params:
            g = getattr(p, "main_grad", None)
            if g is None:
                g = p.grad if p.grad is not None else torch.zeros_like(p)
            gs.append(g.float())
        return gs

    @torch.no_grad()
    def step(self):
        self.finish_grad_sync()
        grads = self._grads()
        dense_g, expert_g = split_grads_for_norm(self.params, grads)
        total_norm = get_grad_norm(dense_g, expert_grads=expert_g)
        if self.config.clip_grad > 0:
            clip_grads_by_total_norm(grads, self.config.clip_grad, total_norm)
        self.step_count += 1
        decay_mask = [_wd_group(p) for p in self.params]
        for apply_wd in (True, False):
            idx = [i for i, m in enumerate(decay_mask) if m == apply_wd]
            if not idx:
                continue
            apply_param_update(
                self.config,
                [self.params[i].data for i in idx],
                [grads[i] for i in idx],
                [self.exp_avg[i] for i in idx],
                [self.exp_avg_sq[i] for i in idx],
                self._lr,
                self._wd if apply_wd else 0.0,
                self.step_count,
            )
        return True, total_norm, None

    def state_dict(self):
        return {
            "step": self.step_count,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        for a, b in zip(self.exp_avg, sd["exp_avg"]):
            a.copy_(b)
        for a, b in zip(self.exp_avg_sq, sd["exp_avg_sq"]):
            a.copy_(b)


class MixedPrecisionOptimizer(_BaseOptimizer):
    """bf16/fp16 model params with fp32 main copies (non-distributed ZeRO-0).

    Reference: Float16OptimizerWithFloat16Params optimizer.py:964.
    """

    def __init__(self, config: OptimizerConfig, model_chunks: List):
        super().__init__(config, model_chunks)
        self.params = _model_chunks_params(model_chunks, self.param_filter)
        self.main_params = [p.detach().clone().float() for p in self.params]
        self.exp_avg = [torch.zeros_like(mp) for mp in self.main_params]
        self.exp_avg_sq = [torch.zeros_like(mp) for mp in self.main_params]
        # fp16 dynamic loss scale
        self.grad_scaler = None
        if config.fp16:
            from megatron_amd.optimizer.grad_scaler import DynamicGradScaler

            self.grad_scaler = DynamicGradScaler(config)

    def reload_model_params(self):
        for mp, p in zip(self.main_params, self.params):
            mp.copy_(p.detach().float())

    @torch.no_grad()
    def step(self):
        self.finish_grad_sync()
        grads = []
        for p in self.params:
            g = getattr(p, "main_grad", None)
            if g is None:
                g = p.grad if p.grad is not None else torch.zeros_like(p)
            grads.append(g.float())
        if self.grad_scaler is not None:
            inv = 1.0 / self.grad_scaler.scale
            torch._foreach_mul_(grads, inv)
            found_inf = torch.zeros((), device=grads[0].device)
            for g in grads:
                found_inf = torch.maximum(found_inf, (~torch.isfinite(g)).any().float())
            if dist.is_initialized():
                from megatron_amd.parallel import grid as G

                if G.grid_initialized():
                    dist.all_reduce(found_inf, op=dist.ReduceOp.MAX, group=G.get_grid().group("mp"))
            self.grad_scaler.update(found_inf.item() > 0)
            if found_inf.item() > 0:
                return False, None, None
        dense_g, expert_g = split_grads_for_norm(self.params, grads)
        total_norm = get_grad_norm(dense_g, expert_grads=expert_g)
        if self.config.clip_grad > 0:
            clip_grads_by_total_norm(grads, self.config.clip_grad, total_norm)
        self.step_count += 1
        decay_mask = [_wd_group(p) for p in self.params]
        for apply_wd in (True, False):
            idx = [i for i, m in enumerate(decay_mask) if m == apply_wd]
            if not idx:
                continue
            apply_param_update(
                self.config,
                [self.main_params[i] for i in idx],
                [grads[i] for i in idx],
                [self.exp_avg[i] for i in idx],
                [self.exp_avg_sq[i] for i in idx],
                self._lr,
                self._wd if apply_wd else 0.0,
                self.step_count,
                model_params_bf16=[self.params[i].data for i in idx],
            )
        return True, total_norm, None

    def state_dict(self):
        return {
            "step": self.step_count,
            "main_params": self.main_params,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        for dst, src in zip(
            self.main_params + self.exp_avg + self.exp_avg_sq,
            sd["main_params"] + sd["exp_avg"] + sd["exp_avg_sq"],
        ):
            dst.copy_(src)
        for p, mp in zip(self.params, self.main_params):
            p.data.copy_(mp.to(p.dtype))


class ChainedOptimizer:
    """Chains sub-optimizers (dense / expert / ...): reference optimizer.py:1419."""

    def __init__(self, optimizers: List[_BaseOptimizer]):
        self.chained_optimizers = optimizers

    def zero_grad(self):
        for o in self.chained_optimizers:
            o.zero_grad()

    @torch.no_grad()
    def step(self):
        ok, norms = True, []
        for o in self.chained_optimizers:
            success, norm, _ = o.step()
            ok = ok and success
            if norm is not None:
                norms.append(norm)
        total = torch.sqrt(sum(n * n for n in norms)) if norms else None
        return ok, total, None

    def set_lr(self, lr):
        for o in self.chained_optimizers:
            o.set_lr(lr)

    def set_wd(self, wd):
        for o in self.chained_optimizers:
            o.set_wd(wd)

    def get_lr(self):
        return self.chained_optimizers[0].get_lr()

    def reload_model_params(self):
        for o in self.chained_optimizers:
            if hasattr(o, "reload_model_params"):
                o.reload_model_params()

    def start_param_sync(self):
        for o in self.chained_optimizers:
            if hasattr(o, "start_param_sync"):
                o.start_param_sync()

    def state_dict(self):
        return [o.state_dict() for o in self.chained_optimizers]

    def load_state_dict(self, sds):
        for o, sd in zip(self.chained_optimizers, sds):
            o.load_state_dict(sd)
