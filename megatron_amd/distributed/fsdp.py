"""Fully-sharded data parallelism (ZeRO-3): param, grad, and optimizer-state
sharding over the DP group.

Capability analog of reference megatron/core/distributed/fsdp/src/
megatron_fsdp/ (fully_shard.py, megatron_fsdp.py, param_and_grad_buffer.py
— ~12.8k LoC built on DTensor): params live as ONE flat local shard per
unit; full weights exist only transiently around each unit's compute.

MI355X-first design decisions (vs the reference's NVSwitch assumptions):
  * xGMI is 7 point-to-point links -> all-gather/reduce-scatter are
    per-link bound; units are whole transformer layers (few, large
    collectives) instead of per-weight sharding.
  * 288 GB HBM3E -> default ``reshard_after_forward=False`` keeps gathered
    weights resident between fwd and bwd (ZeRO-2-like memory, ZeRO-3 comm
    only once per step); flip it on only for models that do not fit.
  * fp32 master shard + fp32 grad shard are owned here, so the optimizer
    runs on 1/dp of the state with no extra copies (the grad
    reduce-scatter lands directly in fp32).

Flow per unit U:
  fwd pre-hook:  all-gather shard -> flat buffer, params become views
  fwd post-hook: optionally drop the flat buffer (reshard_after_forward)
  bwd pre-hook:  re-gather if dropped
  last param grad accumulated (post_accumulate_grad_hook): reduce-scatter
    grads (pre-scaled 1/dp, SUM) into the fp32 grad shard, free full grads
    and the flat buffer.

Optimizer contract: ``fsdp.shard_parameters()`` yields one fp32 Parameter
per unit (with ``.grad`` filled after backward); after ``optimizer.step()``
call ``fsdp.update_model_shards()`` to cast the fp32 master back into the
model-dtype shard that the next all-gather broadcasts.
"""

from __future__ import annotations

import math
from contextlib import contextmanager
from typing import Dict, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from megatron_amd.parallel import grid as G


def _pad_to(x: int, align: int) -> int:
    return int(math.ceil(x / align) * align) if align > 1 else x


class _FSDPUnit:
    """One sharding unit: a module whose params share a flat buffer."""

    def __init__(self, name: str, module: nn.Module, params: List[nn.Parameter],
                 dp_group, reshard_after_forward: bool, grad_scale_denom: int = 0):
        self.name = name
        self.module = module
        self.params = params
        self.dp_group = dp_group
        if dist.is_initialized():
            self.dp_size = dist.get_world_size(dp_group)
            self.dp_rank = dist.get_rank(dp_group)
        else:
            self.dp_size, self.dp_rank = 1, 0
        # expert units reduce-scatter over edp but must average over dp_cp so
        # expert grads carry the same per-token scale as dense grads (same
        # rule as ParamAndGradBuffer.grad_scale_denom)
        self.grad_scale_denom = grad_scale_denom or self.dp_size
        self.reshard_after_forward = reshard_after_forward

        self.dtype = params[0].dtype
        device = params[0].device
        numel = sum(p.numel() for p in params)
        self.flat_size = _pad_to(numel, self.dp_size)
        self.shard_size = self.flat_size // self.dp_size

        # offsets / sizes of each param in the flat buffer (param .data gets
        # replaced by 0-size stubs while sharded, so sizes must be cached)
        self.offsets: List[int] = []
        self.numels: List[int] = [p.numel() for p in params]
        off = 0
        for n in self.numels:
            self.offsets.append(off)
            off += n

        # build the flat buffer once, sync to the group's rank-0 weights
        # (per-rank random init must agree before sharding), keep our shard
        flat = torch.zeros(self.flat_size, dtype=self.dtype, device=device)
        for p, o in zip(params, self.offsets):
            flat[o : o + p.numel()].copy_(p.detach().reshape(-1))
        if dist.is_initialized() and self.dp_size > 1:
            src = dist.get_process_group_ranks(dp_group)[0] if dp_group is not None else 0
            dist.broadcast(flat, src=src, group=dp_group)
        shard_view = flat[self.dp_rank * self.shard_size : (self.dp_rank + 1) * self.shard_size]
        self.model_shard = shard_view.clone()  # model-dtype shard (AG source)
        # fp32 master shard exposed to the optimizer
        self.master_shard = nn.Parameter(self.model_shard.float())
        self.master_shard.fsdp_unit = name
        self.grad_shard = torch.zeros_like(self.master_shard, dtype=torch.float32)

        self._param_shapes = [p.shape for p in params]
        self._flat: Optional[torch.Tensor] = None
        self._grads_ready = 0
        self._ag_handle = None
        self.is_last_microbatch = True
        self._stub = torch.empty(0, dtype=self.dtype, device=device)
        self._set_stub_views()

    # ---- param materialization ----------------------------------------

    def _set_flat_views(self):
        for p, o, n, shp in zip(self.params, self.offsets, self.numels, self._param_shapes):
            p.data = self._flat[o : o + n].view(shp)

    def _set_stub_views(self):
        for p in self.params:
            p.data = self._stub

    def unshard(self, async_op: bool = False):
        if self._flat is not None:
            if self._ag_handle is not None and not async_op:
                self.finish_unshard()  # a prefetch is in flight: just wait
            return
        self._flat = torch.empty(self.flat_size, dtype=self.dtype, device=self.model_shard.device)
        if self.dp_size == 1:
            self._flat.copy_(self.model_shard)
            if not async_op:
                self._set_flat_views()
            else:
                self._ag_handle = None
                self._set_flat_views()
            return
        h = dist.all_gather_into_tensor(self._flat, self.model_shard, group=self.dp_group,
                                        async_op=async_op)
        if async_op:
            self._ag_handle = h
        else:
            self._set_flat_views()

    def finish_unshard(self):
        if self._ag_handle is not None:
            self._ag_handle.wait()
            self._ag_handle = None
            self._set_flat_views()

    def reshard(self):
        if self._flat is None:
            return
        if self._ag_handle is not None:
            # never free a buffer an async gather may still be writing
            self._ag_handle.wait()
            self._ag_handle = None
        self._set_stub_views()
        self._flat = None

    # ---- grad path ------------------------------------------------------

    def on_param_grad(self, param: nn.Parameter):
        self._grads_ready += 1
        if self._grads_ready < sum(1 for p in self.params if p.requires_grad):
            return
        self._grads_ready = 0
        # pack grads into a flat fp32 buffer, pre-scaled for DP averaging
        flat_grad = torch.zeros(self.flat_size, dtype=torch.float32,
                                device=self.model_shard.device)
        inv = 1.0 / self.grad_scale_denom
        for p, o, n in zip(self.params, self.offsets, self.numels):
            if p.grad is not None:
                flat_grad[o : o + n].copy_(p.grad.reshape(-1)).mul_(inv)
                p.grad = None
        if self.dp_size == 1:
            self.grad_shard += flat_grad
        else:
            tmp = torch.empty_like(self.grad_shard)
            dist.reduce_scatter_tensor(tmp, flat_grad, group=self.dp_group)
            self.grad_shard += tmp  # accumulates across microbatches
        if self.is_last_microbatch:
            self.master_shard.grad = self.grad_shard
        self.reshard()

    def update_model_shard(self):
        self.model_shard.copy_(self.master_shard.detach().to(self.dtype))

    def zero_grad(self):
        self.grad_shard.zero_()
        self.master_shard.grad = None


class FullyShardedDataParallel(nn.Module):
    """Wraps a model; shards every leaf-module's params over DP.

    ``unit_modules``: module classes that form sharding units (default:
    any direct child of a ModuleList — i.e. one unit per transformer
    layer); params not claimed by a unit are grouped into a root unit.
    """

    def __init__(self, module: nn.Module, unit_classes: tuple = (), dp_group=None,
                 reshard_after_forward: bool = False):
        super().__init__()
        self.module = module
        self.dp_group = (dp_group if dp_group is not None
                         else (G.get_data_parallel_group() if dist.is_initialized() else None))
        # expert-parallel params are sharded over EP already; their FSDP flat
        # shard lives on the expert-data-parallel (edp) replica group, with
        # grads averaged over dp_cp like the dense ones
        grid_ok = dist.is_initialized() and G.grid_initialized()
        self.edp_group = G.get_grid().group("expert_dp") if grid_ok else None
        dp_cp = (dist.get_world_size(self.dp_group)
                 if (self.dp_group is not None and dist.is_initialized()) else 1)
        self.units: List[_FSDPUnit] = []
        self._param_to_unit: Dict[nn.Parameter, _FSDPUnit] = {}

        claimed = set()
        seen_params = set()

        def make_unit(name, mod, params):
            params = [p for p in params if id(p) not in seen_params]
            if not params:
                return
            seen_params.update(id(p) for p in params)
            dense = [p for p in params if not getattr(p, "is_expert_parallel", False)]
            expert = [p for p in params if getattr(p, "is_expert_parallel", False)]
            if dense:
                u = _FSDPUnit(name, mod, dense, self.dp_group, reshard_after_forward)
                self.units.append(u)
                for p in dense:
                    self._param_to_unit[p] = u
            if expert:
                u = _FSDPUnit(name + ".experts", mod, expert, self.edp_group,
                              reshard_after_forward, grad_scale_denom=dp_cp)
                self.units.append(u)
                for p in expert:
                    self._param_to_unit[p] = u

        for name, mod in module.named_modules():
            is_unit = isinstance(mod, unit_classes) if unit_classes else False
            if not unit_classes:
                # default: one unit per element of any ModuleList (layers)
                parent_is_list = "." in name and isinstance(
                    module.get_submodule(name.rsplit(".", 1)[0]), nn.ModuleList
                )
                is_unit = parent_is_list
            if is_unit and name not in claimed:
                claimed.add(name)
                make_unit(name, mod, list(mod.parameters()))
        # everything else (embeddings, final norm, output layer) -> root unit
        make_unit("root", module, [p for p in module.parameters() if id(p) not in seen_params])

        self._register_hooks()

    # ---- setup ----------------------------------------------------------

    def _register_hooks(self):
        for u in self.units:
            u.module.register_forward_pre_hook(self._fwd_pre(u))
            u.module.register_forward_hook(self._fwd_post(u))
            u.module.register_full_backward_pre_hook(self._bwd_pre(u))
            for p in u.params:
                if p.requires_grad:
                    p.register_post_accumulate_grad_hook(self._grad_hook(u))

    def _fwd_pre(self, u):
        idx = self.units.index(u)

        def hook(mod, args):
            u.unshard()
            # AG-prefetch pipeline (reference megatron_fsdp overlap): kick
            # the NEXT unit's all-gather so it overlaps this unit's compute
            # (no-op at dp=1: nothing to gather, skip the async churn)
            if idx + 1 < len(self.units) and u.dp_size > 1:
                self.units[idx + 1].unshard(async_op=True)
        return hook

    def _fwd_post(self, u):
        def hook(mod, args, out):
            if u.reshard_after_forward and self.training:
                u.reshard()
            if not self.training:
                u.reshard()
        return hook

    def _bwd_pre(self, u):
        idx = self.units.index(u)

        def hook(mod, grad_out):
            u.unshard()
            if idx - 1 >= 0 and u.dp_size > 1:
                self.units[idx - 1].unshard(async_op=True)  # bwd runs in reverse
        return hook

    def _grad_hook(self, u):
        def hook(p):
            u.on_param_grad(p)
        return hook

    # ---- public API ------------------------------------------------------

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def shard_parameters(self):
        """fp32 master shards — feed these to the optimizer."""
        return [u.master_shard for u in self.units]

    def update_model_shards(self):
        for u in self.units:
            u.update_model_shard()

    def zero_grad_buffer(self):
        for u in self.units:
            u.zero_grad()

    @contextmanager
    def no_last_microbatch(self):
        """During grad accumulation: delay exposing .grad until the last
        microbatch (reduce-scatter still runs each microbatch — grads stay
        sharded at all times)."""
        for u in self.units:
            u.is_last_microbatch = False
        try:
            yield
        finally:
            for u in self.units:
                u.is_last_microbatch = True

    def clip_grad_norm(self, max_norm: float) -> torch.Tensor:
        """Global grad-norm over all shards (each element counted once)."""
        sq = torch.zeros((), dtype=torch.float32,
                         device=self.units[0].model_shard.device)
        for u in self.units:
            sq += u.grad_shard.float().pow(2).sum()
        if dist.is_initialized():
            dist.all_reduce(sq, group=self.dp_group)
        norm = sq.sqrt()
        scale = max_norm / (norm + 1e-6)
        if scale < 1.0:
            for u in self.units:
                u.grad_shard.mul_(scale)
        return norm

    def state_dict(self, *args, **kwargs):  # full (unsharded) state dict
        for u in self.units:
            u.unshard()
        sd = self.module.state_dict(*args, **kwargs)
        sd = {k: v.clone() for k, v in sd.items()}
        for u in self.units:
            u.reshard()
        return sd
