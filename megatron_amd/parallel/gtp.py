"""Generalized tensor parallelism (weight-rematerialized sharding).

Capability analog of reference megatron/core/tensor_parallel/
generalized_tensor_parallelism.py (+ gtp_api.py): an axis orthogonal to TP
that shards *weights* (not activations) along out_features over a group;
every forward AND backward all-gathers the weight shards (rematerialization
— the full weight never persists), and the weight gradient is
reduce-scattered back so each rank only stores and optimizes its shard.
Unlike the distributed optimizer (which shards optimizer state only), GTP
removes the full weight itself from resident memory.

MI355X notes: on one node the gather/scatter ride all 7 xGMI links
(fully-connected, ~1 TB/s aggregate per GPU), and with 288 GB HBM3E the
right default is a LARGE gtp degree only for the few giant weights
(embeddings, MoE expert stacks), not every linear; callers opt in per-layer.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from megatron_amd.parallel import grid as G


def _default_group():
    if G.grid_initialized() and dist.is_initialized():
        return G.get_grid().group("dp")
    return None


def _gather_weight(shard: torch.Tensor, group) -> torch.Tensor:
    """all-gather [out/g, in] shards -> [out, in] (concat along dim 0)."""
    world = dist.get_world_size(group) if (dist.is_initialized() and group is not None) else 1
    if world == 1:
        return shard
    parts = [torch.empty_like(shard) for _ in range(world)]
    dist.all_gather(parts, shard.contiguous(), group=group)
    return torch.cat(parts, dim=0)


def _gather_weight_async(shard: torch.Tensor, group):
    """Launch the shard all-gather without blocking; returns (buf, handle)
    (the GTP prefetch chain's primitive; reference gtp streams :352)."""
    world = dist.get_world_size(group) if (dist.is_initialized() and group is not None) else 1
    if world == 1:
        return shard, None
    buf = torch.empty(world * shard.shape[0], *shard.shape[1:],
                      dtype=shard.dtype, device=shard.device)
    with torch.no_grad():
        h = dist.all_gather_into_tensor(buf, shard.contiguous(), group=group, async_op=True)
    return buf, h


class _GTPLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight_shard, bias, group, module):
        full_w = module._take_prefetched() if module is not None else None
        if full_w is None:
            full_w = _gather_weight(weight_shard, group)
        ctx.save_for_backward(x, weight_shard)
        ctx.group = group
        ctx.module = module
        out = torch.matmul(x, full_w.t())
        if bias is not None:
            out = out + bias
        ctx.has_bias = bias is not None
        return out

    @staticmethod
    def backward(ctx, dy):
        x, weight_shard = ctx.saved_tensors
        group = ctx.group
        full_w = ctx.module._take_prefetched() if ctx.module is not None else None
        if full_w is None:
            full_w = _gather_weight(weight_shard, group)  # rematerialize
        dx = torch.matmul(dy, full_w)
        dy2 = dy.reshape(-1, dy.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        dw_full = torch.matmul(dy2.t(), x2)  # [out, in]
        world = dist.get_world_size(group) if (dist.is_initialized() and group is not None) else 1
        if world > 1:
            backend = dist.get_backend(group)
            if backend == "gloo":  # gloo has no reduce-scatter: all-reduce + slice
                dist.all_reduce(dw_full, group=group)
                r = dist.get_rank(group)
                n = weight_shard.shape[0]
                dw_shard = dw_full[r * n:(r + 1) * n].contiguous()
            else:
                dw_shard = torch.empty_like(weight_shard)
                dist.reduce_scatter_tensor(dw_shard, dw_full.contiguous(), group=group)
        else:
            dw_shard = dw_full
        db = dy2.sum(dim=0) if ctx.has_bias else None
        return dx, dw_shard, db, None, None


class GTPLinear(nn.Module):
    """Linear with the weight sharded along out_features over `group`.

    The full [out, in] weight exists only transiently inside fwd/bwd.
    The bias (small) is replicated; its grad is averaged by DDP as usual.
    """

    def __init__(self, in_features: int, out_features: int, bias: bool = True,
                 group=None, dtype=torch.float32, init_method=None):
        super().__init__()
        self.group = group if group is not None else _default_group()
        world = dist.get_world_size(self.group) if (dist.is_initialized() and self.group is not None) else 1
        rank = dist.get_rank(self.group) if (dist.is_initialized() and self.group is not None) else 0
        assert out_features % world == 0, (out_features, world)
        self.in_features = in_features
        self.out_features = out_features
        self.shard_rows = out_features // world
        self.gtp_rank = rank
        self.weight = nn.Parameter(torch.empty(self.shard_rows, in_features, dtype=dtype))
        self.weight.gtp_sharded = True  # checkpoint/DDP: not DP-replicated
        with torch.no_grad():
            if init_method is not None:
                # init the full weight identically on all ranks, keep our rows
                full = torch.empty(out_features, in_features, dtype=dtype)
                init_method(full)
                self.weight.copy_(full[rank * self.shard_rows:(rank + 1) * self.shard_rows])
            else:
                nn.init.kaiming_uniform_(self.weight, a=5 ** 0.5)
        self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype)) if bias else None

        self._prefetched = None  # (buf, handle) set by a GTPChain

    def prefetch(self):
        """Launch this layer's weight all-gather ahead of time (chain)."""
        if self._prefetched is None:
            self._prefetched = _gather_weight_async(self.weight, self.group)

    def _take_prefetched(self):
        if self._prefetched is None:
            return None
        buf, h = self._prefetched
        self._prefetched = None
        if h is not None:
            with torch.no_grad():
                h.wait()  # gloo completes chunk-view copies inside wait
        return buf

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return _GTPLinearFn.apply(x, self.weight, self.bias, self.group, self)


class GTPChain:
    """Weight-gather prefetch chain over GTP layers in execution order
    (reference generalized_tensor_parallelism chains :103): when layer i
    starts its forward, layer i+1's all-gather is already in flight (and
    layer i-1's during backward), hiding gather latency under the GEMMs."""

    def __init__(self, layers):
        self.layers = list(layers)
        for i, m in enumerate(self.layers):
            m.register_forward_pre_hook(self._fwd_pre(i))
            m.register_full_backward_pre_hook(self._bwd_pre(i))

    def _fwd_pre(self, i):
        def hook(mod, args):
            mod.prefetch()
            if i + 1 < len(self.layers):
                self.layers[i + 1].prefetch()
        return hook

    def _bwd_pre(self, i):
        def hook(mod, gout):
            mod.prefetch()
            if i - 1 >= 0:
                self.layers[i - 1].prefetch()
        return hook
