"""Tensor-parallel region mappings (autograd collectives).

Capability analog of reference megatron/core/tensor_parallel/mappings.py
(:22 _reduce, :118 gather, :159 reduce-scatter, :201-424 autograd Functions,
:424 _AllToAll, :492-601 public region API).  All collectives go through
torch.distributed — backend "nccl" is RCCL over xGMI on ROCm.

xGMI note: on one MI355X node every GPU pair has a direct link, so
all_to_all_single is single-hop and reduce_scatter/all_gather engage all 7
links; bucket/chunk sizing decisions live in the callers.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from megatron_amd.parallel import grid as G


def _world(group) -> int:
    if group is None or not dist.is_initialized():
        return 1
    return dist.get_world_size(group=group)


def _split_along_last_dim(x: torch.Tensor, group) -> torch.Tensor:
    world = _world(group)
    if world == 1:
        return x
    assert x.size(-1) % world == 0
    rank = dist.get_rank(group=group) if _world(group) > 1 else 0
    chunk = x.size(-1) // world
    return x[..., rank * chunk : (rank + 1) * chunk].contiguous()


def _split_along_first_dim(x: torch.Tensor, group) -> torch.Tensor:
    world = _world(group)
    if world == 1:
        return x
    assert x.size(0) % world == 0
    rank = dist.get_rank(group=group) if _world(group) > 1 else 0
    chunk = x.size(0) // world
    return x[rank * chunk : (rank + 1) * chunk].contiguous()


def _gather_along_last_dim(x: torch.Tensor, group) -> torch.Tensor:
    world = _world(group)
    if world == 1:
        return x
    shape = list(x.shape)
    shape[0] *= world
    out = torch.empty(shape, dtype=x.dtype, device=x.device)
    dist.all_gather_into_tensor(out, x.contiguous(), group=group)
    # chunks along dim0 -> concat along last dim
    return torch.cat(out.chunk(world, dim=0), dim=-1)


def _gather_along_first_dim(x: torch.Tensor, group, output_buffer: torch.Tensor = None) -> torch.Tensor:
    world = _world(group)
    if world == 1:
        return x
    shape = list(x.shape)
    shape[0] *= world
    out = output_buffer if output_buffer is not None else torch.empty(shape, dtype=x.dtype, device=x.device)
    dist.all_gather_into_tensor(out, x.contiguous(), group=group)
    return out


def _reduce_scatter_along_first_dim(x: torch.Tensor, group) -> torch.Tensor:
    world = _world(group)
    if world == 1:
        return x
    assert x.size(0) % world == 0
    shape = list(x.shape)
    shape[0] //= world
    out = torch.empty(shape, dtype=x.dtype, device=x.device)
    dist.reduce_scatter_tensor(out, x.contiguous(), group=group)
    return out


def _reduce(x: torch.Tensor, group) -> torch.Tensor:
    if _world(group) == 1:
        return x
    dist.all_reduce(x.contiguous(), group=group)
    return x


# ---------------------------------------------------------------------------
# autograd region functions (TP group unless stated)
# ---------------------------------------------------------------------------


class _CopyToModelParallelRegion(torch.autograd.Function):
    """fwd: identity; bwd: all-reduce (input of column-parallel linear)."""

    @staticmethod
    def forward(ctx, x):
        return x

    @staticmethod
    def backward(ctx, grad):
        return _reduce(grad, G.get_tensor_model_parallel_group())


class _ReduceFromModelParallelRegion(torch.autograd.Function):
    """fwd: all-reduce (output of row-parallel linear); bwd: identity."""

    @staticmethod
    def forward(ctx, x):
        return _reduce(x, G.get_tensor_model_parallel_group())

    @staticmethod
    def backward(ctx, grad):
        return grad


class _ScatterToSequenceParallelRegion(torch.autograd.Function):
    """fwd: split along seq (dim 0); bwd: all-gather."""

    @staticmethod
    def forward(ctx, x):
        return _split_along_first_dim(x, G.get_tensor_model_parallel_group())

    @staticmethod
    def backward(ctx, grad):
        return _gather_along_first_dim(grad, G.get_tensor_model_parallel_group())


class _GatherFromSequenceParallelRegion(torch.autograd.Function):
    """fwd: all-gather along seq; bwd: reduce-scatter."""

    @staticmethod
    def forward(ctx, x, tensor_parallel_output_grad=True):
        ctx.tensor_parallel_output_grad = tensor_parallel_output_grad
        return _gather_along_first_dim(x, G.get_tensor_model_parallel_group())

    @staticmethod
    def backward(ctx, grad):
        if ctx.tensor_parallel_output_grad:
            return _reduce_scatter_along_first_dim(grad, G.get_tensor_model_parallel_group()), None
        return _split_along_first_dim(grad, G.get_tensor_model_parallel_group()), None


class _ReduceScatterToSequenceParallelRegion(torch.autograd.Function):
    """fwd: reduce-scatter along seq; bwd: all-gather."""

    @staticmethod
    def forward(ctx, x):
        return _reduce_scatter_along_first_dim(x, G.get_tensor_model_parallel_group())

    @staticmethod
    def backward(ctx, grad):
        return _gather_along_first_dim(grad, G.get_tensor_model_parallel_group())


class _GatherFromModelParallelRegion(torch.autograd.Function):
    """fwd: all-gather along last dim; bwd: split along last dim."""

    @staticmethod
    def forward(ctx, x):
        return _gather_along_last_dim(x, G.get_tensor_model_parallel_group())

    @staticmethod
    def backward(ctx, grad):
        return _split_along_last_dim(grad, G.get_tensor_model_parallel_group())


class _ScatterToModelParallelRegion(torch.autograd.Function):
    """fwd: split along last dim; bwd: all-gather along last dim."""

    @staticmethod
    def forward(ctx, x):
        return _split_along_last_dim(x, G.get_tensor_model_parallel_group())

    @staticmethod
    def backward(ctx, grad):
        return _gather_along_last_dim(grad, G.get_tensor_model_parallel_group())


class _AllToAll(torch.autograd.Function):
    """all_to_all_single with optional uneven splits (reference mappings.py:424)."""

    @staticmethod
    def forward(ctx, group, x, output_split_sizes=None, input_split_sizes=None):
        ctx.group = group
        ctx.output_split_sizes = output_split_sizes
        ctx.input_split_sizes = input_split_sizes
        world = _world(group)
        if world == 1:
            return x
        x = x.contiguous()
        if output_split_sizes is None:
            out = torch.empty_like(x)
        else:
            shape = list(x.shape)
            shape[0] = sum(output_split_sizes)
            out = torch.empty(shape, dtype=x.dtype, device=x.device)
        dist.all_to_all_single(
            out, x, output_split_sizes=output_split_sizes, input_split_sizes=input_split_sizes, group=group
        )
        return out

    @staticmethod
    def backward(ctx, grad):
        return (
            None,
            _AllToAll.apply(ctx.group, grad, ctx.input_split_sizes, ctx.output_split_sizes),
            None,
            None,
        )


# ---------------------------------------------------------------------------
# public API (reference mappings.py:492-601)
# ---------------------------------------------------------------------------


def copy_to_tensor_model_parallel_region(x):
    return _CopyToModelParallelRegion.apply(x)


def reduce_from_tensor_model_parallel_region(x):
    return _ReduceFromModelParallelRegion.apply(x)


def scatter_to_sequence_parallel_region(x):
    return _ScatterToSequenceParallelRegion.apply(x)


def gather_from_sequence_parallel_region(x, tensor_parallel_output_grad=True):
    return _GatherFromSequenceParallelRegion.apply(x, tensor_parallel_output_grad)


def reduce_scatter_to_sequence_parallel_region(x):
    return _ReduceScatterToSequenceParallelRegion.apply(x)


def gather_from_tensor_model_parallel_region(x):
    return _GatherFromModelParallelRegion.apply(x)


def scatter_to_tensor_model_parallel_region(x):
    return _ScatterToModelParallelRegion.apply(x)


def all_to_all(group, x, output_split_sizes=None, input_split_sizes=None):
    return _AllToAll.apply(group, x, output_split_sizes, input_split_sizes)


def all_to_all_sp2hp(x):
    """[s/tp, b, h] -> [s, b, h/tp] over the TP group (Ulysses head-scatter)."""
    group = G.get_tensor_model_parallel_group()
    world = _world(group) if group is not None else 1
    if world == 1:
        return x
    s, b, h = x.shape
    xt = x.reshape(s, b, world, h // world).permute(2, 0, 1, 3).contiguous().view(world * s, b, h // world)
    out = _AllToAll.apply(group, xt, None, None)
    return out.view(world, s, b, h // world).reshape(world * s, b, h // world)


def all_to_all_hp2sp(x):
    """[s, b, h/tp] -> [s/tp, b, h] over the TP group."""
    group = G.get_tensor_model_parallel_group()
    world = _world(group) if group is not None else 1
    if world == 1:
        return x
    s, b, hp = x.shape
    out = _AllToAll.apply(group, x.contiguous(), None, None)
    return out.view(world, s // world, b, hp).permute(1, 2, 0, 3).reshape(s // world, b, hp * world)
