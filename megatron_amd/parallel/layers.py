"""Tensor-parallel linear layers and vocab-parallel embedding.

Capability analog of reference megatron/core/tensor_parallel/layers.py
(ColumnParallelLinear :868, RowParallelLinear :1247, VocabParallelEmbedding
:230, LinearWithGradAccumulationAndAsyncCommunication :524).

MI355X-first design decisions:
  * Plain GEMMs go through torch.matmul -> hipBLASLt (Tensile hand-asm,
    ~80% of bf16 MFMA peak at large N) — no custom GEMM here.
  * The dgrad TP all-reduce is issued async and overlapped with the wgrad
    GEMM (the reference relies on CUDA_DEVICE_MAX_CONNECTIONS=1 ordering;
    on ROCm the async handle + wait gives the same overlap window).
  * Weight-grad accumulation goes straight into a persistent fp32
    ``main_grad`` buffer (gradient_accumulation_fusion): the GEMM runs in
    bf16 on MFMA (fp32 accumulate inside the MFMA), the add into main_grad
    is one fused elementwise pass; a hipBLASLt beta=1/fp32-D path replaces
    it when the native extension is built (ops.wgrad_gemm_accum).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from megatron_amd.parallel import grid as G
from megatron_amd.parallel.mappings import (
    _gather_along_first_dim,
    _reduce_scatter_along_first_dim,
    copy_to_tensor_model_parallel_region,
    gather_from_sequence_parallel_region,
    reduce_from_tensor_model_parallel_region,
    reduce_scatter_to_sequence_parallel_region,
    scatter_to_sequence_parallel_region,
)


def _wgrad_accum(main_grad: torch.Tensor, grad_output_2d: torch.Tensor, input_2d: torch.Tensor):
    """main_grad(fp32) += grad_output^T @ input.

    Fallback path: bf16 GEMM on MFMA (internal fp32 accumulate) + fp32 add.
    Native path (when built): hipBLASLt with fp32 D and beta=1 — see
    megatron_amd/ops (K9 in SURVEY.md §2.3).
    """
    from megatron_amd import ops

    if ops.has_native() and main_grad.is_cuda:
        ops.wgrad_gemm_accum(main_grad, grad_output_2d, input_2d)
    else:
        main_grad.add_(torch.matmul(grad_output_2d.t(), input_2d).to(main_grad.dtype))


class _ParallelLinearFn(torch.autograd.Function):
    """fused forward/backward for column- and row-parallel linears.

    Handles: optional SP all-gather of the input (fwd) / reduce-scatter of
    dgrad (bwd), async TP all-reduce of dgrad overlapped with the wgrad GEMM,
    and grad-accumulation fusion into weight.main_grad.
    Reference: LinearWithGradAccumulationAndAsyncCommunication layers.py:524.
    """

    @staticmethod
    def forward(ctx, inp, weight, bias, grad_accum_fusion, async_grad_allreduce,
                sequence_parallel, fp8_recipes=None, tp_comm_overlap=False):
        ctx.use_bias = bias is not None
        ctx.grad_accum_fusion = grad_accum_fusion
        ctx.async_grad_allreduce = async_grad_allreduce
        ctx.sequence_parallel = sequence_parallel
        if sequence_parallel and tp_comm_overlap and fp8_recipes is None:
            # chunked ring AG+GEMM: each ring hop overlaps the previous
            # chunk's GEMM (parallel/overlap.py; TE ub_overlap_ag analog).
            # Backward is unchanged - it re-gathers with its own async AG.
            from megatron_amd.parallel.overlap import ring_allgather_gemm

            ctx.save_for_backward(inp, weight)
            ctx.fp8 = False
            group = G.get_tensor_model_parallel_group()
            world = dist.get_world_size(group) if dist.is_initialized() else 1
            x2 = inp.reshape(-1, inp.shape[-1])
            y2 = ring_allgather_gemm(x2, weight, group=group)
            output = y2.view(inp.shape[0] * world, *inp.shape[1:-1], weight.shape[0])
            if bias is not None:
                output = output + bias
            return output
        if sequence_parallel:
            # fwd-only gather (backward re-gathers): land it in the reused
            # global scratch (reference GlobalMemoryBuffer, utils.py:693)
            from megatron_amd.parallel.memory_buffer import get_global_memory_buffer

            group = G.get_tensor_model_parallel_group()
            world = dist.get_world_size(group) if dist.is_initialized() else 1
            shape = (inp.shape[0] * world,) + tuple(inp.shape[1:])
            buf = get_global_memory_buffer().get_tensor(shape, inp.dtype, "sp-ag-fwd",
                                                        device=inp.device)
            total_input = _gather_along_first_dim(inp, group, output_buffer=buf)
        else:
            total_input = inp
        ctx.save_for_backward(inp, weight)
        ctx.fp8 = False
        if fp8_recipes is not None:
            from megatron_amd.ops import fp8 as F8

            if F8.fp8_eligible(total_input, weight):
                rx, rw, rg = fp8_recipes
                x2d = total_input.reshape(-1, total_input.shape[-1])
                xq, inv_x = F8.quantize_fp8(x2d, rx.scale_for(x2d))
                wq, inv_w = F8.quantize_fp8(weight, rw.scale_for(weight))
                output = F8._scaled_mm(xq, wq.t(), inv_x, inv_w, total_input.dtype)
                output = output.reshape(*total_input.shape[:-1], weight.shape[0])
                # column-major fp8 weight copy for the dgrad GEMM
                ctx.wq_cm = wq.t().contiguous().t()
                ctx.inv_w = inv_w
                ctx.rg = rg
                ctx.fp8 = True
                if bias is not None:
                    output = output + bias
                return output
        output = torch.matmul(total_input, weight.t())
        if bias is not None:
            output = output + bias
        return output

    @staticmethod
    def backward(ctx, grad_output):
        inp, weight = ctx.saved_tensors
        group = G.get_tensor_model_parallel_group()

        if ctx.sequence_parallel:
            # async all-gather of the (seq-sharded) input for the wgrad GEMM,
            # overlapped with the dgrad GEMM below (reference layers.py:609-616)
            from megatron_amd.parallel.mappings import _world
            from megatron_amd.parallel.memory_buffer import get_global_memory_buffer

            world = _world(group)
            shape = list(inp.shape)
            shape[0] *= world
            # reuse the global scratch (a fresh torch.empty per microbatch
            # churned the allocator on the hot path; distinct tag from the
            # forward AG buffer, which may still be alive in this step)
            total_input = get_global_memory_buffer().get_tensor(
                shape, inp.dtype, "sp-ag-bwd", device=inp.device)
            gather_handle = dist.all_gather_into_tensor(
                total_input, inp.contiguous(), group=group, async_op=True
            )
        else:
            total_input = inp
            gather_handle = None

        if ctx.fp8:
            from megatron_amd.ops import fp8 as F8

            go2d = grad_output.reshape(-1, grad_output.shape[-1])
            dyq, inv_dy = F8.quantize_fp8(go2d, ctx.rg.scale_for(go2d), fmt=torch.float8_e5m2)
            grad_input = torch._scaled_mm(dyq, ctx.wq_cm, scale_a=inv_dy, scale_b=ctx.inv_w,
                                          out_dtype=grad_output.dtype)
            grad_input = grad_input.reshape(*grad_output.shape[:-1], weight.shape[1])
        else:
            grad_input = torch.matmul(grad_output, weight)

        if gather_handle is not None:
            gather_handle.wait()

        allreduce_handle = None
        if ctx.async_grad_allreduce:
            # overlap dgrad all-reduce with wgrad GEMM (reference layers.py:640)
            allreduce_handle = dist.all_reduce(grad_input, group=group, async_op=True)

        go2 = grad_output.reshape(-1, grad_output.shape[-1])
        in2 = total_input.reshape(-1, total_input.shape[-1])

        if ctx.sequence_parallel:
            # reduce-scatter dgrad, async, overlapped with wgrad GEMM
            assert not ctx.async_grad_allreduce
            sub_shape = list(grad_input.shape)
            from megatron_amd.parallel.mappings import _world as _w
            sub_shape[0] //= _w(group)
            grad_input_rs = torch.empty(sub_shape, dtype=grad_input.dtype, device=grad_input.device)
            rs_handle = dist.reduce_scatter_tensor(
                grad_input_rs, grad_input.contiguous(), group=group, async_op=True
            )
        else:
            rs_handle = None

        if ctx.grad_accum_fusion and hasattr(weight, "main_grad"):
            _wgrad_accum(weight.main_grad, go2, in2)
            grad_weight = None
            weight.grad_added_to_main_grad = True
            cb = getattr(weight, "_ddp_grad_ready_cb", None)
            if cb is not None:
                cb()  # bucket-readiness signal for overlapped grad reduce
        else:
            grad_weight = torch.matmul(go2.t(), in2)
        grad_bias = go2.sum(dim=0) if ctx.use_bias else None

        if rs_handle is not None:
            rs_handle.wait()
            grad_input = grad_input_rs
        if allreduce_handle is not None:
            allreduce_handle.wait()

        return grad_input, grad_weight, grad_bias, None, None, None, None, None


def linear_with_grad_accumulation_and_async_allreduce(
    inp, weight, bias, grad_accum_fusion, async_grad_allreduce, sequence_parallel,
    fp8_recipes=None, tp_comm_overlap=False,
):
    return _ParallelLinearFn.apply(inp, weight, bias, grad_accum_fusion,
                                   async_grad_allreduce, sequence_parallel, fp8_recipes,
                                   tp_comm_overlap)


def _init_weight(weight: torch.Tensor, init_std: float, generator: Optional[torch.Generator] = None):
    with torch.no_grad():
        weight.normal_(mean=0.0, std=init_std, generator=generator)


class ColumnParallelLinear(nn.Module):
    """Y = XA^T + b with A sharded along its output dim over TP.

    Reference: layers.py:868.  gather_output=False keeps the TP-sharded
    output (the normal transformer path).
    """

    def __init__(
        self,
        input_size: int,
        output_size: int,
        *,
        config,
        bias: bool = True,
        gather_output: bool = False,
        skip_bias_add: bool = False,
        is_expert: bool = False,
        init_std: Optional[float] = None,
    ):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        tp = G.get_tensor_model_parallel_world_size()
        assert output_size % tp == 0
        self.output_size_per_partition = output_size // tp
        self.gather_output = gather_output
        self.skip_bias_add = skip_bias_add
        self.sequence_parallel = config.sequence_parallel and tp > 1
        self.async_tp_allreduce = (
            config.async_tensor_model_parallel_allreduce and tp > 1 and not self.sequence_parallel
        )
        self.grad_accum_fusion = config.gradient_accumulation_fusion
        self.tp_comm_overlap = getattr(config, "tp_comm_overlap", False)
        self.fp8_recipes = None
        if getattr(config, "fp8", None):
            from megatron_amd.ops.fp8 import make_recipes

            self.fp8_recipes = make_recipes(config.fp8, config.fp8_amax_history_len,
                                            config.fp8_margin)
        self.weight = nn.Parameter(
            torch.empty(self.output_size_per_partition, input_size, dtype=config.params_dtype)
        )
        self.weight.tensor_parallel = tp > 1
        self.weight.partition_dim = 0
        from megatron_amd.parallel.random import get_rng_tracker

        with get_rng_tracker().fork():
            _init_weight(self.weight, init_std if init_std is not None else config.init_method_std)
        if bias:
            self.bias = nn.Parameter(torch.zeros(self.output_size_per_partition, dtype=config.params_dtype))
            self.bias.tensor_parallel = tp > 1
            self.bias.partition_dim = 0
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        bias = self.bias if not self.skip_bias_add else None
        if self.sequence_parallel:
            inp = x  # gather happens inside the fused fn
        elif G.get_tensor_model_parallel_world_size() > 1 and not self.async_tp_allreduce:
            inp = copy_to_tensor_model_parallel_region(x)
        else:
            inp = x
        out = linear_with_grad_accumulation_and_async_allreduce(
            inp, self.weight, bias, self.grad_accum_fusion, self.async_tp_allreduce,
            self.sequence_parallel, self.fp8_recipes if self.training else None,
            self.tp_comm_overlap,
        )
        if self.gather_output:
            from megatron_amd.parallel.mappings import gather_from_tensor_model_parallel_region

            out = gather_from_tensor_model_parallel_region(out)
        out_bias = self.bias if self.skip_bias_add else None
        return out, out_bias


class RowParallelLinear(nn.Module):
    """Y = XA^T + b with A sharded along its input dim over TP.

    Reference: layers.py:1247.  Output is all-reduced over TP (or
    reduce-scattered when sequence_parallel).
    """

    def __init__(
        self,
        input_size: int,
        output_size: int,
        *,
        config,
        bias: bool = True,
        input_is_parallel: bool = True,
        skip_bias_add: bool = False,
        is_expert: bool = False,
        init_std: Optional[float] = None,
    ):
        super().__init__()
        tp = G.get_tensor_model_parallel_world_size()
        assert input_size % tp == 0
        self.input_size_per_partition = input_size // tp
        self.output_size = output_size
        self.input_is_parallel = input_is_parallel
        self.skip_bias_add = skip_bias_add
        self.sequence_parallel = config.sequence_parallel and tp > 1
        self.grad_accum_fusion = config.gradient_accumulation_fusion
        self.tp_comm_overlap = getattr(config, "tp_comm_overlap", False)
        self.fp8_recipes = None
        if getattr(config, "fp8", None):
            from megatron_amd.ops.fp8 import make_recipes

            self.fp8_recipes = make_recipes(config.fp8, config.fp8_amax_history_len,
                                            config.fp8_margin)
        self.weight = nn.Parameter(
            torch.empty(output_size, self.input_size_per_partition, dtype=config.params_dtype)
        )
        self.weight.tensor_parallel = tp > 1
        self.weight.partition_dim = 1
        from megatron_amd.parallel.random import get_rng_tracker

        with get_rng_tracker().fork():
            _init_weight(self.weight, init_std if init_std is not None else config.init_method_std)
        if bias:
            # bias is replicated; added after the reduction
            self.bias = nn.Parameter(torch.zeros(output_size, dtype=config.params_dtype))
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        assert self.input_is_parallel
        out_parallel = linear_with_grad_accumulation_and_async_allreduce(
            x, self.weight, None, self.grad_accum_fusion, False, False,
            self.fp8_recipes if self.training else None
        )
        if self.sequence_parallel:
            out = reduce_scatter_to_sequence_parallel_region(out_parallel)
        else:
            out = reduce_from_tensor_model_parallel_region(out_parallel)
        if self.skip_bias_add:
            return out, self.bias
        if self.bias is not None:
            out = out + self.bias
        return out, None


class VocabParallelEmbedding(nn.Module):
    """Embedding sharded along vocab over TP (reference layers.py:230)."""

    def __init__(self, num_embeddings: int, embedding_dim: int, *, config, init_std: Optional[float] = None):
        super().__init__()
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        tp = G.get_tensor_model_parallel_world_size()
        rank = G.get_tensor_model_parallel_rank()
        assert num_embeddings % tp == 0, "pad vocab to a multiple of tp"
        self.vocab_per_partition = num_embeddings // tp
        self.vocab_start = rank * self.vocab_per_partition
        self.vocab_end = self.vocab_start + self.vocab_per_partition
        self.sequence_parallel = config.sequence_parallel and tp > 1
        self.weight = nn.Parameter(
            torch.empty(self.vocab_per_partition, embedding_dim, dtype=config.params_dtype)
        )
        self.weight.tensor_parallel = tp > 1
        self.weight.partition_dim = 0
        _init_weight(self.weight, init_std if init_std is not None else config.init_method_std)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        tp = G.get_tensor_model_parallel_world_size()
        if tp > 1:
            mask = (input_ids < self.vocab_start) | (input_ids >= self.vocab_end)
            masked = input_ids.clone() - self.vocab_start
            masked[mask] = 0
            out = F.embedding(masked, self.weight)
            out = out.masked_fill(mask.unsqueeze(-1), 0.0)
        else:
            out = F.embedding(input_ids, self.weight)
        # [b, s, h] -> [s, b, h]
        out = out.transpose(0, 1).contiguous()
        if self.sequence_parallel:
            out = reduce_scatter_to_sequence_parallel_region(out)
        elif tp > 1:
            out = reduce_from_tensor_model_parallel_region(out)
        return out
