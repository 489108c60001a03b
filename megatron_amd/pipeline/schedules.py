"""Pipeline schedules.

Capability analog of reference megatron/core/pipeline_parallel/schedules.py
(get_forward_backward_func :48, no-pipelining :705, non-interleaved 1F1B
:2129, interleaved 1F1B :1001).

Contract (same as reference): ``forward_step_func(data_iterator, model)``
returns ``(output_tensor, loss_func)`` where ``loss_func(output_tensor)``
returns ``(loss, num_tokens, {metrics})``; loss is the micro-batch SUM loss
and the schedule divides by the total token count at the end.

The three DDP/optimizer hooks ride on config (reference
model_parallel_config.py:211-223): no_sync_func, grad_sync_func,
param_sync_func.
"""

from __future__ import annotations

import contextlib
from typing import Callable, List, Optional

import torch

from megatron_amd.parallel import grid as G


def get_forward_backward_func(config=None):
    combined = config is not None and getattr(config, "overlap_moe_expert_parallel_comm", False)
    if combined and not (G.grid_initialized() and G.get_pipeline_model_parallel_world_size() > 1):
        from megatron_amd.pipeline.combined_1f1b import forward_backward_no_pipelining_combined

        return forward_backward_no_pipelining_combined
    if (combined and G.grid_initialized() and G.get_pipeline_model_parallel_world_size() > 1
            and (G.get_grid().vpp is None or G.get_grid().vpp <= 1)):
        from megatron_amd.pipeline.combined_1f1b import (
            forward_backward_pipelining_without_interleaving_combined,
        )

        return forward_backward_pipelining_without_interleaving_combined
    if G.grid_initialized() and G.get_pipeline_model_parallel_world_size() > 1:
        grid = G.get_grid()
        if grid.vpp is not None and grid.vpp > 1:
            from megatron_amd.pipeline.pipelined import forward_backward_pipelining_with_interleaving

            return forward_backward_pipelining_with_interleaving
        from megatron_amd.pipeline.pipelined import forward_backward_pipelining_without_interleaving

        return forward_backward_pipelining_without_interleaving
    return forward_backward_no_pipelining


def _forward_step(forward_step_func, data_iterator, model, losses, num_tokens_acc, num_microbatches, config):
    """Returns the loss scaled for backward: loss_sum / (ntok * num_microbatches)
    so the accumulated gradient is the gradient of the per-token-average loss
    over the rank's whole batch (DP buckets then AVG across ranks)."""
    output, loss_func = forward_step_func(data_iterator, model)
    loss, num_tokens, metrics = loss_func(output)
    losses.append(metrics)
    num_tokens_acc.add_(num_tokens)
    scale = 1.0 / (max(int(num_tokens), 1) * num_microbatches)
    from megatron_amd.moe.router import AuxLossScaler

    AuxLossScaler.bind_scale(scale)
    if config.grad_scale_func is not None:
        return config.grad_scale_func(loss * scale)
    return loss * scale


def forward_backward_no_pipelining(
    *,
    forward_step_func: Callable,
    data_iterator,
    model,
    num_microbatches: int,
    seq_length: int = None,
    micro_batch_size: int = None,
    forward_only: bool = False,
    **kw,
):
    if isinstance(model, list):
        assert len(model) == 1
        model = model[0]
    if isinstance(data_iterator, list):
        assert len(data_iterator) == 1
        data_iterator = data_iterator[0]
    config = (model.module if hasattr(model, "module") else model).config

    no_sync = config.no_sync_func
    if no_sync is None and hasattr(model, "no_sync"):
        no_sync = model.no_sync
    if no_sync is None:
        no_sync = contextlib.nullcontext

    losses: List[dict] = []
    device = next(model.parameters()).device
    num_tokens_acc = torch.zeros((), dtype=torch.long, device=device)

    with no_sync():
        for _ in range(num_microbatches - 1):
            loss = _forward_step(
                forward_step_func, data_iterator, model, losses, num_tokens_acc, num_microbatches, config
            )
            if not forward_only:
                loss.backward()
    loss = _forward_step(forward_step_func, data_iterator, model, losses, num_tokens_acc, num_microbatches, config)
    if not forward_only:
        loss.backward()

    if not forward_only and config.finalize_model_grads_func is not None:
        config.finalize_model_grads_func([model], config)

    return losses, num_tokens_acc
