"""1F1B and interleaved-1F1B pipeline schedules.

Capability analog of reference megatron/core/pipeline_parallel/schedules.py
(non-interleaved :2129, interleaved :1001, schedule table :971, warmup math
:911-959 — see SURVEY.md §8.1).
"""

from __future__ import annotations

import contextlib
from typing import Callable, List, Optional

import torch

from megatron_amd.parallel import grid as G
from megatron_amd.pipeline.p2p import P2PCommunicator


def _get_config(model):
    core = model.module if hasattr(model, "module") else model
    return core.config


def _fwd(forward_step_func, data_iterator, model, input_tensor, losses, num_tokens_acc,
         num_microbatches, config, is_last_stage):
    core = model.module if hasattr(model, "module") else model
    core.set_input_tensor(input_tensor)
    output, loss_func = forward_step_func(data_iterator, model)
    if is_last_stage:
        loss, num_tokens, metrics = loss_func(output)
        losses.append(metrics)
        num_tokens_acc.add_(num_tokens)
        scale = 1.0 / (max(int(num_tokens), 1) * num_microbatches)
        from megatron_amd.moe.router import AuxLossScaler

        AuxLossScaler.bind_scale(scale)
        out = loss * scale
        if config.grad_scale_func is not None:
            out = config.grad_scale_func(out)
        return out
    # Non-last stages never see loss_func's token count, but their MoE aux
    # losses still need the main-loss scale.  Estimate tokens from the stage
    # output shape [s, b, h] (exact when the loss mask is all-ones).
    if output.dim() >= 2:
        est_tokens = output.shape[0] * output.shape[1]
        from megatron_amd.moe.router import AuxLossScaler

        AuxLossScaler.bind_scale(1.0 / (max(est_tokens, 1) * num_microbatches))
    return output


def _bwd(input_tensor, output_tensor, output_grad):
    """Run backward for one microbatch chunk; returns grad wrt input."""
    if input_tensor is not None:
        input_tensor.retain_grad()
    if output_grad is None:
        torch.autograd.backward(output_tensor)
    else:
        torch.autograd.backward(output_tensor, grad_tensors=output_grad)
    return input_tensor.grad if input_tensor is not None else None


def forward_backward_pipelining_without_interleaving(
    *,
    forward_step_func: Callable,
    data_iterator,
    model,
    num_microbatches: int,
    seq_length: int,
    micro_batch_size: int,
    forward_only: bool = False,
    **kw,
):
    """Non-interleaved 1F1B (reference schedules.py:2129)."""
    if isinstance(model, list):
        assert len(model) == 1
        model = model[0]
    if isinstance(data_iterator, list):
        data_iterator = data_iterator[0]
    config = _get_config(model)
    grid = G.get_grid()
    comm = P2PCommunicator(config, seq_length, micro_batch_size)
    is_first = grid.is_pipeline_first_stage(ignore_virtual=True)
    is_last = grid.is_pipeline_last_stage(ignore_virtual=True)
    pp, pp_rank = grid.pp, grid.pp_rank

    no_sync = config.no_sync_func
    if no_sync is None and hasattr(model, "no_sync"):
        no_sync = model.no_sync
    if no_sync is None:
        no_sync = contextlib.nullcontext

    losses: List[dict] = []
    device = next(model.parameters()).device
    num_tokens_acc = torch.zeros((), dtype=torch.long, device=device)

    num_warmup = min(pp - pp_rank - 1, num_microbatches)
    num_steady = num_microbatches - num_warmup

    input_tensors, output_tensors = [], []

    with no_sync():
        # ---- warmup ----
        for _ in range(num_warmup):
            input_tensor = comm.recv_forward(is_first)
            output = _fwd(forward_step_func, data_iterator, model, input_tensor, losses,
                          num_tokens_acc, num_microbatches, config, is_last)
            comm.send_forward(output, is_last)
            if not forward_only:
                input_tensors.append(input_tensor)
                output_tensors.append(output)

        # ---- steady 1F1B ----
        if num_steady > 0:
            input_tensor = comm.recv_forward(is_first)
        for i in range(num_steady):
            last_iter = i == num_steady - 1
            output = _fwd(forward_step_func, data_iterator, model, input_tensor, losses,
                          num_tokens_acc, num_microbatches, config, is_last)
            if forward_only:
                comm.send_forward(output, is_last)
                if not last_iter:
                    input_tensor = comm.recv_forward(is_first)
                continue
            output_grad = comm.send_forward_recv_backward(output, is_last)
            input_tensors.append(input_tensor)
            output_tensors.append(output)
            in_t, out_t = input_tensors.pop(0), output_tensors.pop(0)
            input_grad = _bwd(in_t, out_t, output_grad)
            if last_iter:
                comm.send_backward(input_grad, is_first)
            else:
                input_tensor = comm.send_backward_recv_forward(input_grad, is_first)

        # ---- cooldown backwards ----
        if not forward_only:
            for i in range(num_warmup):
                if i == num_warmup - 1 and num_steady == 0:
                    pass  # last backward below handles grad sync outside no_sync
                in_t, out_t = input_tensors.pop(0), output_tensors.pop(0)
                output_grad = comm.recv_backward(is_last)
                input_grad = _bwd(in_t, out_t, output_grad)
                comm.send_backward(input_grad, is_first)

    if not forward_only:
        # trigger bucket grad reduce (it was suppressed by no_sync for every
        # microbatch; launch now, once)
        if hasattr(model, "start_grad_sync"):
            model.start_grad_sync()
        if config.finalize_model_grads_func is not None:
            config.finalize_model_grads_func([model], config)
    return losses, num_tokens_acc


def get_schedule_table(num_microbatches: int, num_chunks: int, group_size: int):
    """virtual step -> (microbatch, chunk) (reference schedules.py:971)."""
    table = []
    for g0 in range(0, num_microbatches, group_size):
        group = list(range(g0, min(g0 + group_size, num_microbatches)))
        for chunk in range(num_chunks):
            for mb in group:
                table.append((mb, chunk))
    return table


def forward_backward_pipelining_with_interleaving(
    *,
    forward_step_func: Callable,
    data_iterator,
    model: List,
    num_microbatches: int,
    seq_length: int,
    micro_batch_size: int,
    forward_only: bool = False,
    **kw,
):
    """Interleaved 1F1B over virtual chunks (reference schedules.py:1001).

    Deadlock-freedom: ONE fused batched p2p per schedule tick (all four
    directions posted together, reference send_forward_backward_recv_
    forward_backward); the grad for backward step j is prefetched by the
    previous tick's fused call.  Backward walks the forward table with the
    chunk axis reversed.
    """
    assert isinstance(model, list) and len(model) > 1
    if not isinstance(data_iterator, list) or len(data_iterator) == 1:
        it0 = data_iterator[0] if isinstance(data_iterator, list) else data_iterator
        data_iterator = [it0] * len(model)
    config = _get_config(model[0])
    grid = G.get_grid()
    comm = P2PCommunicator(config, seq_length, micro_batch_size)
    pp, pp_rank = grid.pp, grid.pp_rank
    num_chunks = len(model)
    group_size = config.microbatch_group_size_per_vp_stage or pp
    # num_microbatches need not divide pp: the schedule table's last group is
    # simply smaller and the warmup count is clamped to the table length
    # (reference schedules.py:959 handles the general case the same way).

    no_syncs = [m.no_sync() for m in model if hasattr(m, "no_sync")]
    for c in no_syncs:
        c.__enter__()

    losses: List[dict] = []
    device = next(model[0].parameters()).device
    num_tokens_acc = torch.zeros((), dtype=torch.long, device=device)

    table = get_schedule_table(num_microbatches, num_chunks, group_size)
    # backward visits chunks in reverse order (last global stage first)
    btable = [(mb, num_chunks - 1 - c) for (mb, c) in table]
    total = len(table)
    num_warmup = (pp - pp_rank - 1) * 2 + (num_chunks - 1) * group_size
    num_warmup = min(num_warmup, total)
    if forward_only:
        num_warmup = total
    num_steady = total - num_warmup

    input_q = [[] for _ in range(num_chunks)]
    output_q = [[] for _ in range(num_chunks)]

    def first_stage(c):
        return pp_rank == 0 and c == 0

    def last_stage(c):
        return pp_rank == pp - 1 and c == num_chunks - 1

    def fwd_step(k, input_tensor):
        mb, c = table[k]
        grid.set_vpp_rank(c)
        out = _fwd(forward_step_func, data_iterator[c], model[c], input_tensor, losses,
                   num_tokens_acc, num_microbatches, config, last_stage(c))
        if not forward_only:
            input_q[c].append(input_tensor)
            output_q[c].append(out)
        return out

    def bwd_step(j, output_grad):
        mb, c = btable[j]
        grid.set_vpp_rank(c)
        in_t = input_q[c].pop(0)
        out_t = output_q[c].pop(0)
        return _bwd(in_t, out_t, output_grad)

    # ---- initial recv for table[0] ----
    input_tensor = comm.recv_forward(first_stage(table[0][1]))
    output_grad = None
    fwd_k, bwd_j = 0, 0

    # ---- warmup forwards ----
    for i in range(num_warmup):
        out = fwd_step(fwd_k, input_tensor)
        send_next = not last_stage(table[fwd_k][1])
        fwd_k += 1
        recv_prev = fwd_k < total and not first_stage(table[fwd_k][1])
        last_tick = i == num_warmup - 1 and not forward_only and total > 0
        recv_next = last_tick and not last_stage(btable[0][1])
        input_tensor, output_grad = comm.communicate(
            tensor_send_next=out if send_next else None,
            recv_prev=recv_prev, recv_next=recv_next,
        )

    if forward_only:
        for c in reversed(no_syncs):
            c.__exit__(None, None, None)
        grid.set_vpp_rank(0)
        return losses, num_tokens_acc

    # ---- steady 1F1B: one fused p2p per tick ----
    for i in range(num_steady):
        out = fwd_step(fwd_k, input_tensor)
        send_next = not last_stage(table[fwd_k][1])
        fwd_k += 1
        input_grad = bwd_step(bwd_j, output_grad)
        send_prev = not first_stage(btable[bwd_j][1])
        bwd_j += 1
        recv_prev = fwd_k < total and not first_stage(table[fwd_k][1])
        recv_next = bwd_j < total and not last_stage(btable[bwd_j][1])
        input_tensor, output_grad = comm.communicate(
            tensor_send_next=out if send_next else None,
            tensor_send_prev=input_grad if send_prev else None,
            recv_prev=recv_prev, recv_next=recv_next,
        )

    # ---- cooldown backwards ----
    while bwd_j < total:
        input_grad = bwd_step(bwd_j, output_grad)
        send_prev = not first_stage(btable[bwd_j][1])
        bwd_j += 1
        recv_next = bwd_j < total and not last_stage(btable[bwd_j][1])
        _, output_grad = comm.communicate(
            tensor_send_prev=input_grad if send_prev else None, recv_next=recv_next
        )

    for c in reversed(no_syncs):
        c.__exit__(None, None, None)
    for m in model:
        if hasattr(m, "start_grad_sync"):
            m.start_grad_sync()
    if config.finalize_model_grads_func is not None:
        config.finalize_model_grads_func(model, config)
    grid.set_vpp_rank(0)
    return losses, num_tokens_acc
