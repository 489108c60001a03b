"""Vocab-parallel cross-entropy.

Capability analog of reference megatron/core/tensor_parallel/cross_entropy.py
(:13 _VocabParallelCrossEntropy, :213 vocab_parallel_cross_entropy) and the
fused variant fusions/fused_cross_entropy.py:13-65.

Native path (GPU, label_smoothing=0): ONE online HIP kernel pass over the
bf16 logits in forward (running (max, sumexp) merge + target pick —
`ops/csrc/cross_entropy.hip`), two tiny [T] all-reduces over TP (row max,
then rebased sumexp + target logit packed into one collective), and ONE
backward kernel pass that writes the bf16 softmax-grad IN-PLACE over the
logits buffer.  Nothing of shape [T, V] is ever materialized beyond the
logits themselves (the previous torch-composed path kept an fp32 softmax
copy alive from forward to backward: 8.4 GB at T=16k, V=128k).

Fallback (CPU tensors, or label_smoothing > 0): composed torch ops with
the same two-all-reduce algorithm.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from megatron_amd.parallel import grid as G


class _VocabParallelCrossEntropyNative(torch.autograd.Function):
    """Fused kernel path: loss [T] fp32 from logits [T, V/tp] bf16."""

    @staticmethod
    def forward(ctx, logits, target):
        from megatron_amd import ops

        group = G.get_tensor_model_parallel_group()
        tp = G.get_tensor_model_parallel_world_size()
        rank = G.get_tensor_model_parallel_rank()
        part_vocab = logits.size(-1)
        vocab_start = rank * part_vocab

        logits = logits.contiguous()
        m_loc, s_loc, pred_raw = ops._C.ce_fwd(logits, target, vocab_start)
        if tp > 1:
            m = m_loc.clone()
            dist.all_reduce(m, op=dist.ReduceOp.MAX, group=group)
            # rebase local sumexp from the local max to the global max,
            # then one packed SUM collective for (sumexp, target_logit)
            sp = torch.stack((s_loc * torch.exp(m_loc - m), pred_raw))
            dist.all_reduce(sp, op=dist.ReduceOp.SUM, group=group)
            s, pred_raw = sp[0], sp[1]
        else:
            m, s = m_loc, s_loc
        loss = torch.log(s) - (pred_raw - m)
        ctx.save_for_backward(logits, target, m, s)
        ctx.vocab_start = vocab_start
        return loss

    @staticmethod
    def backward(ctx, grad_output):
        from megatron_amd import ops

        logits, target, m, s = ctx.saved_tensors
        # write (softmax - onehot) * grad_output into the logits buffer —
        # the logits are dead after this point (the producing GEMM's
        # backward needs its own inputs, not its output)
        ops._C.ce_bwd(logits, target, m, s.reciprocal(), grad_output.float(), ctx.vocab_start)
        return logits, None


class _VocabParallelCrossEntropy(torch.autograd.Function):
    """Composed-torch fallback (CPU / label smoothing)."""

    @staticmethod
    def forward(ctx, logits, target, label_smoothing=0.0):
        # logits: [tokens, V/tp] (any leading dims flattened by caller), target: [tokens]
        group = G.get_tensor_model_parallel_group()
        tp = G.get_tensor_model_parallel_world_size()
        rank = G.get_tensor_model_parallel_rank()
        part_vocab = logits.size(-1)
        vocab_start = rank * part_vocab
        vocab_end = vocab_start + part_vocab

        logits_max = torch.max(logits, dim=-1)[0]
        if tp > 1:
            dist.all_reduce(logits_max, op=dist.ReduceOp.MAX, group=group)
        shifted = logits.float() - logits_max.unsqueeze(-1).float()

        target_mask = (target < vocab_start) | (target >= vocab_end)
        masked_target = target - vocab_start
        masked_target = masked_target.masked_fill(target_mask, 0)
        # gather predicted (shifted) logit for the target
        pred = shifted.gather(-1, masked_target.unsqueeze(-1)).squeeze(-1)
        pred = pred.masked_fill(target_mask, 0.0)

        exp_logits = torch.exp(shifted)
        sum_exp = exp_logits.sum(dim=-1)
        if tp > 1:
            dist.all_reduce(pred, op=dist.ReduceOp.SUM, group=group)
            dist.all_reduce(sum_exp, op=dist.ReduceOp.SUM, group=group)

        loss = torch.log(sum_exp) - pred

        softmax = exp_logits.div_(sum_exp.unsqueeze(-1))

        vocab_size = part_vocab * tp
        if label_smoothing > 0:
            smoothing = label_smoothing * vocab_size / (vocab_size - 1)
            log_probs = torch.log(softmax.clamp(min=1e-20))
            mean_log_probs = log_probs.mean(dim=-1)
            if tp > 1:
                dist.all_reduce(mean_log_probs, op=dist.ReduceOp.SUM, group=group)
                mean_log_probs = mean_log_probs / tp
            loss = (1.0 - smoothing) * loss - smoothing * mean_log_probs
        ctx.label_smoothing = label_smoothing
        ctx.vocab_size = vocab_size
        ctx.logits_dtype = logits.dtype
        ctx.save_for_backward(softmax, target_mask, masked_target)
        return loss

    @staticmethod
    def backward(ctx, grad_output):
        softmax, target_mask, masked_target = ctx.saved_tensors
        grad = softmax  # in-place reuse; [tokens, V/tp] fp32
        rows = torch.arange(grad.size(0), device=grad.device)
        update = (~target_mask).float()
        if ctx.label_smoothing > 0:
            smoothing = ctx.label_smoothing * ctx.vocab_size / (ctx.vocab_size - 1)
            grad.sub_(smoothing / ctx.vocab_size)
            grad[rows, masked_target] -= (1.0 - smoothing) * update
        else:
            grad[rows, masked_target] -= update
        grad.mul_(grad_output.unsqueeze(-1))
        return grad.to(ctx.logits_dtype), None, None


def vocab_parallel_cross_entropy(logits: torch.Tensor, target: torch.Tensor, label_smoothing: float = 0.0):
    """logits [s, b, V/tp], target [s, b] -> loss [s, b] (fp32)."""
    from megatron_amd import ops

    s, b, v = logits.shape
    flat = logits.reshape(s * b, v)
    tgt = target.reshape(-1)
    if (
        label_smoothing == 0.0
        and flat.is_cuda
        and flat.dtype == torch.bfloat16
        and v % 8 == 0
        and ops.has_native()
    ):
        loss = _VocabParallelCrossEntropyNative.apply(flat, tgt)
    else:
        loss = _VocabParallelCrossEntropy.apply(flat, tgt, label_smoothing)
    return loss.view(s, b)
