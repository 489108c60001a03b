"""Combined 1F1B: layer-granular co-scheduling of forward and backward.

Capability analog of reference megatron/core/pipeline_parallel/
combined_1f1b.py:35 (+ pipeline_parallel/utils.py `AbstractSchedulePlan` /
`ScheduleNode`, models/common/model_chunk_schedule_plan.py): in the 1F1B
steady state, the forward of microbatch i and the backward of microbatch
i-1 are both resident; instead of running them whole-model-at-a-time, each
is decomposed into per-layer ScheduleNodes and executed interleaved —
fwd layer k of microbatch i, then bwd layer N-1-k of microbatch i-1, ...

Why: with MoE layers, a forward node's EP all-to-all (already launched on
the dedicated comm stream by MoELayer) overlaps the *backward* node's
GEMMs instead of idling the GPU, and vice versa — the reference needs
NVSHMEM+DeepEP for this; on one MI355X node the a2a is a single-hop xGMI
transfer on its own stream, so plain interleaving exposes the overlap.

Mechanics: a ScheduleNode detaches its input (requires_grad), runs the
layer, and keeps (input, output); node.backward(gout) runs
torch.autograd.backward on just that segment and hands back input.grad —
manual chain rule over the layer boundary, numerically identical to one
whole backward (same ops, same order within a layer).
"""

from __future__ import annotations

from typing import Callable, List, Optional

import torch


class ScheduleNode:
    """One schedulable segment (a transformer layer or embedding/head)."""

    def __init__(self, fn: Callable, name: str = ""):
        self.fn = fn
        self.name = name
        self._inp: Optional[torch.Tensor] = None
        self._out: Optional[torch.Tensor] = None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        self._inp = x.detach().requires_grad_(x.requires_grad or x.is_floating_point())
        self._out = self.fn(self._inp)
        return self._out

    def backward(self, grad_output: Optional[torch.Tensor]) -> Optional[torch.Tensor]:
        if self._out is None:
            raise RuntimeError(f"backward before forward on node {self.name!r}")
        if grad_output is None:  # loss node: scalar output
            torch.autograd.backward(self._out)
        else:
            torch.autograd.backward(self._out, grad_output)
        g = self._inp.grad
        self._inp = self._out = None  # free activations
        return g


class ModelChunkSchedulePlan:
    """Decomposes a model chunk into nodes: [pre] + one per layer + [post].

    `pre_fn` / `post_fn` wrap embedding / final-norm+head+loss; layer nodes
    call decoder layers with the step-invariant kwargs closed over."""

    def __init__(self, layers: List[Callable], pre_fn: Optional[Callable] = None,
                 post_fn: Optional[Callable] = None):
        self.nodes: List[ScheduleNode] = []
        if pre_fn is not None:
            self.nodes.append(ScheduleNode(pre_fn, "pre"))
        for i, layer in enumerate(layers):
            self.nodes.append(ScheduleNode(layer, f"layer{i}"))
        if post_fn is not None:
            self.nodes.append(ScheduleNode(post_fn, "post"))

    @classmethod
    def from_gpt(cls, model, rotary_freqs=None, attention_mask=None,
                 loss_fn: Optional[Callable] = None):
        """Plan for a full (pp=1) GPT chunk: embedding node + layer nodes +
        final-norm/head/loss node.  Under pipeline parallelism the pre/post
        nodes exist only on the first/last stage (reference
        model_chunk_schedule_plan.py)."""
        core = model.module if hasattr(model, "module") else model

        layer_fns = [
            (lambda h, _l=l: _l(h, rotary_freqs=rotary_freqs, attention_mask=attention_mask))
            for l in core.decoder.layers
        ]

        def pre(tokens):
            return core.embedding(tokens)

        def post(h):
            h = core.decoder.final_layernorm(h)
            logits, _ = core.output_layer(h)
            return loss_fn(logits) if loss_fn is not None else logits

        return cls(layer_fns, pre_fn=pre if core.pre_process else None,
                   post_fn=post if core.post_process else None)

    def forward_node(self, k: int, x: torch.Tensor) -> torch.Tensor:
        return self.nodes[k].forward(x)

    def backward_node(self, k: int, g: Optional[torch.Tensor]) -> Optional[torch.Tensor]:
        return self.nodes[k].backward(g)

    def __len__(self):
        return len(self.nodes)


def combined_1f1b_step(fwd_plan: ModelChunkSchedulePlan, fwd_input: torch.Tensor,
                       bwd_plan: Optional[ModelChunkSchedulePlan] = None,
                       bwd_grad: Optional[torch.Tensor] = None):
    """One steady-state slot: forward `fwd_plan` on `fwd_input` while
    draining `bwd_plan`'s backward, node-interleaved.  Returns
    (fwd_output, bwd_input_grad)."""
    n_f = len(fwd_plan)
    n_b = len(bwd_plan) if bwd_plan is not None else 0
    steps = max(n_f, n_b)
    x = fwd_input
    g = bwd_grad
    for k in range(steps):
        if k < n_f:
            x = fwd_plan.forward_node(k, x)
        if bwd_plan is not None and k < n_b:
            g = bwd_plan.backward_node(n_b - 1 - k, g)
    return (x if n_f else None), (g if bwd_plan is not None else None)


def forward_backward_no_pipelining_combined(
    *,
    forward_step_func,  # unused: the combined path owns the batch->loss plumbing
    data_iterator,
    model,
    num_microbatches: int,
    seq_length: int = None,
    micro_batch_size: int = None,
    forward_only: bool = False,
    **kw,
):
    """pp=1 schedule with layer-granular fwd/bwd co-scheduling
    (config.overlap_moe_expert_parallel_comm): the forward of microbatch i
    runs node-interleaved with the backward of microbatch i-1, so each MoE
    layer's EP all-to-all (on the comm stream) overlaps the neighbor node's
    GEMMs (reference combined_1f1b.py:35).

    Contract: GPT-family model; data_iterator yields {"tokens", "labels"}
    [b, s]; per-token loss averaged over the whole batch (the bench/pretrain
    protocol).  CP/SP stay on the standard path.
    """
    import contextlib

    from megatron_amd.moe.router import AuxLossScaler
    from megatron_amd.parallel.cross_entropy import vocab_parallel_cross_entropy

    if isinstance(model, list):
        assert len(model) == 1
        model = model[0]
    if isinstance(data_iterator, list):
        data_iterator = data_iterator[0]
    core = model.module if hasattr(model, "module") else model
    config = core.config
    assert config.context_parallel_size == 1 and not config.sequence_parallel, (
        "combined 1F1B co-schedule supports the dense/MoE pp=1 path; CP/SP use the standard schedule")

    no_sync = config.no_sync_func
    if no_sync is None and hasattr(model, "no_sync"):
        no_sync = model.no_sync
    if no_sync is None:
        no_sync = contextlib.nullcontext

    losses: List[dict] = []
    device = next(core.parameters()).device
    num_tokens_acc = torch.zeros((), dtype=torch.long, device=device)

    def make_plan(batch):
        tokens, labels = batch["tokens"], batch["labels"]
        seq = tokens.shape[1]
        freqs = core._rotary_freqs(seq, tokens.device)
        labels_sb = labels.transpose(0, 1).contiguous()

        def loss_fn(logits):
            loss_sb = vocab_parallel_cross_entropy(
                logits, labels_sb, label_smoothing=config.label_smoothing)
            ntok = loss_sb.numel()
            ssum = loss_sb.sum()
            losses.append({"loss_sum": ssum.detach()})
            num_tokens_acc.add_(ntok)
            scale = 1.0 / (max(int(ntok), 1) * num_microbatches)
            AuxLossScaler.bind_scale(scale)
            out = ssum * scale
            if config.grad_scale_func is not None:
                out = config.grad_scale_func(out)
            return out

        plan = ModelChunkSchedulePlan.from_gpt(core, rotary_freqs=freqs, loss_fn=loss_fn)
        return plan, tokens

    with no_sync():
        prev_plan = None
        for _ in range(num_microbatches):
            plan, tokens = make_plan(next(data_iterator))
            if forward_only:
                x = tokens
                for k in range(len(plan)):
                    x = plan.forward_node(k, x)
                continue
            combined_1f1b_step(plan, tokens, prev_plan, None)
            prev_plan = plan
        if prev_plan is not None:
            n = len(prev_plan)
            g = None
            for k in range(n):
                g = prev_plan.backward_node(n - 1 - k, g)

    if not forward_only:
        if hasattr(model, "start_grad_sync"):
            model.start_grad_sync()
        if config.finalize_model_grads_func is not None:
            config.finalize_model_grads_func([model], config)
    return losses, num_tokens_acc


def forward_backward_pipelining_without_interleaving_combined(
    *,
    forward_step_func,  # unused: the combined path owns batch->loss plumbing
    data_iterator,
    model,
    num_microbatches: int,
    seq_length: int,
    micro_batch_size: int,
    forward_only: bool = False,
    **kw,
):
    """Non-interleaved 1F1B with layer-granular fwd/bwd co-scheduling in the
    steady state (reference combined_1f1b.py:35): the backward of microbatch
    j is deferred one slot and interleaved node-by-node with the forward of
    microbatch j+warmup+1, so MoE a2a on either side overlaps the other's
    compute.  One extra microbatch of activations stays live vs standard
    1F1B.  Contract: GPT-family chunks; data_iterator yields
    {"tokens", "labels"}.
    """
    import contextlib
    from collections import deque

    from megatron_amd.moe.router import AuxLossScaler
    from megatron_amd.parallel import grid as G
    from megatron_amd.parallel.cross_entropy import vocab_parallel_cross_entropy
    from megatron_amd.pipeline.p2p import P2PCommunicator

    if isinstance(model, list):
        assert len(model) == 1
        model = model[0]
    if isinstance(data_iterator, list):
        data_iterator = data_iterator[0]
    core = model.module if hasattr(model, "module") else model
    config = core.config
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

    losses = []
    device = next(core.parameters()).device
    num_tokens_acc = torch.zeros((), dtype=torch.long, device=device)

    def make_plan(batch):
        tokens = batch["tokens"]
        seq = tokens.shape[1]
        freqs = core._rotary_freqs(seq, tokens.device)
        loss_fn = None
        if is_last:
            labels_sb = batch["labels"].transpose(0, 1).contiguous()

            def loss_fn(logits):
                loss_sb = vocab_parallel_cross_entropy(
                    logits, labels_sb, label_smoothing=config.label_smoothing)
                ntok = loss_sb.numel()
                ssum = loss_sb.sum()
                losses.append({"loss_sum": ssum.detach()})
                num_tokens_acc.add_(ntok)
                scale = 1.0 / (max(int(ntok), 1) * num_microbatches)
                AuxLossScaler.bind_scale(scale)
                out = ssum * scale
                if config.grad_scale_func is not None:
                    out = config.grad_scale_func(out)
                return out

        plan = ModelChunkSchedulePlan.from_gpt(core, rotary_freqs=freqs, loss_fn=loss_fn)
        return plan, tokens

    def _bind_est_scale(out):
        # Non-last stages never see the token count; estimate from the stage
        # output shape [s, b, h] (same rule as pipelined._fwd).
        if not is_last and out is not None and out.dim() >= 2:
            AuxLossScaler.bind_scale(
                1.0 / (max(out.shape[0] * out.shape[1], 1) * num_microbatches))

    def fwd_whole(plan, x):
        for k in range(len(plan)):
            x = plan.forward_node(k, x)
        _bind_est_scale(x)
        return x

    def bwd_whole(plan, g):
        n = len(plan)
        for k in range(n):
            g = plan.backward_node(n - 1 - k, g)
        return g

    num_warmup = min(pp - pp_rank - 1, num_microbatches)
    num_steady = num_microbatches - num_warmup
    plans = deque()

    with no_sync():
        # ---- warmup forwards ----
        for _ in range(num_warmup):
            input_tensor = comm.recv_forward(is_first)
            plan, tokens = make_plan(next(data_iterator))
            out = fwd_whole(plan, tokens if is_first else input_tensor)
            comm.send_forward(out, is_last)
            if not forward_only:
                plans.append(plan)

        pending = None  # (plan, grad) whose backward interleaves the next fwd
        if num_steady > 0:
            input_tensor = comm.recv_forward(is_first)
        for i in range(num_steady):
            plan, tokens = make_plan(next(data_iterator))
            x = tokens if is_first else input_tensor
            if forward_only:
                out = fwd_whole(plan, x)
                comm.send_forward(out, is_last)
                if i < num_steady - 1:
                    input_tensor = comm.recv_forward(is_first)
                continue
            if pending is None:
                out = fwd_whole(plan, x)
            else:
                bplan, bgrad = pending
                out, in_grad = combined_1f1b_step(plan, x, bplan, bgrad)
                _bind_est_scale(out)
                comm.send_backward(in_grad, is_first)
            grad = comm.send_forward_recv_backward(out, is_last)
            plans.append(plan)
            pending = (plans.popleft(), grad)
            if i < num_steady - 1:
                input_tensor = comm.recv_forward(is_first)

        if not forward_only:
            # ---- cooldown: drain the deferred backward, then the warmup ones
            if pending is not None:
                in_grad = bwd_whole(*pending)
                comm.send_backward(in_grad, is_first)
            while plans:
                grad = comm.recv_backward(is_last)
                in_grad = bwd_whole(plans.popleft(), grad)
                comm.send_backward(in_grad, is_first)

    if not forward_only:
        if hasattr(model, "start_grad_sync"):
            model.start_grad_sync()
        if config.finalize_model_grads_func is not None:
            config.finalize_model_grads_func([model], config)
    return losses, num_tokens_acc
