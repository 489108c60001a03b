"""hipGraph capture of transformer layers for training.

Capability analog of reference megatron/core/transformer/cuda_graphs.py
(`CudaGraphManager`, per-layer capture) and full_cuda_graph.py: short
launch-bound inner loops (small models, decode-sized microbatches) replay a
captured hipGraph instead of relaunching every kernel.  torch.cuda.CUDAGraph
IS hipGraph on ROCm.

Unlike `torch.cuda.make_graphed_callables` (whose multi-graph shared-mempool
capture segfaults in capture_end on ROCm 7.2 — measured on MI355X), each
layer here gets its OWN forward and backward graph with default pools:
  * capture-time: warmup on a side stream, then record fwd with grad enabled
    into static input/output buffers, then record bwd as autograd.grad of
    the captured forward's graph (retained) into static grad buffers;
  * run-time: a custom autograd.Function copies into the static input,
    replays the fwd graph, and on backward copies the incoming grad in,
    replays the bwd graph, and hands autograd clones of the static grads
    (params are explicit Function inputs so their grads accumulate
    normally, which keeps DDP bucket hooks outside the graphed region —
    the reference's `external` grad mode).

Scope: static shapes (one (s, b, h) per capture; eager fallback otherwise,
and always under activation recompute or inference contexts).  With DDP,
disable overlap_grad_reduce: replay re-runs the captured kernels (including
fused main_grad accumulation) but NOT python grad-ready callbacks, so
bucket overlap would wait forever — reduce at finish_grad_sync instead.  Rotary
freqs / masks are closed over as static tensors (step-invariant in
pretraining).  The dynamic inference engine has its own decode-step graph
runner (inference/engine.py _DecodeGraphRunner); this module is the
training side.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn


class _LayerGraphs:
    """Captured state for one layer at one shape."""

    def __init__(self, layer: nn.Module, sample: torch.Tensor,
                 rotary_freqs: Optional[torch.Tensor],
                 attention_mask: Optional[torch.Tensor],
                 num_warmup_iters: int = 3):
        self.layer = layer
        self.params: List[torch.nn.Parameter] = [p for p in layer.parameters() if p.requires_grad]

        def fwd(inp):
            return layer(inp, rotary_freqs=rotary_freqs, attention_mask=attention_mask)

        # warmup must run off the default stream (capture prerequisite)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(num_warmup_iters):
                xi = sample.clone().requires_grad_(True)
                out = fwd(xi)
                torch.autograd.grad(out, [xi] + self.params, torch.ones_like(out),
                                    allow_unused=True)
        torch.cuda.current_stream().wait_stream(side)
        for p in self.params:
            p.grad = None
        torch.cuda.synchronize()

        self.static_input = sample.clone().requires_grad_(True)
        self.g_fwd = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_fwd):
            self.static_output = fwd(self.static_input)

        self.static_grad_output = torch.empty_like(self.static_output)
        self.g_bwd = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_bwd):
            grads = torch.autograd.grad(
                self.static_output, [self.static_input] + self.params,
                self.static_grad_output, retain_graph=True, allow_unused=True)
        self.static_grad_input = grads[0]
        self.static_param_grads = grads[1:]


class _GraphedLayerFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, g: _LayerGraphs, x, *params):
        ctx.g = g
        g.static_input.copy_(x.detach())
        g.g_fwd.replay()
        return g.static_output.detach().clone()

    @staticmethod
    def backward(ctx, grad_output):
        g = ctx.g
        g.static_grad_output.copy_(grad_output)
        g.g_bwd.replay()
        dx = g.static_grad_input.clone() if g.static_grad_input is not None else None
        dps = tuple(p.clone() if p is not None else None for p in g.static_param_grads)
        # overlap_grad_reduce coexistence: params whose wgrad is fused into
        # the graph (static grad None -> main_grad accumulated in-replay)
        # never reach autograd's post-accumulate hooks, so fire their DDP
        # grad-ready callbacks here.  The replay is enqueued on the current
        # stream and torch.distributed orders its RCCL stream after it, so
        # the bucket reduce launched by the callback sees the final grads.
        for p, sg in zip(g.params, g.static_param_grads):
            if sg is None:
                cb = getattr(p, "_ddp_grad_ready_cb", None)
                if cb is not None:
                    cb()
        return (None, dx) + dps


class GraphedLayer(nn.Module):
    def __init__(self, graphs: _LayerGraphs):
        super().__init__()
        self.graphs = graphs

    def forward(self, x):
        return _GraphedLayerFn.apply(self.graphs, x, *self.graphs.params)


def capture_block_hip_graphs(block, sample_hidden: torch.Tensor,
                             rotary_freqs: Optional[torch.Tensor] = None,
                             attention_mask: Optional[torch.Tensor] = None,
                             num_warmup_iters: int = 3) -> int:
    """Graph every layer of a TransformerBlock for the given activation
    shape.  Returns the number of captured layers.  The block's forward
    uses the graphs for matching shapes and falls back to eager otherwise
    (and always under activation recompute or inference contexts)."""
    if not torch.cuda.is_available():
        raise RuntimeError("hipGraph capture requires a GPU")
    assert sample_hidden.is_cuda, "sample must live on the device"
    # ROCm 7.2: hipGraph capture_end segfaults if any LIVE autograd graph
    # exists at capture time (measured on MI355X — a dangling eager forward
    # is enough).  Capture at a step boundary and drop dead references here.
    import gc

    gc.collect()
    torch.cuda.synchronize()
    from megatron_amd.moe.moe_layer import MoELayer

    for layer in block.layers:
        for m in layer.modules():
            if isinstance(m, MoELayer):
                raise RuntimeError(
                    "hipGraph capture requires static shapes; MoE token routing "
                    "is data-dependent — capture dense blocks only")
    graphed = []
    for layer in block.layers:
        graphs = _LayerGraphs(layer, sample_hidden, rotary_freqs, attention_mask,
                              num_warmup_iters=num_warmup_iters)
        graphed.append(GraphedLayer(graphs))
    block._graphed_layers = graphed
    block._graph_shape = tuple(sample_hidden.shape)
    return len(graphed)


def graphed_layer_or_none(block, index: int, hidden_states: torch.Tensor,
                          inference_context=None):
    """The block's dispatch hook: the captured callable when usable."""
    graphed = getattr(block, "_graphed_layers", None)
    if graphed is None or inference_context is not None:
        return None
    if tuple(hidden_states.shape) != block._graph_shape:
        return None
    return graphed[index]
