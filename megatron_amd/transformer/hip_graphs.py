"""hipGraph capture of transformer layers for training.

Capability analog of reference megatron/core/transformer/cuda_graphs.py
(`CudaGraphManager`, per-layer capture) and full_cuda_graph.py: short
launch-bound inner loops (small models, decode-sized microbatches) replay a
captured hipGraph instead of relaunching every kernel.  torch.cuda.CUDAGraph
IS hipGraph on ROCm; `torch.cuda.make_graphed_callables` records one
forward and one backward graph per layer, with shared memory pools.

Scope notes (vs the reference's 3.5k-LoC manager):
  * capture the plain module stack BEFORE DDP wrapping (bucket grad hooks
    fire outside the graphed region — same constraint as the reference's
    `external` grad mode);
  * static shapes: one (s, b, h) per capture, the wrapper falls back to
    eager for any other shape;
  * rotary freqs / masks are closed over as static tensors (they are
    step-invariant in pretraining).

The dynamic inference engine has its own decode-step graph runner
(inference/engine.py _DecodeGraphRunner); this module is the training side.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn


class _StaticArgLayer(nn.Module):
    """Close rotary freqs / mask over a layer so its graphed signature is a
    single positional hidden-states tensor."""

    def __init__(self, layer: nn.Module, rotary_freqs: Optional[torch.Tensor],
                 attention_mask: Optional[torch.Tensor]):
        super().__init__()
        self.inner = layer  # shared parameters, not a copy
        self.rotary_freqs = rotary_freqs
        self.attention_mask = attention_mask

    def forward(self, hidden_states):
        return self.inner(hidden_states, rotary_freqs=self.rotary_freqs,
                          attention_mask=self.attention_mask)


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
    wrappers = tuple(
        _StaticArgLayer(layer, rotary_freqs, attention_mask) for layer in block.layers
    )
    sample_args = tuple((sample_hidden.clone().requires_grad_(True),) for _ in wrappers)
    graphed = torch.cuda.make_graphed_callables(
        wrappers, sample_args, num_warmup_iters=num_warmup_iters
    )
    block._graphed_layers = list(graphed)
    block._graph_shape = tuple(sample_hidden.shape)
    return len(block._graphed_layers)


def graphed_layer_or_none(block, index: int, hidden_states: torch.Tensor,
                          inference_context=None):
    """The block's dispatch hook: the captured callable when usable."""
    graphed = getattr(block, "_graphed_layers", None)
    if graphed is None or inference_context is not None:
        return None
    if tuple(hidden_states.shape) != block._graph_shape:
        return None
    return graphed[index]
