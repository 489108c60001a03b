"""Elastic-width (Flextron-style) networks.

Capability analog of reference megatron/elastification/ (4,781 LoC): train
one network whose linear layers expose nested sub-networks — the first
`frac` of their rows/columns form a smaller, deployable model sharing the
full model's weights.  A width router (or a static profile) picks the
active fraction per layer; at export time the active slice is materialized
as a standalone dense model.

This module provides the mechanism (elastic linears, width switching,
memory profiles); training recipes (sandwich sampling, distillation from
the full net — see post_training.distillation) compose on top.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.nn as nn


class ElasticLinear(nn.Module):
    """Linear whose active in/out features are runtime-selectable prefixes.

    Weight is stored at full size; forward uses weight[:out_active, :in_active].
    Gradients flow only into the active slice, which is exactly Flextron's
    weight-sharing semantics."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True,
                 dtype=torch.float32):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features, dtype=dtype))
        nn.init.kaiming_uniform_(self.weight, a=5 ** 0.5)
        self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype)) if bias else None
        self.out_active = out_features
        self.in_active = in_features

    def set_active(self, out_frac: float = 1.0, in_frac: float = 1.0):
        self.out_active = max(1, int(round(self.out_features * out_frac)))
        self.in_active = max(1, int(round(self.in_features * in_frac)))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        w = self.weight[: self.out_active, : self.in_active]
        b = self.bias[: self.out_active] if self.bias is not None else None
        return torch.nn.functional.linear(x[..., : self.in_active], w, b)

    def materialize(self) -> nn.Linear:
        """Export the active sub-network as a plain Linear."""
        m = nn.Linear(self.in_active, self.out_active, bias=self.bias is not None)
        with torch.no_grad():
            m.weight.copy_(self.weight[: self.out_active, : self.in_active])
            if self.bias is not None:
                m.bias.copy_(self.bias[: self.out_active])
        return m


def set_active_width(model: nn.Module, frac: float,
                     per_layer: Optional[Dict[str, float]] = None) -> int:
    """Set every ElasticLinear's width fraction (hidden dims elastic, the
    model's input/output interfaces stay full: in_frac of the first elastic
    layer and out_frac of the last are the caller's business via per_layer).
    Returns the number of layers switched."""
    n = 0
    for name, m in model.named_modules():
        if isinstance(m, ElasticLinear):
            f = per_layer.get(name, frac) if per_layer else frac
            m.set_active(out_frac=f, in_frac=f)
            n += 1
    return n


def elastic_memory_profile(model: nn.Module, fracs: List[float]) -> Dict[float, int]:
    """Active-parameter count at each width fraction (reference memory
    profiles used to pick deployable sub-nets for a memory budget)."""
    out = {}
    saved = [(m, m.out_active, m.in_active) for m in model.modules() if isinstance(m, ElasticLinear)]
    for f in fracs:
        set_active_width(model, f)
        n = 0
        for m in model.modules():
            if isinstance(m, ElasticLinear):
                n += m.out_active * m.in_active
                if m.bias is not None:
                    n += m.out_active
            elif isinstance(m, nn.Linear):
                n += m.weight.numel() + (m.bias.numel() if m.bias is not None else 0)
        out[f] = n
    for m, oa, ia in saved:
        m.out_active, m.in_active = oa, ia
    return out
