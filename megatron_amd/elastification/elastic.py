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


# ---------------------------------------------------------------------------
# Flextron-style elastic transformer pieces: elastic MLP widths inside a real
# GPT, an input-adaptive width router, sandwich-rule training, and export
# (reference megatron/elastification/ core loop).
# ---------------------------------------------------------------------------


class ElasticMLP(nn.Module):
    """Wraps a dense MLP so its ffn width is a runtime-selectable prefix.

    The full weights stay shared; fraction f uses gate rows [0 : f*ffn],
    up rows [ffn : ffn + f*ffn] of fc1 and columns [0 : f*ffn] of fc2 —
    Flextron's nested sub-network semantics.  TP=1 scope (elastic widths
    and TP sharding compose in the reference via per-shard fractions; out
    of scope here)."""

    def __init__(self, mlp, widths=(0.25, 0.5, 1.0)):
        super().__init__()
        assert getattr(mlp, "gated", False), "elastic MLP expects a gated (swiglu/geglu) MLP"
        self.mlp = mlp
        self.widths = tuple(sorted(widths))
        self.frac = 1.0
        self.ffn = mlp.linear_fc2.weight.shape[1]

    def set_width(self, frac: float):
        self.frac = float(frac)

    def forward(self, hidden_states: torch.Tensor) -> torch.Tensor:
        import torch.nn.functional as F

        k = max(1, int(round(self.ffn * self.frac)))
        w1 = self.mlp.linear_fc1.weight
        w2 = self.mlp.linear_fc2.weight
        w1s = torch.cat([w1[:k], w1[self.ffn : self.ffn + k]], dim=0)
        x = F.linear(hidden_states, w1s)
        x = self.mlp._act(x)
        return F.linear(x, w2[:, :k])


class WidthRouter(nn.Module):
    """Input-adaptive width selection (Flextron router): a tiny classifier
    over the width set from the mean-pooled hidden state; straight-through
    hard choice in training, argmax at eval.  Returns (frac, aux) where aux
    is a differentiable surrogate that lets the router learn."""

    def __init__(self, hidden_size: int, widths=(0.25, 0.5, 1.0), latency_penalty: float = 0.0):
        super().__init__()
        self.widths = tuple(sorted(widths))
        self.proj = nn.Linear(hidden_size, len(self.widths))
        self.latency_penalty = latency_penalty

    def forward(self, hidden_states: torch.Tensor):
        import torch.nn.functional as F

        pooled = hidden_states.float().mean(dim=tuple(range(hidden_states.dim() - 1)))
        logits = self.proj(pooled)
        probs = F.softmax(logits, dim=-1)
        idx = int(probs.argmax())
        # straight-through scale: multiplying the MLP output by
        # (1 - p_sel.detach() + p_sel) routes gradient into the router
        p_sel = probs[idx]
        st_scale = 1.0 - p_sel.detach() + p_sel
        # latency-aware regularizer: expected width fraction
        aux = self.latency_penalty * (probs * torch.tensor(self.widths, dtype=probs.dtype)).sum()
        return self.widths[idx], st_scale, aux


def elastify_gpt(model, widths=(0.25, 0.5, 1.0), with_router: bool = False,
                 latency_penalty: float = 0.0) -> int:
    """Replace every decoder layer's MLP with an ElasticMLP (optionally
    routed).  Returns the number of layers elastified."""
    core = model.module if hasattr(model, "module") else model
    n = 0
    for layer in core.decoder.layers:
        if hasattr(layer, "mlp") and getattr(layer.mlp, "gated", False):
            em = ElasticMLP(layer.mlp, widths)
            if with_router:
                em.router = WidthRouter(core.config.hidden_size, widths, latency_penalty)

                orig_forward = em.forward

                def routed_forward(h, _em=em, _orig=orig_forward):
                    frac, st, aux = _em.router(h)
                    _em.set_width(frac)
                    out = _orig(h) * st.to(h.dtype)
                    _em.router_aux = aux
                    return out

                em.forward = routed_forward
            layer.mlp = em
            n += 1
    return n


def set_gpt_width(model, frac: float) -> int:
    core = model.module if hasattr(model, "module") else model
    n = 0
    for m in core.modules():
        if isinstance(m, ElasticMLP):
            m.set_width(frac)
            n += 1
    return n


def sandwich_step(model, loss_fn, widths=(0.25, 0.5, 1.0), num_random: int = 1, rng=None):
    """Flextron/slimmable sandwich rule: accumulate grads at the largest
    width, the smallest, and `num_random` middle widths.  `loss_fn(model)`
    runs one forward and returns a scalar loss.  Returns {frac: loss}."""
    import random as _random

    rng = rng or _random.Random(0)
    fracs = [max(widths), min(widths)]
    middle = [w for w in widths if min(widths) < w < max(widths)]
    for _ in range(num_random):
        if middle:
            fracs.append(rng.choice(middle))
    out = {}
    for f in fracs:
        set_gpt_width(model, f)
        loss = loss_fn(model)
        loss.backward()
        out[f] = float(loss.detach())
    set_gpt_width(model, max(widths))
    return out


def materialize_gpt(model, frac: float):
    """Export the active sub-network as a standalone dense GPTModel with
    ffn_hidden_size = frac * ffn (weights sliced, not shared)."""
    import dataclasses

    from megatron_amd.models.gpt import GPTModel

    core = model.module if hasattr(model, "module") else model
    cfg = dataclasses.replace(core.config)
    k = max(1, int(round(core.config.ffn_hidden_size * frac)))
    cfg.ffn_hidden_size = k
    new = GPTModel(cfg, pre_process=core.pre_process, post_process=core.post_process)
    # elastified layers nest the dense MLP one level deeper (mlp.mlp.*)
    src = {n.replace(".mlp.mlp.", ".mlp."): p for n, p in core.named_parameters()}
    with torch.no_grad():
        for name, p in new.named_parameters():
            if name not in src:
                continue
            s = src[name]
            if "linear_fc1.weight" in name:
                ffn = core.config.ffn_hidden_size
                p.copy_(torch.cat([s[:k], s[ffn : ffn + k]], dim=0))
            elif "linear_fc2.weight" in name:
                p.copy_(s[:, :k])
            elif p.shape == s.shape:
                p.copy_(s)
    return new
