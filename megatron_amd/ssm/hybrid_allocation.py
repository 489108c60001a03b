"""Hybrid Mamba/attention/MLP layer allocation.

Capability analog of reference
megatron/core/models/hybrid/hybrid_layer_allocation.py (Symbols:14,
pattern_from_ratios:74): a model is described either by an explicit pattern
string ("M" mamba, "*" attention, "-" MLP) or by target attention/MLP
ratios from which a maximally-even pattern is derived.
"""

from __future__ import annotations

from typing import List


class Symbols:
    MAMBA = "M"
    ATTENTION = "*"
    MLP = "-"
    VALID = {MAMBA, ATTENTION, MLP}


def _spread(num_layers: int, count: int) -> List[int]:
    """Indices of `count` layers spread evenly over `num_layers` slots."""
    if count <= 0:
        return []
    # Place each of the `count` picks at the center of its stride-span.
    return sorted({int((i + 0.5) * num_layers / count) for i in range(count)})


def pattern_from_ratios(num_layers: int, attention_ratio: float = 0.0, mlp_ratio: float = 0.0) -> str:
    """Derive a hybrid pattern from target ratios (reference :74).

    Attention layers are spread evenly; MLP layers are then spread evenly
    over the remaining Mamba slots.
    """
    assert 0.0 <= attention_ratio <= 1.0 and 0.0 <= mlp_ratio <= 1.0
    assert attention_ratio + mlp_ratio <= 1.0
    layers = [Symbols.MAMBA] * num_layers
    n_attn = round(attention_ratio * num_layers)
    for i in _spread(num_layers, n_attn):
        layers[i] = Symbols.ATTENTION
    n_mlp = round(mlp_ratio * num_layers)
    mamba_slots = [i for i, s in enumerate(layers) if s == Symbols.MAMBA]
    for j in _spread(len(mamba_slots), n_mlp):
        layers[mamba_slots[j]] = Symbols.MLP
    return "".join(layers)


def allocate_layers(
    num_layers: int,
    override_pattern: str = "",
    attention_ratio: float = 0.0,
    mlp_ratio: float = 0.0,
) -> List[str]:
    """Resolve the per-layer type list for a hybrid model."""
    if override_pattern:
        bad = set(override_pattern) - Symbols.VALID
        if bad:
            raise ValueError(f"invalid layer symbols {bad!r}; valid: {Symbols.VALID}")
        if len(override_pattern) != num_layers:
            raise ValueError(
                f"hybrid override pattern has {len(override_pattern)} layers, model has {num_layers}"
            )
        return list(override_pattern)
    return list(pattern_from_ratios(num_layers, attention_ratio, mlp_ratio))
