"""Per-layer quantization recipe selection.

Capability analog of reference megatron/core/quantization/quant_config.py
(+ extensions/kitchen.py): a config maps layer-name patterns to quantization
recipes so e.g. all attention/MLP GEMMs run fp8 while the first/last layers
and anything vocab-sized stay bf16.  Recipes resolve against the CDNA4 fp8
path in ops/fp8.py (e4m3/e5m2 MFMA) — 'none' means stay in the params dtype.
"""

from __future__ import annotations

import fnmatch
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple


@dataclass(frozen=True)
class QuantRecipe:
    """One quantization scheme for a matched layer."""

    name: str = "bf16"                 # 'bf16' | 'fp8' | 'fp8-current' | 'mxfp4'
    fmt: str = "hybrid"                # fp8: 'e4m3' | 'e5m2' | 'hybrid' (fwd e4m3 / bwd e5m2)
    granularity: str = "tensor"        # 'tensor' | 'block'
    block_size: int = 32               # for block-scaled formats (mxfp4)

    @property
    def is_quantized(self) -> bool:
        return self.name not in ("bf16", "none", "fp32")


@dataclass
class QuantRecipeConfig:
    """Ordered (pattern, recipe) rules; first match wins, default last."""

    rules: List[Tuple[str, QuantRecipe]] = field(default_factory=list)
    default: QuantRecipe = field(default_factory=QuantRecipe)

    def recipe_for(self, layer_name: str) -> QuantRecipe:
        for pattern, recipe in self.rules:
            if fnmatch.fnmatch(layer_name, pattern):
                return recipe
        return self.default

    @classmethod
    def from_dict(cls, d: dict) -> "QuantRecipeConfig":
        """{'rules': [{'match': 'decoder.layers.*.mlp.*', 'name': 'fp8', ...}],
            'default': {'name': 'bf16'}}"""
        rules = []
        for r in d.get("rules", []):
            r = dict(r)
            pattern = r.pop("match")
            rules.append((pattern, QuantRecipe(**r)))
        default = QuantRecipe(**d.get("default", {}))
        return cls(rules=rules, default=default)


def resolve_layer_recipes(model, config: QuantRecipeConfig) -> Dict[str, QuantRecipe]:
    """Walk the model's linear-like modules and pin a recipe on each
    (module.quant_recipe), returning the resolved name->recipe map."""
    import torch.nn as nn

    from megatron_amd.parallel.layers import ColumnParallelLinear, RowParallelLinear

    resolved: Dict[str, QuantRecipe] = {}
    for name, mod in model.named_modules():
        if isinstance(mod, (nn.Linear, ColumnParallelLinear, RowParallelLinear)):
            recipe = config.recipe_for(name)
            mod.quant_recipe = recipe
            resolved[name] = recipe
    return resolved
