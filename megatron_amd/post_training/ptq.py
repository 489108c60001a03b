"""Post-training quantization: calibration -> fake-quant -> export.

Capability analog of the reference's ModelOpt integration role
(megatron/post_training + core/quantization): an offline PTQ flow that

  1. calibrates activation/weight ranges over real batches (forward hooks
     collecting running amax per matched linear),
  2. fake-quantizes weights in place (int8 per-channel or fp8 per-tensor,
     quantize->dequantize so evaluation measures the true quantized
     accuracy with unmodified kernels),
  3. exports a quantized state dict (int8/fp8 payloads + scales) for a
     serving runtime.

Recipes/patterns come from quant_config.QuantRecipeConfig; vocab-sized and
first/last layers are typically left in bf16 by the pattern rules.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.nn as nn

from megatron_amd.post_training.quant_config import QuantRecipe, QuantRecipeConfig


def _linear_types():
    from megatron_amd.parallel.layers import ColumnParallelLinear, RowParallelLinear

    return (nn.Linear, ColumnParallelLinear, RowParallelLinear)


class CalibrationCollector:
    """Forward hooks recording running activation amax per matched module."""

    def __init__(self, model: nn.Module, config: Optional[QuantRecipeConfig] = None):
        self.model = model
        self.config = config or QuantRecipeConfig()
        self.act_amax: Dict[str, torch.Tensor] = {}
        self.samples: Dict[str, int] = {}
        self._handles = []

    def __enter__(self):
        for name, mod in self.model.named_modules():
            if not isinstance(mod, _linear_types()):
                continue
            if not self.config.recipe_for(name).is_quantized:
                continue
            self._handles.append(mod.register_forward_pre_hook(self._hook(name)))
        return self

    def _hook(self, name):
        def hook(mod, args):
            x = args[0]
            amax = x.detach().abs().amax().float().cpu()
            prev = self.act_amax.get(name)
            self.act_amax[name] = amax if prev is None else torch.maximum(prev, amax)
            self.samples[name] = self.samples.get(name, 0) + 1
        return hook

    def __exit__(self, *exc):
        for h in self._handles:
            h.remove()
        self._handles.clear()
        return False


def _fake_quant_int8_per_channel(w: torch.Tensor):
    """w [out, in] -> (dequantized w, scale [out]) symmetric int8."""
    amax = w.detach().abs().amax(dim=1, keepdim=True).clamp(min=1e-8)
    scale = amax / 127.0
    q = torch.clamp(torch.round(w / scale), -127, 127)
    return (q * scale).to(w.dtype), scale.squeeze(1)


def _fake_quant_fp8_per_tensor(w: torch.Tensor):
    from megatron_amd.ops.fp8 import E4M3_MAX

    amax = w.detach().abs().amax().clamp(min=1e-8)
    scale = E4M3_MAX / amax
    q = (w.float() * scale).clamp(-E4M3_MAX, E4M3_MAX).to(torch.float8_e4m3fn)
    return (q.float() / scale).to(w.dtype), scale


def quantize_model_weights(model: nn.Module, config: QuantRecipeConfig,
                           mode: str = "int8") -> Dict[str, torch.Tensor]:
    """Fake-quantize matched linears IN PLACE; returns name -> scale."""
    scales: Dict[str, torch.Tensor] = {}
    with torch.no_grad():
        for name, mod in model.named_modules():
            if not isinstance(mod, _linear_types()):
                continue
            recipe = config.recipe_for(name)
            if not recipe.is_quantized:
                continue
            if mode == "int8":
                wq, s = _fake_quant_int8_per_channel(mod.weight)
            else:
                wq, s = _fake_quant_fp8_per_tensor(mod.weight)
            mod.weight.copy_(wq)
            scales[name] = s
            mod.quant_recipe = recipe
    return scales


def export_quantized_state_dict(model: nn.Module, config: QuantRecipeConfig,
                                mode: str = "int8") -> Dict[str, torch.Tensor]:
    """Real-quantized export: int8 weights + fp32 scales for matched layers,
    original tensors elsewhere (serving-runtime payload)."""
    out: Dict[str, torch.Tensor] = {}
    quant_names = set()
    for name, mod in model.named_modules():
        if isinstance(mod, _linear_types()) and config.recipe_for(name).is_quantized:
            quant_names.add(name + ".weight")
            w = mod.weight.detach()
            if mode == "int8":
                amax = w.abs().amax(dim=1, keepdim=True).clamp(min=1e-8)
                scale = amax / 127.0
                out[name + ".weight"] = torch.clamp(torch.round(w / scale), -127, 127).to(torch.int8)
                out[name + ".weight_scale"] = scale.squeeze(1).float()
            else:
                amax = w.abs().amax().clamp(min=1e-8)
                scale = (448.0 / amax).float()
                out[name + ".weight"] = (w.float() * scale).clamp(-448, 448).to(torch.float8_e4m3fn)
                out[name + ".weight_scale"] = scale
    for name, t in model.state_dict().items():
        if name not in quant_names and name not in out:
            out[name] = t
    return out
