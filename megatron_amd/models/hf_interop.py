"""HuggingFace interop modules.

Capability analog of reference megatron/core/models/huggingface/ (module.py,
clip_model.py, qwen_model.py): wrap a `transformers` model so it slots into
this framework's module contract — `set_input_tensor`, grads synchronized
across TP replicas (HF modules are replicated, not sharded), and an
encoder-style forward that yields [s, b, h] hidden states for multimodal
pipelines (LLaVA-style vision/audio towers).

Offline-friendly: models build either from a local `from_pretrained` path or
from an explicit `transformers` config object with random init (no network).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn


class HuggingFaceModule(nn.Module):
    """Base wrapper: replicated-across-TP module with grad averaging.

    Every parameter is tagged ``average_gradients_across_tp_domain`` so
    finalize_model_grads all-reduces them over TP — replicated HF weights
    stay bit-synchronized even when nondeterministic kernels produce
    slightly different grads per rank (reference huggingface/module.py).
    """

    def __init__(self, config):
        super().__init__()
        self.config = config
        self.input_tensor: Optional[torch.Tensor] = None

    def set_input_tensor(self, input_tensor):
        self.input_tensor = input_tensor

    def __setattr__(self, name, value):
        super().__setattr__(name, value)
        if isinstance(value, nn.Module):
            for p in value.parameters(recurse=True):
                p.average_gradients_across_tp_domain = True


class AutoHuggingFaceModel(HuggingFaceModule):
    """Wraps `transformers` AutoModel; builds from a local path OR an
    explicit HF config object (random init — works with no network)."""

    def __init__(self, config, hf_model_name_or_path: Optional[str] = None,
                 hf_config=None):
        super().__init__(config)
        from transformers import AutoModel

        if hf_config is not None:
            self.model = AutoModel.from_config(hf_config)
        else:
            path = hf_model_name_or_path or getattr(
                config, "huggingface_model_name_or_path", None)
            assert path, "need hf_model_name_or_path or hf_config"
            self.model = AutoModel.from_pretrained(path)

    def forward(self, *args, **kwargs):
        return self.model(*args, **kwargs)


class HuggingFaceEncoderTower(HuggingFaceModule):
    """Encoder tower for multimodal pipelines: HF backbone + linear
    projection into the language model's hidden size, emitting [s, b, h]
    (reference clip_model.py role).  `freeze=True` keeps the backbone
    frozen and trains only the projector (the usual LLaVA recipe)."""

    def __init__(self, config, hf_config=None, hf_model_name_or_path: Optional[str] = None,
                 freeze: bool = True):
        super().__init__(config)
        from transformers import AutoModel

        self.backbone = (AutoModel.from_config(hf_config) if hf_config is not None
                         else AutoModel.from_pretrained(hf_model_name_or_path))
        enc_h = getattr(self.backbone.config, "hidden_size")
        self.projector = nn.Linear(enc_h, config.hidden_size,
                                   dtype=config.params_dtype)
        if freeze:
            for p in self.backbone.parameters():
                p.requires_grad_(False)

    def forward(self, *args, **kwargs) -> torch.Tensor:
        out = self.backbone(*args, **kwargs)
        hidden = out.last_hidden_state  # [b, s, enc_h]
        return self.projector(hidden.to(self.projector.weight.dtype)).transpose(0, 1)
