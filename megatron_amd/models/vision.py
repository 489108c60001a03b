"""CLIP-style vision transformer encoder.

Capability analog of reference megatron/core/models/vision/clip_vit_model.py
(CLIPViTModel:26): conv patch embedding + class token + learned positions ->
bidirectional TransformerBlock.  Reuses the same block the language models
use (causal_attention=False), so the MFMA attention kernel serves both
modalities.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from megatron_amd.transformer.block import TransformerBlock


class CLIPViTModel(nn.Module):
    def __init__(self, config, img_h: int = 336, img_w: int = 336, patch_dim: int = 14,
                 in_channels: int = 3, add_class_token: bool = True, class_token_len: int = 1):
        super().__init__()
        config.causal_attention = False
        config.position_embedding_type = "none"  # ViT uses learned positions below
        self.config = config
        assert img_h % patch_dim == 0 and img_w % patch_dim == 0
        self.num_patches = (img_h // patch_dim) * (img_w // patch_dim)
        self.add_class_token = add_class_token
        self.class_token_len = class_token_len if add_class_token else 0
        self.seq_length = self.num_patches + self.class_token_len

        self.conv1 = nn.Conv2d(in_channels, config.hidden_size, kernel_size=patch_dim,
                               stride=patch_dim, bias=False, dtype=config.params_dtype)
        self.position_embeddings = nn.Embedding(self.seq_length, config.hidden_size,
                                                dtype=config.params_dtype)
        if add_class_token:
            self.class_token = nn.Parameter(
                torch.randn(1, self.class_token_len, config.hidden_size,
                            dtype=config.params_dtype) * config.init_method_std
            )
        self.ln_pre = nn.LayerNorm(config.hidden_size, eps=config.layernorm_epsilon,
                                   dtype=config.params_dtype)
        self.decoder = TransformerBlock(config, pre_process=True, post_process=True)

    def forward(self, images: torch.Tensor) -> torch.Tensor:
        """images [b, c, H, W] -> patch features [b, seq, hidden]."""
        x = self.conv1(images.to(self.conv1.weight.dtype))  # [b, h, H/p, W/p]
        x = x.flatten(2).transpose(1, 2)  # [b, patches, hidden]
        if self.add_class_token:
            x = torch.cat([self.class_token.expand(x.shape[0], -1, -1), x], dim=1)
        pos = torch.arange(x.shape[1], device=x.device)
        x = x + self.position_embeddings(pos).unsqueeze(0)
        x = self.ln_pre(x)
        x = x.transpose(0, 1).contiguous()  # [s, b, h] (block layout)
        x = self.decoder(x, rotary_freqs=None)
        return x.transpose(0, 1)  # [b, s, h]
