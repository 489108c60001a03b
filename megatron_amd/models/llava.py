"""LLaVA-style multimodal model: vision encoder + projector + language model.

Capability analog of reference megatron/core/models/multimodal/llava_model.py
(LLaVAModel:57): image tokens (id = image_token_index, default -200) in the
text sequence are replaced by projected vision-encoder patch embeddings; the
combined sequence runs through the causal language model; loss is computed
only on text positions (image positions get label -100 / masked).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from megatron_amd.models.gpt import GPTModel
from megatron_amd.models.vision import CLIPViTModel

DEFAULT_IMAGE_TOKEN_INDEX = -200
IGNORE_INDEX = -100


class MultimodalProjector(nn.Module):
    """2-layer MLP projector (reference vision_projection.py 'mlp' type)."""

    def __init__(self, vision_hidden: int, language_hidden: int, dtype):
        super().__init__()
        self.fc1 = nn.Linear(vision_hidden, language_hidden, dtype=dtype)
        self.fc2 = nn.Linear(language_hidden, language_hidden, dtype=dtype)

    def forward(self, x):
        return self.fc2(torch.nn.functional.gelu(self.fc1(x)))


class LLaVAModel(nn.Module):
    def __init__(
        self,
        language_config,
        vision_config,
        img_h: int = 336,
        img_w: int = 336,
        patch_dim: int = 14,
        image_token_index: int = DEFAULT_IMAGE_TOKEN_INDEX,
        drop_vision_class_token: bool = True,
        freeze_language_model: bool = False,
        freeze_vision_model: bool = False,
    ):
        super().__init__()
        self.image_token_index = image_token_index
        self.drop_vision_class_token = drop_vision_class_token
        self.language_model = GPTModel(language_config)
        self.config = language_config  # training stack (DDP/optimizer) reads .config
        self.img_h, self.img_w = img_h, img_w
        self.vision_model = CLIPViTModel(vision_config, img_h=img_h, img_w=img_w,
                                         patch_dim=patch_dim)
        self.vision_projection = MultimodalProjector(
            vision_config.hidden_size, language_config.hidden_size, language_config.params_dtype
        )
        self.img_seq_len = self.vision_model.num_patches
        if freeze_language_model:
            for p in self.language_model.parameters():
                p.requires_grad = False
        if freeze_vision_model:
            for p in self.vision_model.parameters():
                p.requires_grad = False

    def _embed_mixed(self, input_ids: torch.Tensor, image_embeds: Optional[torch.Tensor]):
        """Replace each image-token position with img_seq_len patch embeddings.

        input_ids [b, s] may contain image_token_index once per sample;
        image_embeds [b, img_seq_len, h_lang].  Returns (embeddings [s', b, h],
        text_position_mask [b, s']) with s' = s - 1 + img_seq_len.
        """
        b, s = input_ids.shape
        emb_table = self.language_model.embedding
        if image_embeds is None:
            safe = input_ids.clamp(min=0)
            return emb_table(safe), torch.ones_like(input_ids, dtype=torch.bool)
        out_len = s - 1 + self.img_seq_len
        text_ids = input_ids.clamp(min=0)
        text_emb = emb_table(text_ids)  # [s, b, h]
        text_emb = text_emb.transpose(0, 1)  # [b, s, h]
        mixed = torch.empty(b, out_len, text_emb.shape[-1], dtype=text_emb.dtype,
                            device=text_emb.device)
        text_mask = torch.zeros(b, out_len, dtype=torch.bool, device=text_emb.device)
        for i in range(b):
            pos = (input_ids[i] == self.image_token_index).nonzero(as_tuple=True)[0]
            assert pos.numel() == 1, "exactly one image token per sample (v1)"
            p = int(pos[0])
            mixed[i, :p] = text_emb[i, :p]
            mixed[i, p : p + self.img_seq_len] = image_embeds[i]
            mixed[i, p + self.img_seq_len :] = text_emb[i, p + 1 :]
            text_mask[i, :p] = True
            text_mask[i, p + self.img_seq_len :] = True
        return mixed.transpose(0, 1).contiguous(), text_mask

    def forward(
        self,
        images: Optional[torch.Tensor] = None,
        input_ids: Optional[torch.Tensor] = None,
        labels: Optional[torch.Tensor] = None,
        loss_mask: Optional[torch.Tensor] = None,
    ):
        """images [b, c, H, W]; input_ids/labels [b, s] with one
        image_token_index per sample.  Returns per-token loss [s', b] over the
        expanded sequence (image positions zeroed) or logits."""
        image_embeds = None
        if images is not None:
            feats = self.vision_model(images)  # [b, s_v, h_v]
            if self.drop_vision_class_token and self.vision_model.add_class_token:
                feats = feats[:, self.vision_model.class_token_len :]
            image_embeds = self.vision_projection(feats)  # [b, img_seq_len, h_lang]
        hidden, text_mask = self._embed_mixed(input_ids, image_embeds)

        lm = self.language_model
        rotary = lm._rotary_freqs(hidden.shape[0], hidden.device)
        hidden = lm.decoder(hidden, rotary_freqs=rotary)
        logits, _ = lm.output_layer(hidden)  # [s', b, V/tp]
        if labels is None:
            return logits

        # expand labels to the mixed sequence: image positions -> ignore
        if image_embeds is None:
            full_labels = labels
        else:
            b, s = labels.shape
            out_len = logits.shape[0]
            full_labels = torch.zeros(b, out_len, dtype=labels.dtype, device=labels.device)
            for i in range(b):
                pos = (input_ids[i] == self.image_token_index).nonzero(as_tuple=True)[0]
                p = int(pos[0])
                full_labels[i, :p] = labels[i, :p]
                full_labels[i, p + self.img_seq_len :] = labels[i, p + 1 :]
        from megatron_amd.parallel.cross_entropy import vocab_parallel_cross_entropy

        loss = vocab_parallel_cross_entropy(logits, full_labels.transpose(0, 1).contiguous())
        return loss * text_mask.transpose(0, 1).to(loss.dtype)
