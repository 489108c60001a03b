"""Audio-language multimodal model.

Capability analog of reference megatron/core/models/audio/ (audio_projector.py
AudioProjection, packed_audio.py): mel/SSL feature frames are time-stacked
(stack_factor consecutive frames concatenated per step — the usual 4-8x
subsampling), projected into the language hidden size, and spliced into the
token stream at an audio placeholder token, LLaVA-style.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from megatron_amd.models.gpt import GPTModel
from megatron_amd.parallel.cross_entropy import vocab_parallel_cross_entropy

DEFAULT_AUDIO_TOKEN_INDEX = -300


class AudioFeatureProjector(nn.Module):
    """[b, t, feat] -> [b, ceil(t/stack), h_lang]: stack `stack_factor`
    consecutive frames, then a 2-layer GELU MLP projector."""

    def __init__(self, feat_dim: int, language_hidden: int, stack_factor: int = 4,
                 dtype=torch.float32):
        super().__init__()
        assert stack_factor >= 1
        self.stack_factor = stack_factor
        self.fc1 = nn.Linear(feat_dim * stack_factor, language_hidden, dtype=dtype)
        self.fc2 = nn.Linear(language_hidden, language_hidden, dtype=dtype)

    def forward(self, feats: torch.Tensor) -> torch.Tensor:
        b, t, f = feats.shape
        k = self.stack_factor
        pad = (-t) % k
        if pad:
            feats = torch.cat([feats, feats.new_zeros(b, pad, f)], dim=1)
        stacked = feats.reshape(b, (t + pad) // k, f * k)
        return self.fc2(torch.nn.functional.gelu(
            self.fc1(stacked.to(self.fc1.weight.dtype))))


class AudioLanguageModel(nn.Module):
    """GPT language model with audio feature segments spliced at the audio
    placeholder token (one segment per sample, mirroring llava.py v1)."""

    def __init__(self, language_config, feat_dim: int = 80, stack_factor: int = 4,
                 audio_token_index: int = DEFAULT_AUDIO_TOKEN_INDEX,
                 freeze_language_model: bool = False):
        super().__init__()
        self.audio_token_index = audio_token_index
        self.language_model = GPTModel(language_config)
        self.config = language_config
        self.audio_projector = AudioFeatureProjector(
            feat_dim, language_config.hidden_size, stack_factor,
            dtype=language_config.params_dtype)
        if freeze_language_model:
            for p in self.language_model.parameters():
                p.requires_grad = False

    def forward(self, audio_feats: Optional[torch.Tensor] = None,
                input_ids: torch.Tensor = None, labels: Optional[torch.Tensor] = None):
        """audio_feats [b, t, feat]; input_ids/labels [b, s] with one
        audio_token_index per sample when audio is present.  Returns
        per-token loss [s', b] (audio positions zeroed) or logits."""
        lm = self.language_model
        if audio_feats is None:
            return lm(input_ids.clamp(min=0), labels=labels)
        audio_embeds = self.audio_projector(audio_feats)  # [b, a, h]
        b, s = input_ids.shape
        a = audio_embeds.shape[1]
        out_len = s - 1 + a
        text_emb = lm.embedding(input_ids.clamp(min=0)).transpose(0, 1)  # [b, s, h]
        mixed = torch.empty(b, out_len, text_emb.shape[-1], dtype=text_emb.dtype,
                            device=text_emb.device)
        for i in range(b):
            pos = (input_ids[i] == self.audio_token_index).nonzero(as_tuple=True)[0]
            assert pos.numel() == 1, "exactly one audio token per sample (v1)"
            p = int(pos[0])
            mixed[i, :p] = text_emb[i, :p]
            mixed[i, p : p + a] = audio_embeds[i].to(text_emb.dtype)
            mixed[i, p + a :] = text_emb[i, p + 1 :]
        hidden = mixed.transpose(0, 1).contiguous()
        rotary = lm._rotary_freqs(hidden.shape[0], hidden.device)
        hidden = lm.decoder(hidden, rotary_freqs=rotary)
        logits, _ = lm.output_layer(hidden)
        if labels is None:
            return logits
        full_labels = torch.zeros(b, out_len, dtype=labels.dtype, device=labels.device)
        for i in range(b):
            p = int((input_ids[i] == self.audio_token_index).nonzero(as_tuple=True)[0][0])
            full_labels[i, :p] = labels[i, :p]
            full_labels[i, p + a :] = labels[i, p + 1 :]
        loss = vocab_parallel_cross_entropy(logits, full_labels.transpose(0, 1).contiguous())
        mask = (full_labels != 0).transpose(0, 1).to(loss.dtype)
        return loss * mask
