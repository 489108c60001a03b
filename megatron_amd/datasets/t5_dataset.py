"""Span-corruption dataset for T5 pretraining.

Capability analog of reference megatron/core/datasets/t5_dataset.py (whose
sample maps come from helpers.cpp build_blocks_mapping): contiguous spans
covering ~15% of the input (mean span length 3) are replaced by sentinel
tokens in the encoder input; the decoder target is the sequence of
sentinels each followed by the tokens of its span, with standard
teacher-forced shift for the decoder input.
"""

from __future__ import annotations

from typing import List, Tuple

import numpy as np
import torch


def sample_spans(length: int, masking_prob: float, mean_span: float,
                 rng: np.random.RandomState) -> List[Tuple[int, int]]:
    """Draw non-overlapping (start, end) spans covering ~masking_prob of
    `length` positions, geometric span lengths with the given mean."""
    n_mask = max(1, int(round(length * masking_prob)))
    spans: List[Tuple[int, int]] = []
    covered = np.zeros(length, dtype=bool)
    budget = n_mask
    attempts = 0
    while budget > 0 and attempts < 10 * n_mask:
        attempts += 1
        span_len = min(budget, max(1, int(rng.geometric(1.0 / mean_span))))
        start = int(rng.randint(0, max(1, length - span_len)))
        if covered[start : start + span_len].any():
            continue
        covered[start : start + span_len] = True
        spans.append((start, start + span_len))
        budget -= span_len
    spans.sort()
    return spans


class T5SpanCorruptionDataset(torch.utils.data.Dataset):
    """Wraps a token-producing base dataset into (encoder, decoder) samples.

    Sentinels are the last `max_sentinels` ids of the vocabulary (T5
    convention, descending: first span gets vocab_size-1, second
    vocab_size-2, ...).
    """

    def __init__(self, base: torch.utils.data.Dataset, vocab_size: int,
                 bos_id: int, eos_id: int, masking_prob: float = 0.15,
                 mean_span: float = 3.0, max_sentinels: int = 100,
                 seed: int = 1234):
        self.base = base
        self.vocab_size = vocab_size
        self.bos_id = bos_id
        self.eos_id = eos_id
        self.p = masking_prob
        self.mean_span = mean_span
        self.max_sentinels = max_sentinels
        self.seed = seed

    def __len__(self):
        return len(self.base)

    def sentinel(self, i: int) -> int:
        assert i < self.max_sentinels, "too many spans for sentinel vocabulary"
        return self.vocab_size - 1 - i

    def __getitem__(self, idx: int) -> dict:
        tokens = self.base[idx]["tokens"]
        s = tokens.numel()
        rng = np.random.RandomState((self.seed * 2_000_003 + idx) % 2**31)
        spans = sample_spans(s, self.p, self.mean_span, rng)[: self.max_sentinels]

        enc: List[int] = []
        tgt: List[int] = []
        cursor = 0
        for i, (a, b) in enumerate(spans):
            enc.extend(tokens[cursor:a].tolist())
            enc.append(self.sentinel(i))
            tgt.append(self.sentinel(i))
            tgt.extend(tokens[a:b].tolist())
            cursor = b
        enc.extend(tokens[cursor:].tolist())
        tgt.append(self.eos_id)

        enc_t = torch.tensor(enc, dtype=torch.long)
        labels = torch.tensor(tgt, dtype=torch.long)
        dec_in = torch.cat([torch.tensor([self.bos_id]), labels[:-1]])
        loss_mask = torch.ones(labels.numel(), dtype=torch.float32)
        return {
            "encoder_tokens": enc_t,
            "decoder_tokens": dec_in,
            "labels": labels,
            "loss_mask": loss_mask,
        }


def pad_t5_batch(samples: List[dict], enc_len: int, dec_len: int, pad_id: int = 0) -> dict:
    """Collate variable-length span-corruption samples to fixed shapes."""
    B = len(samples)
    out = {
        "encoder_tokens": torch.full((B, enc_len), pad_id, dtype=torch.long),
        "decoder_tokens": torch.full((B, dec_len), pad_id, dtype=torch.long),
        "labels": torch.full((B, dec_len), pad_id, dtype=torch.long),
        "loss_mask": torch.zeros((B, dec_len), dtype=torch.float32),
        "encoder_mask": torch.zeros((B, enc_len), dtype=torch.bool),
    }
    for i, s in enumerate(samples):
        e = min(enc_len, s["encoder_tokens"].numel())
        d = min(dec_len, s["decoder_tokens"].numel())
        out["encoder_tokens"][i, :e] = s["encoder_tokens"][:e]
        out["decoder_tokens"][i, :d] = s["decoder_tokens"][:d]
        out["labels"][i, :d] = s["labels"][:d]
        out["loss_mask"][i, :d] = s["loss_mask"][:d]
        out["encoder_mask"][i, :e] = True
    return out
