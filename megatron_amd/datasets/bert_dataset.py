"""Masked-LM dataset for BERT pretraining.

Capability analog of reference megatron/core/datasets/masked_dataset.py +
bert_dataset.py (whose sample maps come from helpers.cpp build_mapping):
BERT-style dynamic masking over an indexed corpus — 15% of positions
selected; of those 80% -> [MASK], 10% -> random token, 10% kept."""

from __future__ import annotations

import numpy as np
import torch


class BertMaskedDataset(torch.utils.data.Dataset):
    """MLM masking over a base token dataset; with ``sentence_pairs=True``
    each sample becomes the classic NSP pair: [first half | second half]
    where the second half is swapped with another sample's 50% of the time
    (is_next label + tokentype segmentation — reference bert_dataset.py)."""

    def __init__(self, base: torch.utils.data.Dataset, vocab_size: int, mask_id: int,
                 masking_prob: float = 0.15, seed: int = 1234,
                 sentence_pairs: bool = False):
        self.base = base
        self.vocab_size = vocab_size
        self.mask_id = mask_id
        self.p = masking_prob
        self.seed = seed
        self.sentence_pairs = sentence_pairs

    def __len__(self):
        return len(self.base)

    def __getitem__(self, idx: int) -> dict:
        sample = self.base[idx]
        tokens = sample["tokens"].clone()
        rng = np.random.RandomState((self.seed * 1_000_003 + idx) % 2**31)
        s = tokens.numel()
        is_next = 1
        tokentype = torch.zeros(s, dtype=torch.long)
        if self.sentence_pairs:
            half = s // 2
            tokentype[half:] = 1
            if rng.rand() < 0.5:
                other = int(rng.randint(0, len(self.base)))
                if other != idx:
                    is_next = 0
                    tokens[half:] = self.base[other]["tokens"][half:s]
        n_mask = max(1, int(self.p * s))
        pos = torch.from_numpy(rng.choice(s, size=n_mask, replace=False))
        labels = torch.full_like(tokens, 0)
        labels[pos] = tokens[pos]
        loss_mask = torch.zeros(s, dtype=torch.float32)
        loss_mask[pos] = 1.0
        action = rng.rand(n_mask)
        masked = tokens.clone()
        for i, p_ in enumerate(pos.tolist()):
            if action[i] < 0.8:
                masked[p_] = self.mask_id
            elif action[i] < 0.9:
                masked[p_] = int(rng.randint(0, self.vocab_size))
        out = {"tokens": masked, "labels": labels, "loss_mask": loss_mask,
               "tokentype_ids": tokentype}
        if self.sentence_pairs:
            out["is_next"] = torch.tensor(is_next, dtype=torch.long)
        return out
