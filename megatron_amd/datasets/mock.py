"""Synthetic token stream for benchmarking/tests (reference MockGPTDataset
analog, megatron/core/datasets/gpt_dataset.py): deterministic per (seed,
dp_rank) random tokens of the benchmark shape; no files, no network."""

from __future__ import annotations

import torch


class MockGPTDataIterator:
    def __init__(self, micro_batch_size: int, seq_length: int, vocab_size: int,
                 seed: int = 1234, device: str = "cpu", dp_rank: int = 0):
        self.mbs = micro_batch_size
        self.seq = seq_length
        self.vocab = vocab_size
        self.device = device
        self.gen = torch.Generator().manual_seed(seed + 101 * dp_rank)

    def __iter__(self):
        return self

    def __next__(self):
        tokens = torch.randint(
            0, self.vocab, (self.mbs, self.seq + 1), generator=self.gen
        )
        batch = {
            "tokens": tokens[:, :-1].to(self.device, non_blocking=True),
            "labels": tokens[:, 1:].to(self.device, non_blocking=True),
        }
        batch["loss_mask"] = torch.ones(self.mbs, self.seq, device=self.device)
        return batch
