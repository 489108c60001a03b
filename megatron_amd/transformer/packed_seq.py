"""Packed (THD) sequence support.

Capability analog of reference megatron/core/packed_seq_params.py +
the THD paths through TE fused attention (cu_seqlens) and
GPTDataset-side sequence packing: multiple variable-length documents are
concatenated into one [t, 1, h] activation row; attention is block-diagonal
(each document attends only within itself), RoPE positions restart at each
document boundary, and loss masks follow the pack.

The attention math runs through ops.reference.attention_varlen on the torch
path; the MFMA flash kernel processes packs as b=1 with the segment mask
applied via the same varlen path until a dedicated thd kernel variant lands
(tracked as a perf follow-up, not a capability gap — numerics are exact).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Sequence, Tuple

import torch


@dataclass
class PackedSeqParams:
    """cu_seqlens in token offsets, [n_docs + 1], starting at 0."""

    cu_seqlens: torch.Tensor
    max_seqlen: int

    @classmethod
    def from_lengths(cls, lengths: Sequence[int], device=None) -> "PackedSeqParams":
        cu = torch.zeros(len(lengths) + 1, dtype=torch.long, device=device)
        cu[1:] = torch.cumsum(torch.as_tensor(lengths, dtype=torch.long, device=device), 0)
        return cls(cu_seqlens=cu, max_seqlen=max(lengths) if lengths else 0)

    @property
    def total_tokens(self) -> int:
        return int(self.cu_seqlens[-1])

    def segment_ids(self) -> torch.Tensor:
        """[t] document index per token."""
        t = self.total_tokens
        return torch.bucketize(
            torch.arange(t, device=self.cu_seqlens.device),
            self.cu_seqlens[1:-1], right=True,
        )

    def positions(self) -> torch.Tensor:
        """[t] RoPE position per token (restarts at 0 in each document)."""
        t = self.total_tokens
        idx = torch.arange(t, device=self.cu_seqlens.device)
        starts = self.cu_seqlens[self.segment_ids()]
        return idx - starts


def pack_sequences(docs: List[torch.Tensor], seq_length: int,
                   pad_id: int = 0) -> List[dict]:
    """First-fit pack token documents into rows of at most `seq_length`
    tokens (reference GPT packed/THD sample construction).  Returns a list
    of {tokens [seq_length], labels, loss_mask, cu_seqlens} — the tail of
    each row is padding with loss_mask 0 and its own terminal cu entry."""
    packs: List[List[torch.Tensor]] = []
    fills: List[int] = []
    for d in docs:
        d = d[:seq_length]
        placed = False
        for i, f in enumerate(fills):
            if f + d.numel() <= seq_length:
                packs[i].append(d)
                fills[i] += d.numel()
                placed = True
                break
        if not placed:
            packs.append([d])
            fills.append(d.numel())
    out = []
    for segs in packs:
        toks = torch.cat(segs)
        n = toks.numel()
        lengths = [s.numel() for s in segs]
        row = torch.full((seq_length,), pad_id, dtype=torch.long)
        row[:n] = toks
        labels = torch.full((seq_length,), pad_id, dtype=torch.long)
        labels[: n - 1] = toks[1:]  # next-token within the pack; boundaries masked below
        loss_mask = torch.zeros(seq_length)
        loss_mask[: n - 1] = 1.0
        # mask the last token of each doc (its "label" is the next doc's first token)
        cu = [0]
        for L in lengths:
            cu.append(cu[-1] + L)
            if cu[-1] - 1 < seq_length:
                loss_mask[cu[-1] - 1] = 0.0
        if n < seq_length:  # padding segment
            cu.append(seq_length)
        out.append({
            "tokens": row,
            "labels": labels,
            "loss_mask": loss_mask,
            "cu_seqlens": torch.tensor(cu, dtype=torch.long),
        })
    return out
