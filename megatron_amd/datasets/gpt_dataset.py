"""GPT pretraining dataset over indexed corpora.

Capability analog of reference megatron/core/datasets/gpt_dataset.py
(GPTDataset doc/sample/shuffle index construction) and blended_dataset.py
(weighted corpus mixing), built on our IndexedDataset and the C++ helpers.

Index construction (seeded, deterministic, cached to .npy next to the data):
  doc_idx     — documents of every epoch, shuffled per epoch
  sample_idx  — (doc position, token offset) per sample (C++ build_sample_idx)
  shuffle_idx — random permutation of samples
A sample is seq_length + 1 tokens (inputs + shifted labels share the buffer).
"""

from __future__ import annotations

import hashlib
import os
from typing import Optional, Sequence

import numpy as np
import torch

from megatron_amd.datasets.helpers import build_blending_indices, build_sample_idx
from megatron_amd.datasets.indexed import IndexedDataset


def eod_boundaries(tokens: torch.Tensor, eod: int, seq_length: int,
                   max_docs: int = 64) -> torch.Tensor:
    """Document end offsets within one sample (cumulative, ending at
    seq_length), zero-padded to a fixed width so samples collate.  The
    training loop turns these into varlen cu_seqlens -> block-diagonal
    attention + per-document position restart (reference
    get_ltor_masks_and_position_ids, gpt_dataset.py) — mapped onto the
    flash-varlen path instead of a materialized [s, s] mask."""
    ends = (tokens == eod).nonzero(as_tuple=True)[0] + 1
    cu = ends[ends < seq_length].tolist() + [seq_length]
    cu = cu[: max_docs]
    if cu[-1] != seq_length:
        cu[-1] = seq_length
    out = torch.zeros(max_docs, dtype=torch.long)
    out[: len(cu)] = torch.tensor(cu, dtype=torch.long)
    return out


class GPTDataset(torch.utils.data.Dataset):
    def __init__(self, indexed: IndexedDataset, num_samples: Optional[int],
                 seq_length: int, seed: int = 1234, cache_dir: Optional[str] = None,
                 document_subset: Optional[np.ndarray] = None, name: str = "train",
                 eod: Optional[int] = None, reset_attention_mask: bool = False,
                 eod_mask_loss: bool = False):
        self.indexed = indexed
        self.seq_length = seq_length
        self.seed = seed
        self.name = name
        self.eod = eod
        self.reset_attention_mask = reset_attention_mask
        self.eod_mask_loss = eod_mask_loss
        if document_subset is None:
            document_subset = np.arange(len(indexed.document_indices) - 1, dtype=np.int64)
        self.documents = document_subset
        sizes = indexed.sequence_lengths
        # document -> first-sequence map: sequence i belongs to doc d iff
        # document_indices[d] <= i < document_indices[d+1]; GPT corpora are
        # one-sequence-per-document, enforce that here.
        assert len(indexed.document_indices) - 1 == len(indexed.sequence_lengths), (
            "GPTDataset expects one sequence per document")
        doc_sizes = sizes[self.documents]
        tokens_per_epoch = int(doc_sizes.sum())
        assert tokens_per_epoch > seq_length, "corpus smaller than one sample"
        if num_samples is None:
            num_samples = (tokens_per_epoch - 1) // seq_length
        self.num_samples = num_samples
        num_epochs = max(1, int(np.ceil((num_samples * seq_length + 1) / tokens_per_epoch)))

        key = hashlib.md5(
            f"{indexed.path_prefix}|{name}|{seq_length}|{seed}|{num_epochs}|"
            f"{len(self.documents)}|{self.documents[:16].tobytes().hex()}".encode()
        ).hexdigest()[:16]
        cache_dir = cache_dir or os.path.dirname(indexed.path_prefix + ".bin") or "."
        cache = os.path.join(cache_dir, f"{os.path.basename(indexed.path_prefix)}_{name}_{key}")

        if os.path.exists(cache + "_sample_idx.npy"):
            self.doc_idx = np.load(cache + "_doc_idx.npy", mmap_mode="r")
            self.sample_idx = np.load(cache + "_sample_idx.npy", mmap_mode="r")
            self.shuffle_idx = np.load(cache + "_shuffle_idx.npy", mmap_mode="r")
        else:
            rng = np.random.RandomState(seed)
            doc_idx = np.tile(self.documents, num_epochs).astype(np.int32)
            for e in range(num_epochs):  # shuffle each epoch independently
                rng.shuffle(doc_idx[e * len(self.documents):(e + 1) * len(self.documents)])
            sample_idx = build_sample_idx(sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch)
            total = sample_idx.shape[0] - 1
            shuffle_idx = np.arange(total, dtype=np.int64)
            rng.shuffle(shuffle_idx)
            self.doc_idx, self.sample_idx, self.shuffle_idx = doc_idx, sample_idx, shuffle_idx
            try:
                np.save(cache + "_doc_idx.npy", doc_idx)
                np.save(cache + "_sample_idx.npy", sample_idx)
                np.save(cache + "_shuffle_idx.npy", shuffle_idx)
            except OSError:
                pass  # read-only data dir: rebuild next time
        assert self.sample_idx.shape[0] - 1 >= num_samples, (
            f"only {self.sample_idx.shape[0]-1} samples available, need {num_samples}")

    def __len__(self) -> int:
        return self.num_samples

    def _sample_tokens(self, sample: int) -> np.ndarray:
        doc_a, off_a = self.sample_idx[sample]
        doc_b, off_b = self.sample_idx[sample + 1]
        if doc_a == doc_b:
            return self.indexed.get(int(self.doc_idx[doc_a]), int(off_a),
                                    int(off_b) - int(off_a) + 1)
        parts = [self.indexed.get(int(self.doc_idx[doc_a]), int(off_a))]
        for d in range(int(doc_a) + 1, int(doc_b)):
            parts.append(self.indexed.get(int(self.doc_idx[d])))
        parts.append(self.indexed.get(int(self.doc_idx[doc_b]), 0, int(off_b) + 1))
        return np.concatenate(parts)

    def __getitem__(self, idx: int) -> dict:
        tokens = self._sample_tokens(int(self.shuffle_idx[idx % len(self.shuffle_idx)]))
        tokens = torch.from_numpy(tokens.astype(np.int64))
        assert tokens.numel() == self.seq_length + 1
        out = {
            "tokens": tokens[:-1],
            "labels": tokens[1:],
            "loss_mask": torch.ones(self.seq_length, dtype=torch.float32),
        }
        if self.eod is not None:
            if self.eod_mask_loss:
                out["loss_mask"][out["labels"] == self.eod] = 0.0
            if self.reset_attention_mask:
                out["cu_seqlens"] = eod_boundaries(out["tokens"], self.eod, self.seq_length)
        return out


class BlendedDataset(torch.utils.data.Dataset):
    """Weighted mix of GPTDatasets (reference blended_dataset.py +
    helpers.cpp:77 build_blending_indices)."""

    def __init__(self, datasets: Sequence[torch.utils.data.Dataset],
                 weights: Sequence[float], num_samples: int):
        assert len(datasets) == len(weights) > 0
        w = np.asarray(weights, dtype=np.float64)
        w = w / w.sum()
        self.datasets = list(datasets)
        self.dataset_index, self.dataset_sample_index = build_blending_indices(w, num_samples)
        self.num_samples = num_samples

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx: int):
        d = int(self.dataset_index[idx])
        s = int(self.dataset_sample_index[idx])
        return self.datasets[d][s % len(self.datasets[d])]


def _parse_split(split: str) -> np.ndarray:
    parts = np.array([float(x) for x in split.split(",")], dtype=np.float64)
    return parts / parts.sum()


def build_gpt_datasets(data_paths: Sequence, seq_length: int, seed: int,
                       train_samples: int, split: str = "969,30,1",
                       cache_dir: Optional[str] = None, eod: Optional[int] = None,
                       reset_attention_mask: bool = False, eod_mask_loss: bool = False):
    """data_paths: [prefix] or [w1, prefix1, w2, prefix2, ...]. Returns
    (train, valid, test) datasets; splits partition each corpus by document."""
    if len(data_paths) == 1:
        weights, prefixes = [1.0], [str(data_paths[0])]
    else:
        assert len(data_paths) % 2 == 0, "expect [weight path]* pairs"
        weights = [float(x) for x in data_paths[0::2]]
        prefixes = [str(x) for x in data_paths[1::2]]
    fracs = _parse_split(split)
    out = []
    per_split_counts = [train_samples,
                        max(1, int(train_samples * fracs[1] / max(fracs[0], 1e-9))),
                        max(1, int(train_samples * fracs[2] / max(fracs[0], 1e-9)))]
    names = ["train", "valid", "test"]
    wnorm = np.asarray(weights, dtype=np.float64)
    wnorm = wnorm / wnorm.sum()
    for si in range(3):
        count = per_split_counts[si]
        subsets = []
        for pi, prefix in enumerate(prefixes):
            indexed = IndexedDataset(prefix)
            ndocs = len(indexed.document_indices) - 1
            bounds = np.floor(np.cumsum(np.concatenate([[0.0], fracs])) * ndocs).astype(np.int64)
            docs = np.arange(bounds[si], bounds[si + 1], dtype=np.int64)
            if len(docs) == 0:
                continue
            # each constituent sized to its weighted share (+margin, reference
            # blended builder uses the same 0.5% pad)
            want = count if len(prefixes) == 1 else int(wnorm[pi] * count * 1.005) + 1
            subsets.append(GPTDataset(indexed, want, seq_length, seed,
                                      cache_dir, docs, name=names[si], eod=eod,
                                      reset_attention_mask=reset_attention_mask,
                                      eod_mask_loss=eod_mask_loss))
        if not subsets:
            out.append(None)
        elif len(subsets) == 1:
            out.append(subsets[0])
        else:
            out.append(BlendedDataset(subsets, weights[: len(subsets)], count))
    return tuple(out)


def build_gpt_train_iterator(args, device, dp_rank: int, dp_size: int,
                             start_sample: int = 0, split: str = "train"):
    """Rank-sharded infinite iterator over one split (used by pretrain).
    ``start_sample`` resumes mid-epoch after a checkpoint load."""
    train_samples = args.train_iters * args.global_batch_size
    ds = build_gpt_datasets(args.data_path, args.seq_length, args.seed,
                            train_samples, args.split,
                            cache_dir=getattr(args, "data_cache_path", None),
                            eod=getattr(args, "eod_id", None),
                            reset_attention_mask=getattr(args, "reset_attention_mask", False),
                            eod_mask_loss=getattr(args, "eod_mask_loss", False))
    train = {"train": ds[0], "valid": ds[1] or ds[0], "test": ds[2] or ds[0]}[split]
    sampler = _ShardedSequentialSampler(len(train), args.micro_batch_size, dp_rank, dp_size,
                                        start_sample=start_sample)
    loader = torch.utils.data.DataLoader(
        train, batch_sampler=sampler, num_workers=args.num_workers, pin_memory=device != "cpu")

    def gen():
        while True:
            for batch in loader:
                yield {k: v.to(device, non_blocking=True) for k, v in batch.items()}

    return gen()


class _ShardedSequentialSampler(torch.utils.data.Sampler):
    """Contiguous global-batch order, strided over DP ranks (matches the
    reference MegatronPretrainingSampler semantics)."""

    def __init__(self, total: int, micro_batch: int, dp_rank: int, dp_size: int,
                 start_sample: int = 0):
        self.total, self.mbs, self.dp_rank, self.dp_size = total, micro_batch, dp_rank, dp_size
        self.start = start_sample

    def __iter__(self):
        stride = self.mbs * self.dp_size
        lo = self.dp_rank * self.mbs
        for start in range(self.start, self.total - stride + 1, stride):
            yield list(range(start + lo, start + lo + self.mbs))

    def __len__(self):
        return self.total // (self.mbs * self.dp_size)
