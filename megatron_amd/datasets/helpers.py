"""Index-building helpers: native C++ module (megatron_amd/datasets/csrc/
data_helpers.cpp, built in-tree by setup.py) with a pure-numpy fallback of the
same semantics (reference analog: megatron/core/datasets/helpers.cpp:77,:145)."""

from __future__ import annotations

import numpy as np

try:
    from megatron_amd.datasets import _data_helpers as _native
except ImportError:
    _native = None


def has_native() -> bool:
    return _native is not None


def build_sample_idx(sizes: np.ndarray, doc_idx: np.ndarray, seq_length: int,
                     num_epochs: int, tokens_per_epoch: int) -> np.ndarray:
    """[num_samples+1, 2] int64 of (doc_idx position, token offset) per sample."""
    if _native is not None:
        return _native.build_sample_idx(
            np.ascontiguousarray(sizes, dtype=np.int32),
            np.ascontiguousarray(doc_idx, dtype=np.int32),
            seq_length, num_epochs, tokens_per_epoch)
    return _build_sample_idx_py(sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch)


def _build_sample_idx_py(sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch):
    num_samples = (num_epochs * tokens_per_epoch - 1) // seq_length
    out = np.zeros((num_samples + 1, 2), dtype=np.int64)
    doc_cursor = 0
    offset = 0
    for i in range(1, num_samples + 1):
        remaining = seq_length + 1
        while remaining > 0:
            doc_len = int(sizes[doc_idx[doc_cursor]]) - offset
            if doc_len >= remaining:
                offset += remaining - 1
                remaining = 0
            else:
                remaining -= doc_len
                doc_cursor += 1
                offset = 0
        out[i, 0] = doc_cursor
        out[i, 1] = offset
    return out


def build_blending_indices(weights: np.ndarray, size: int):
    """Greedy error-minimizing interleave: returns (dataset_index int16[size],
    dataset_sample_index int64[size])."""
    if _native is not None:
        return _native.build_blending_indices(
            np.ascontiguousarray(weights, dtype=np.float64), size)
    w = np.asarray(weights, dtype=np.float64)
    n = len(w)
    di = np.zeros(size, dtype=np.int16)
    dsi = np.zeros(size, dtype=np.int64)
    taken = np.zeros(n, dtype=np.int64)
    for i in range(size):
        step = max(i, 1)
        err = w * step - taken
        best = int(np.argmax(err))
        di[i] = best
        dsi[i] = taken[best]
        taken[best] += 1
    return di, dsi
