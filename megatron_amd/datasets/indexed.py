"""Memory-mapped indexed token dataset, on-disk compatible with the reference
MMIDIDX format (reference megatron/core/datasets/indexed_dataset.py:47 header,
:120 _IndexWriter layout, :597 IndexedDataset) so corpora preprocessed by
either framework interoperate.

Layout of `<prefix>.idx` (all little-endian):
  bytes 0..8    magic b"MMIDIDX\\x00\\x00"
  bytes 9..16   version: u64 == 1
  byte  17      dtype code (see DTYPES)
  bytes 18..25  sequence_count: u64
  bytes 26..33  document_count: u64
  then  int32[sequence_count]  sequence lengths (tokens)
        int64[sequence_count]  byte offset of each sequence in the .bin
        int64[document_count]  sequence index marking each document boundary
        int8 [sequence_count]  (optional, multimodal) per-sequence mode

`<prefix>.bin` is the raw token stream.
"""

from __future__ import annotations

import os
import struct
from typing import Optional

import numpy as np

_INDEX_HEADER = b"MMIDIDX\x00\x00"

# code -> numpy dtype (reference indexed_dataset.py:50 DType enum)
DTYPES = {
    1: np.uint8,
    2: np.int8,
    3: np.int16,
    4: np.int32,
    5: np.int64,
    6: np.float64,
    7: np.float32,
    8: np.uint16,
}
DTYPE_CODES = {v: k for k, v in DTYPES.items()}


def optimal_token_dtype(vocab_size: Optional[int]):
    """Smallest integer dtype that holds token ids of this vocabulary."""
    if vocab_size is not None and vocab_size < 65500:
        return np.uint16
    return np.int32


class IndexedDataset:
    """Read-only mmap view over a (.bin, .idx) pair.

    ``dataset[i]`` -> np.ndarray of the i-th sequence's tokens.
    ``dataset.get(i, offset, length)`` -> a slice of sequence i.
    """

    def __init__(self, path_prefix: str):
        self.path_prefix = path_prefix
        idx_path = path_prefix + ".idx"
        bin_path = path_prefix + ".bin"
        if not (os.path.exists(idx_path) and os.path.exists(bin_path)):
            raise FileNotFoundError(f"indexed dataset not found: {path_prefix}(.bin/.idx)")

        with open(idx_path, "rb") as f:
            header = f.read(9)
            assert header == _INDEX_HEADER, f"bad index header in {idx_path}"
            (version,) = struct.unpack("<Q", f.read(8))
            assert version == 1, f"unsupported index version {version}"
            (code,) = struct.unpack("<B", f.read(1))
            self.dtype = DTYPES[code]
            (self.sequence_count,) = struct.unpack("<Q", f.read(8))
            (self.document_count,) = struct.unpack("<Q", f.read(8))
            offset = f.tell()

        self._idx_mmap = np.memmap(idx_path, mode="r", order="C")
        self.sequence_lengths = np.frombuffer(
            self._idx_mmap, dtype=np.int32, count=self.sequence_count, offset=offset
        )
        offset += self.sequence_lengths.nbytes
        self.sequence_pointers = np.frombuffer(
            self._idx_mmap, dtype=np.int64, count=self.sequence_count, offset=offset
        )
        offset += self.sequence_pointers.nbytes
        self.document_indices = np.frombuffer(
            self._idx_mmap, dtype=np.int64, count=self.document_count, offset=offset
        )
        self._bin_mmap = np.memmap(bin_path, mode="r", order="C")
        self._itemsize = self.dtype().itemsize

    def __len__(self) -> int:
        return int(self.sequence_count)

    def __getitem__(self, idx: int) -> np.ndarray:
        return self.get(idx)

    def get(self, idx: int, offset: int = 0, length: Optional[int] = None) -> np.ndarray:
        ptr = int(self.sequence_pointers[idx]) + offset * self._itemsize
        if length is None:
            length = int(self.sequence_lengths[idx]) - offset
        return np.frombuffer(self._bin_mmap, dtype=self.dtype, count=length, offset=ptr)

    @property
    def num_tokens(self) -> int:
        return int(self.sequence_lengths.sum())


class IndexedDatasetBuilder:
    """Streaming writer for a (.bin, .idx) pair."""

    def __init__(self, path_prefix: str, dtype=np.int32):
        self.path_prefix = path_prefix
        self.dtype = np.dtype(dtype).type
        self._bin = open(path_prefix + ".bin", "wb")
        self._lengths: list[int] = []
        self._doc_indices: list[int] = [0]

    def add_document(self, tokens) -> None:
        arr = np.asarray(tokens, dtype=self.dtype)
        self._bin.write(arr.tobytes(order="C"))
        self._lengths.append(len(arr))
        self._doc_indices.append(len(self._lengths))

    def add_sequence(self, tokens) -> None:
        """Add one sequence without closing the current document."""
        arr = np.asarray(tokens, dtype=self.dtype)
        self._bin.write(arr.tobytes(order="C"))
        self._lengths.append(len(arr))

    def end_document(self) -> None:
        self._doc_indices.append(len(self._lengths))

    def merge(self, other_prefix: str) -> None:
        """Append another on-disk indexed dataset (for parallel preprocessing shards)."""
        other = IndexedDataset(other_prefix)
        assert other.dtype == self.dtype
        with open(other_prefix + ".bin", "rb") as f:
            while True:
                chunk = f.read(1 << 24)
                if not chunk:
                    break
                self._bin.write(chunk)
        base = len(self._lengths)
        self._lengths.extend(int(x) for x in other.sequence_lengths)
        self._doc_indices.extend(base + int(x) for x in other.document_indices[1:])

    def finalize(self) -> None:
        self._bin.close()
        lengths = np.asarray(self._lengths, dtype=np.int32)
        pointers = np.zeros(len(lengths), dtype=np.int64)
        if len(lengths) > 1:
            np.cumsum(lengths[:-1].astype(np.int64) * self.dtype().itemsize, out=pointers[1:])
        with open(self.path_prefix + ".idx", "wb") as f:
            f.write(_INDEX_HEADER)
            f.write(struct.pack("<Q", 1))
            f.write(struct.pack("<B", DTYPE_CODES[self.dtype]))
            f.write(struct.pack("<Q", len(lengths)))
            f.write(struct.pack("<Q", len(self._doc_indices)))
            f.write(lengths.tobytes(order="C"))
            f.write(pointers.tobytes(order="C"))
            f.write(np.asarray(self._doc_indices, dtype=np.int64).tobytes(order="C"))
