"""Reused scratch for sequence-parallel all-gathers.

Capability analog of reference megatron/core/utils.py:693
(`GlobalMemoryBuffer`): the SP forward all-gather materializes a [s, b, h]
tensor per column-linear call that is consumed immediately by the GEMM and
never saved for backward (backward re-gathers).  Allocating it fresh every
call pressures the caching allocator (fragmentation across the 288 GB HBM3E
pool under big activations); instead one named, grow-only buffer per
(dtype, name) is reused."""

from __future__ import annotations

from typing import Dict, Tuple

import torch


class GlobalMemoryBuffer:
    def __init__(self):
        self._buffers: Dict[Tuple[str, torch.dtype], torch.Tensor] = {}

    def get_tensor(self, shape, dtype: torch.dtype, name: str,
                   device=None) -> torch.Tensor:
        numel = 1
        for s in shape:
            numel *= s
        key = (name, dtype)
        buf = self._buffers.get(key)
        if buf is None or buf.numel() < numel or (device is not None and buf.device != torch.device(device)):
            buf = torch.empty(numel, dtype=dtype,
                              device=device if device is not None else
                              (buf.device if buf is not None else None))
            self._buffers[key] = buf
        return buf[:numel].view(*shape)


_GLOBAL = GlobalMemoryBuffer()


def get_global_memory_buffer() -> GlobalMemoryBuffer:
    return _GLOBAL
