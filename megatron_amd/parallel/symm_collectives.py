"""Symmetric-memory one-shot collectives for TP decode latency.

Capability analog of reference inference/communication/torch_symm_triton/
(one-shot allreduce over multimem/symmetric memory): for the tiny [b, h]
activations of a TP decode step, RCCL ring latency dominates; with every
rank's buffer peer-mapped over xGMI (hipIpc, dmabuf mode —
HSA_ENABLE_IPC_MODE_LEGACY=0), one kernel does flag-barrier + fan-in sum in
a few microseconds (kernel: ops/csrc/symm_allreduce.hip).

Usage: construct once per TP group after init; `all_reduce(t)` returns the
summed tensor.  Falls back to dist.all_reduce transparently when IPC
mapping is unavailable (single rank, non-CUDA tensors, or mapping failure).
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


class SymmetricAllReduce:
    def __init__(self, group=None, max_bytes: int = 1 << 20, device=None):
        from megatron_amd import ops

        self._C = ops._C if hasattr(ops, "_C") else None
        self.group = group
        self.world = dist.get_world_size(group) if dist.is_initialized() else 1
        self.rank = dist.get_rank(group) if dist.is_initialized() else 0
        self.max_bytes = max_bytes
        self.seq = 0
        self.enabled = False
        self._peer_ptrs: List[int] = []
        self._opened: List[int] = []
        if (self.world > 1 and torch.cuda.is_available() and self._C is not None
                and hasattr(self._C, "symm_ipc_handle")):
            try:
                self._setup(device or torch.device("cuda", torch.cuda.current_device()))
                self.enabled = True
            except Exception:
                self._teardown()

    def _setup(self, device):
        flags_bytes = 8 * 16 * 4  # MAX_WORLD slots, 64B apart
        self._buf = torch.zeros(self.max_bytes + flags_bytes, dtype=torch.uint8,
                                device=device)
        handle = self._C.symm_ipc_handle(self._buf)
        gathered: List[Optional[list]] = [None] * self.world
        dist.all_gather_object(gathered, handle, group=self.group)
        for r, h in enumerate(gathered):
            if r == self.rank:
                self._peer_ptrs.append(self._buf.data_ptr())
            else:
                ptr = self._C.symm_open_handle(h)
                self._peer_ptrs.append(ptr)
                self._opened.append(ptr)
        dist.barrier(group=self.group)

    def _teardown(self):
        for p in self._opened:
            try:
                self._C.symm_close_handle(p)
            except Exception:
                pass
        self._opened = []
        self._peer_ptrs = []
        self.enabled = False

    def eligible(self, t: torch.Tensor) -> bool:
        return (self.enabled and t.is_cuda and t.is_contiguous()
                and t.dtype in (torch.bfloat16, torch.float32)
                and t.numel() * t.element_size() <= self.max_bytes)

    def all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if not self.eligible(t):
            if dist.is_initialized() and self.world > 1:
                dist.all_reduce(t, group=self.group)
            return t
        self.seq += 1
        out = torch.empty_like(t)
        self._C.symm_allreduce(self._peer_ptrs, self.max_bytes, t, out,
                               self.rank, self.seq)
        return out

    def __del__(self):
        self._teardown()
