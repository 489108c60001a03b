"""RCCL-registered memory for communication buffers.

Capability analog of reference megatron/core/nccl_allocator.py (N2 in
SURVEY.md §2.3): DDP grad/param buffers allocated inside a dedicated
memory pool that is REGISTERED with the RCCL communicator
(ncclCommRegister under the hood), letting RCCL use zero-copy /
pre-registered transports on xGMI instead of staging through its internal
buffers.

MI355X-native route: torch 2.10's `torch.cuda.MemPool` +
`ProcessGroupNCCL.register_mem_pool` — no inline-compiled allocator like
the reference needs (its CUDAPluggableAllocator + ncclMemAlloc dance
predates the MemPool API).  Usage:

    with registered_comm_pool(group) as pool:
        buf = torch.empty(n, device="cuda")   # lands in the pool, registered

Falls back to a plain context (allocations behave normally) when CUDA or
the NCCL backend is unavailable, so callers need no branching.
"""

from __future__ import annotations

import contextlib
from typing import Optional

import torch
import torch.distributed as dist


def _nccl_pg(group):
    """The ProcessGroupNCCL backend object of `group`, or None."""
    if not (dist.is_initialized() and torch.cuda.is_available()):
        return None
    try:
        g = group if group is not None else dist.group.WORLD
        backend = g._get_backend(torch.device("cuda"))
        if hasattr(backend, "register_mem_pool"):
            return backend
    except (RuntimeError, AttributeError):
        pass
    return None


class RcclRegisteredPool:
    """Owns a MemPool registered with one process group's communicator."""

    def __init__(self, group=None):
        self.backend = _nccl_pg(group)
        self.pool: Optional[torch.cuda.MemPool] = None
        if self.backend is not None:
            self.pool = torch.cuda.MemPool()
            self.backend.register_mem_pool(self.pool)

    @property
    def active(self) -> bool:
        return self.pool is not None

    @contextlib.contextmanager
    def use(self):
        """Allocations inside land in the registered pool."""
        if self.pool is None:
            yield
            return
        with torch.cuda.use_mem_pool(self.pool):
            yield

    def close(self):
        if self.pool is not None and self.backend is not None:
            try:
                self.backend.deregister_mem_pool(self.pool)
            except RuntimeError:
                pass
        self.pool = None


@contextlib.contextmanager
def registered_comm_pool(group=None):
    """Context manager form: allocate comm buffers inside, auto-deregisters."""
    p = RcclRegisteredPool(group)
    try:
        with p.use():
            yield p
    finally:
        p.close()
