"""Test helpers: single-process grid setup and multi-process (gloo) launcher.

Mirrors the reference's tests/unit_tests/test_utilities.py pattern
(Utils.initialize_model_parallel / fake_initialize_model_parallel) with a
spawn-based gloo runner so world_size>1 paths run on CPU-only CI.
"""

from __future__ import annotations

import os
import socket
from datetime import timedelta
from typing import Callable

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from megatron_amd.parallel import grid as G
from megatron_amd.parallel.random import model_parallel_seed


def init_single(tp: int = 1, pp: int = 1, cp: int = 1, ep: int = 1, seed: int = 1234, world_size: int = 1, rank: int = 0):
    """Single-process grid (no torch.distributed) — the reference's
    fake_initialize_model_parallel analog."""
    G.destroy_model_parallel()
    grid = G.initialize_model_parallel(
        tensor_parallel_size=tp,
        pipeline_parallel_size=pp,
        context_parallel_size=cp,
        expert_parallel_size=ep,
        world_size=world_size,
        rank=rank,
    )
    model_parallel_seed(seed)
    return grid


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _worker(rank, world_size, port, fn, args, kwargs):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world_size, timeout=timedelta(minutes=5))
    torch.manual_seed(1234)
    try:
        fn(rank, world_size, *args, **kwargs)
    finally:
        G.destroy_model_parallel()
        dist.barrier()
        dist.destroy_process_group()


def spawn_dist(fn: Callable, world_size: int = 2, *args, **kwargs):
    """Run fn(rank, world_size, *args) in world_size processes over gloo."""
    port = _free_port()
    mp.spawn(_worker, args=(world_size, port, fn, args, kwargs), nprocs=world_size, join=True)


def assert_close(a: torch.Tensor, b: torch.Tensor, rtol=1e-4, atol=1e-4, msg=""):
    torch.testing.assert_close(a, b, rtol=rtol, atol=atol, msg=msg or None)
