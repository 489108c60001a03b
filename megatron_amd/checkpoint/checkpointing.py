"""High-level checkpoint save/load with tracker file and exact resume.

Capability analog of reference megatron/training/checkpointing.py
(save_checkpoint :570, load_checkpoint, tracker file :340, rng gather :416).
"""

from __future__ import annotations

import os
from typing import List, Optional

import torch
import torch.distributed as dist

from megatron_amd.checkpoint.sharded import load as sharded_load
from megatron_amd.checkpoint.sharded import save as sharded_save
from megatron_amd.checkpoint.state_dict import model_sharded_state_dict, optimizer_sharded_state_dict
from megatron_amd.parallel.random import get_rng_tracker

TRACKER = "latest_checkpointed_iteration.txt"


def _ckpt_dir(root: str, iteration: int) -> str:
    return os.path.join(root, f"iter_{iteration:07d}")


def save_checkpoint(root: str, models: List, optimizer, iteration: int,
                    scheduler=None, extra: Optional[dict] = None, async_save: bool = False):
    sharded = {}
    param_maps = []
    for i, m in enumerate(models):
        # One canonical namespace regardless of VPP chunk count: layer keys are
        # globally numbered, so chunks never collide and a checkpoint written
        # at any (pp, vpp) layout loads at any other.
        sd, pm = model_sharded_state_dict(m, prefix="model.")
        sharded.update(sd)
        param_maps.append(pm)
    if optimizer is not None:
        sharded.update(optimizer_sharded_state_dict(optimizer, param_maps))
    common = {
        "iteration": iteration,
        "scheduler": scheduler.state_dict() if scheduler is not None else None,
        "optim_steps": [sub.step_count for sub in optimizer.chained_optimizers] if optimizer else None,
        "grad_scaler": [
            sub.grad_scaler.state_dict() if getattr(sub, "grad_scaler", None) is not None else None
            for sub in optimizer.chained_optimizers
        ] if optimizer else None,
        "extra": extra or {},
    }
    path = _ckpt_dir(root, iteration)
    # per-rank RNG state rides in this rank's shard file via a fake sharded key?
    # -> stored separately (valid only for same-layout resume)
    rank = dist.get_rank() if dist.is_initialized() else 0
    os.makedirs(path, exist_ok=True)
    rng_state = {
        "torch": torch.get_rng_state(),
        "cuda": torch.cuda.get_rng_state() if torch.cuda.is_available() else None,
        "tracker": get_rng_tracker().get_states(),
    }
    torch.save(rng_state, os.path.join(path, f"rng_r{rank}.pt"))
    writer = sharded_save(sharded, common, path, async_save=async_save)
    if rank == 0:
        with open(os.path.join(root, TRACKER), "w") as f:
            f.write(str(iteration))
    return writer


def save_non_persistent_checkpoint(local_root: str, models: List, optimizer,
                                   iteration: int, scheduler=None,
                                   extra: Optional[dict] = None, retain: int = 1):
    """Fast in-job restart checkpoints on node-local storage (SSD/ramdisk)
    — reference checkpointing.py:1528 non-persistent local ckpts.  Same
    format as regular checkpoints; only the newest `retain` iterations are
    kept (older ones deleted after a successful save)."""
    import re
    import shutil

    save_checkpoint(local_root, models, optimizer, iteration, scheduler, extra)
    rank = dist.get_rank() if dist.is_initialized() else 0
    if rank == 0:
        kept = sorted(
            (int(m.group(1)) for d in os.listdir(local_root)
             for m in [re.fullmatch(r"iter_(\d+)", d)] if m),
            reverse=True)
        for it in kept[retain:]:
            shutil.rmtree(os.path.join(local_root, f"iter_{it:07d}"), ignore_errors=True)
    if dist.is_initialized():
        dist.barrier()


def latest_checkpoint_iteration(root: str) -> Optional[int]:
    try:
        with open(os.path.join(root, TRACKER)) as f:
            return int(f.read().strip())
    except (OSError, ValueError):
        return None


def resolve_resume_source(persistent_root: Optional[str],
                          non_persistent_root: Optional[str]):
    """(root, iteration) of the newest available checkpoint across the
    persistent and node-local non-persistent trees (reference: prefer the
    non-persistent copy when it is newer)."""
    candidates = []
    for root in (persistent_root, non_persistent_root):
        if root:
            it = latest_checkpoint_iteration(root)
            if it is not None:
                candidates.append((it, root))
    if not candidates:
        return None, None
    it, root = max(candidates)
    return root, it


def load_checkpoint(root: str, models: List, optimizer, scheduler=None,
                    iteration: Optional[int] = None, load_rng: bool = True,
                    load_optim: bool = True) -> int:
    """load_optim=False (--finetune / --no-load-optim): restore model weights
    only — optimizer state, scheduler and step counts start fresh."""
    if not load_optim:
        optimizer_arg = optimizer
        optimizer = None
        scheduler = None
    if iteration is None:
        with open(os.path.join(root, TRACKER)) as f:
            iteration = int(f.read().strip())
    path = _ckpt_dir(root, iteration)
    sharded = {}
    param_maps = []
    for i, m in enumerate(models):
        # One canonical namespace regardless of VPP chunk count: layer keys are
        # globally numbered, so chunks never collide and a checkpoint written
        # at any (pp, vpp) layout loads at any other.
        sd, pm = model_sharded_state_dict(m, prefix="model.")
        sharded.update(sd)
        param_maps.append(pm)
    if optimizer is not None:
        sharded.update(optimizer_sharded_state_dict(optimizer, param_maps))
    common = sharded_load(sharded, path)
    # propagate loaded shards into runtime state
    if optimizer is not None:
        if common.get("optim_steps"):
            for sub, sc in zip(optimizer.chained_optimizers, common["optim_steps"]):
                sub.step_count = sc
        if common.get("grad_scaler"):
            for sub, gs in zip(optimizer.chained_optimizers, common["grad_scaler"]):
                if gs is not None and getattr(sub, "grad_scaler", None) is not None:
                    sub.grad_scaler.load_state_dict(gs)
        from megatron_amd.optimizer.dist_optimizer import DistributedOptimizer

        for sub in optimizer.chained_optimizers:
            if isinstance(sub, DistributedOptimizer):
                for seg in sub.segments:
                    seg.model_view.copy_(seg.main.to(seg.model_view.dtype))
                for chunk in sub.model_chunks:
                    chunk.start_param_sync()
                    chunk.finish_param_sync()
            elif hasattr(sub, "main_params"):
                for p, mp in zip(sub.params, sub.main_params):
                    p.data.copy_(mp.to(p.dtype))
    if scheduler is not None and common.get("scheduler") is not None:
        scheduler.load_state_dict(common["scheduler"])
    if not load_optim and optimizer_arg is not None:
        # weights changed under the optimizer: refresh fp32 main copies
        for sub in optimizer_arg.chained_optimizers:
            if hasattr(sub, "reload_model_params"):
                sub.reload_model_params()
    rank = dist.get_rank() if dist.is_initialized() else 0
    rng_file = os.path.join(path, f"rng_r{rank}.pt")
    if load_rng and os.path.exists(rng_file):
        rng = torch.load(rng_file, weights_only=False)
        torch.set_rng_state(rng["torch"])
        if rng["cuda"] is not None and torch.cuda.is_available():
            torch.cuda.set_rng_state(rng["cuda"])
        get_rng_tracker().set_states(rng["tracker"])
    return iteration
