"""RNG state tracker + activation recompute (checkpointing).

Capability analog of reference megatron/core/tensor_parallel/random.py
(CudaRNGStatesTracker :229, checkpoint/CheckpointFunction :585).

Named RNG streams fork the device RNG so that TP ranks get identical
initialization / dropout where the tensor is replicated and different where
it is sharded.  Works on both HIP devices (torch.cuda on ROCm) and CPU
(gloo-tested paths).
"""

from __future__ import annotations

import contextlib
from typing import Dict, Optional

import torch

_MODEL_PARALLEL_RNG = "model-parallel-rng"
_EXPERT_PARALLEL_RNG = "expert-parallel-rng"
_DATA_PARALLEL_RNG = "data-parallel-rng"


def _get_state(device: Optional[torch.device] = None):
    if torch.cuda.is_available() and (device is None or device.type == "cuda"):
        return torch.cuda.get_rng_state()
    return torch.get_rng_state()


def _set_state(state, device: Optional[torch.device] = None):
    if torch.cuda.is_available() and (device is None or device.type == "cuda"):
        torch.cuda.set_rng_state(state)
    else:
        torch.set_rng_state(state)


def _manual_seed(seed: int):
    if torch.cuda.is_available():
        torch.cuda.manual_seed(seed)
    else:
        torch.manual_seed(seed)


class RNGStatesTracker:
    """Named RNG streams, forked per region (reference random.py:229)."""

    def __init__(self):
        self.states: Dict[str, object] = {}
        self.seeds = set()

    def reset(self):
        self.states = {}
        self.seeds = set()

    def get_states(self):
        return dict(self.states)

    def set_states(self, states):
        self.states = dict(states)

    def add(self, name: str, seed: int):
        if seed in self.seeds:
            raise ValueError(f"seed {seed} already present")
        self.seeds.add(seed)
        if name in self.states:
            raise ValueError(f"rng state {name} already present")
        orig = _get_state()
        _manual_seed(seed)
        self.states[name] = _get_state()
        _set_state(orig)

    @contextlib.contextmanager
    def fork(self, name: str = _MODEL_PARALLEL_RNG):
        if name not in self.states:
            # tolerate un-seeded use (single-process tests): behave as identity
            yield
            return
        orig = _get_state()
        _set_state(self.states[name])
        try:
            yield
        finally:
            self.states[name] = _get_state()
            _set_state(orig)


_TRACKER = RNGStatesTracker()


def get_rng_tracker() -> RNGStatesTracker:
    return _TRACKER


def model_parallel_seed(seed: int):
    """Seed global + tracker streams with TP/EP/DP-aware offsets
    (reference random.py:meeting `model_parallel_cuda_manual_seed`)."""
    from megatron_amd.parallel import grid as G

    tp_rank = G.get_tensor_model_parallel_rank()
    ep_rank = G.get_expert_model_parallel_rank() if G.grid_initialized() else 0
    pp_rank = G.get_pipeline_model_parallel_rank() if G.grid_initialized() else 0
    offset = seed + 2718
    tp_seed = offset + tp_rank + pp_rank * 1024
    ep_seed = offset + 40000 + ep_rank * 131072 + tp_rank + pp_rank * 1024
    dp_seed = seed

    _TRACKER.reset()
    torch.manual_seed(dp_seed)
    _manual_seed(dp_seed)
    _TRACKER.add(_MODEL_PARALLEL_RNG, tp_seed)
    _TRACKER.add(_EXPERT_PARALLEL_RNG, ep_seed)


class CheckpointFunction(torch.autograd.Function):
    """Activation recompute with exact RNG replay (reference random.py:585)."""

    @staticmethod
    def forward(ctx, run_function, distribute_saved_activations, *args):
        ctx.run_function = run_function
        ctx.fwd_rng_state = _get_state()
        ctx.fwd_tracker_states = _TRACKER.get_states()
        ctx.fwd_cpu_state = torch.get_rng_state()
        with torch.no_grad():
            outputs = run_function(*args)
        ctx.inputs = [a if not torch.is_tensor(a) else None for a in args]
        tensor_inputs = [a for a in args if torch.is_tensor(a)]
        ctx.save_for_backward(*tensor_inputs)
        return outputs

    @staticmethod
    def backward(ctx, *grad_outputs):
        tensors = list(ctx.saved_tensors)
        inputs = []
        for a in ctx.inputs:
            inputs.append(tensors.pop(0) if a is None else a)
        # restore RNG to forward-time state
        cur_rng = _get_state()
        cur_tracker = _TRACKER.get_states()
        cur_cpu = torch.get_rng_state()
        _set_state(ctx.fwd_rng_state)
        _TRACKER.set_states(ctx.fwd_tracker_states)
        torch.set_rng_state(ctx.fwd_cpu_state)
        detached = [x.detach().requires_grad_(x.requires_grad) if torch.is_tensor(x) else x for x in inputs]
        with torch.enable_grad():
            outputs = ctx.run_function(*detached)
        _set_state(cur_rng)
        _TRACKER.set_states(cur_tracker)
        torch.set_rng_state(cur_cpu)
        if torch.is_tensor(outputs):
            outputs = (outputs,)
        out_tensors, out_grads = [], []
        for o, g in zip(outputs, grad_outputs):
            if torch.is_tensor(o) and o.requires_grad:
                out_tensors.append(o)
                out_grads.append(g)
        torch.autograd.backward(out_tensors, out_grads)
        grads = tuple(
            x.grad if torch.is_tensor(x) and x.requires_grad else None for x in detached
        )
        return (None, None) + grads


def checkpoint(run_function, distribute_saved_activations, *args):
    return CheckpointFunction.apply(run_function, distribute_saved_activations, *args)
