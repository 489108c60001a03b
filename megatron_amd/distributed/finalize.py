"""End-of-step cross-group gradient reductions.

Capability analog of reference megatron/core/distributed/finalize_model_grads.py
(:164 word-embedding grads over the embd group, :416 non-TP-replicated
(sequence-parallel-duplicated) grads over TP, :560 entry point).
"""

from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist

from megatron_amd.parallel import grid as G


def _grad_of(param):
    return getattr(param, "main_grad", None) if getattr(param, "main_grad", None) is not None else param.grad


def _allreduce_layernorm_grads(models: List[torch.nn.Module], config):
    """SP shards the sequence over TP: norm/bias grads (computed from a seq
    slice) must be summed over the TP group."""
    if not (config.sequence_parallel and config.tensor_parallel_size > 1):
        return
    grid = G.get_grid()
    group = grid.group("tp")
    grads = []
    for model in models:
        for p in model.parameters():
            if getattr(p, "sequence_parallel_dup", False) and p.requires_grad:
                g = _grad_of(p)
                if g is not None:
                    grads.append(g.data)
    if grads and group is not None:
        flat = torch._utils._flatten_dense_tensors(grads)
        dist.all_reduce(flat, group=group)
        for g, synced in zip(grads, torch._utils._unflatten_dense_tensors(flat, grads)):
            g.copy_(synced)


def _allreduce_word_embedding_grads(models: List[torch.nn.Module], config):
    """Tied input/output embeddings live on first and last PP stage: their
    grads are summed over the embd group (reference finalize_model_grads:164)."""
    grid = G.get_grid()
    if grid.pp == 1:
        return
    if not (grid.is_pipeline_first_stage(ignore_virtual=True) or grid.is_pipeline_last_stage(ignore_virtual=True)):
        return
    group = grid.group("embd")
    if group is None or len(grid.ranks("embd")) < 2:
        return
    for model in models:
        core = model.module if hasattr(model, "module") else model
        if getattr(core, "share_embeddings_and_output_weights", False):
            w = core.shared_embedding_or_output_weight()
            if w is not None and w.requires_grad:
                g = _grad_of(w)
                if g is not None:
                    dist.all_reduce(g.data, group=group)
        # MTP with untied embeddings: the first stage's input embedding and
        # the last stage's mtp_embedding replica also form a tied pair
        # (tied models already covered above: mtp_embedding IS output weight)
        cfg = getattr(core, "config", None)
        if cfg is not None and getattr(cfg, "mtp_num_layers", 0) and not getattr(
                core, "share_embeddings_and_output_weights", False):
            w = None
            if getattr(core, "pre_process", False) and getattr(core, "embedding", None) is not None:
                w = core.embedding.weight
            elif getattr(core, "mtp_embedding", None) is not None:
                w = core.mtp_embedding.weight
            if w is not None and w.requires_grad:
                g = _grad_of(w)
                if g is not None:
                    dist.all_reduce(g.data, group=group)


def update_router_expert_bias(models: List[torch.nn.Module], config):
    """Aux-loss-free load balancing (reference finalize_model_grads.py:334,
    `_update_router_expert_bias`): sum each router's per-step routed-token
    counts over all data-parallel replicas, then nudge the routing bias toward
    the mean load: bias += rate * sign(mean_load - expert_load)."""
    if not getattr(config, "moe_router_enable_expert_bias", False):
        return
    from megatron_amd.moe.router import TopKRouter

    routers, counts = [], []
    for model in models:
        for m in model.modules():
            if isinstance(m, TopKRouter):
                routers.append(m)
                counts.append(m.local_tokens_per_expert)
    if not routers:
        return
    stacked = torch.stack(counts)  # [L, E]
    if G.grid_initialized() and dist.is_initialized():
        group = G.get_grid().group("dp_cp")
        if group is not None and dist.get_world_size(group) > 1:
            dist.all_reduce(stacked, group=group)
    rate = config.moe_router_bias_update_rate
    mean_load = stacked.mean(dim=-1, keepdim=True)
    update = rate * torch.sign(mean_load - stacked)
    for r, u in zip(routers, update):
        r.expert_bias += u.to(r.expert_bias.dtype)
        r.local_tokens_per_expert.zero_()


def _allreduce_tp_replicated_grads(models):
    """Params tagged ``average_gradients_across_tp_domain`` (HF-interop
    modules: replicated over TP, not sharded) get their grads averaged over
    the TP group so replicas never drift (reference huggingface/module.py)."""
    group = G.get_tensor_model_parallel_group()
    if group is None or G.get_tensor_model_parallel_world_size() <= 1:
        return
    for model in models:
        core = model.module if hasattr(model, "module") else model
        for p in core.parameters():
            if not getattr(p, "average_gradients_across_tp_domain", False):
                continue
            g = getattr(p, "main_grad", None)
            if g is None:
                g = p.grad
            if g is not None:
                dist.all_reduce(g, group=group)
                g /= G.get_tensor_model_parallel_world_size()


def finalize_model_grads(models: List[torch.nn.Module], config=None):
    if config is None:
        core = models[0].module if hasattr(models[0], "module") else models[0]
        config = core.config
    if not G.grid_initialized():
        return
    _allreduce_word_embedding_grads(models, config)
    _allreduce_layernorm_grads(models, config)
    _allreduce_tp_replicated_grads(models)
    update_router_expert_bias(models, config)
