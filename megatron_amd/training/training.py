"""Training step + setup helpers.

Capability analog of reference megatron/training/training.py (train_step
:2372, setup_model_and_optimizer :2042, get_model :1732) — lean version;
the full pretrain() CLI app lives in pretrain_gpt.py at the repo root.
"""

from __future__ import annotations

from typing import Callable, List, Optional

import torch
import torch.distributed as dist

from megatron_amd.config import DDPConfig, OptimizerConfig, TransformerConfig
from megatron_amd.distributed import DistributedDataParallel, finalize_model_grads
from megatron_amd.optimizer import ChainedOptimizer, get_optimizer
from megatron_amd.parallel import grid as G
from megatron_amd.pipeline.schedules import get_forward_backward_func
from megatron_amd.utils.rerun_state_machine import get_rerun_state_machine


class FSDPTrainAdapter:
    """Adapts FullyShardedDataParallel + a plain AdamW over its fp32 master
    shards to the train_step optimizer/chunk contract (reference
    torch_fully_sharded_data_parallel.py wrapper role)."""

    def __init__(self, fsdp, opt_config: OptimizerConfig):
        self.fsdp = fsdp
        self.config = opt_config
        shards = fsdp.shard_parameters()
        self.inner = torch.optim.AdamW(
            shards, lr=opt_config.lr, weight_decay=opt_config.weight_decay,
            betas=(opt_config.adam_beta1, opt_config.adam_beta2), eps=opt_config.adam_eps)
        self.chained_optimizers = [self]
        self.step_count = 0

    # -- optimizer contract used by train_step / scheduler ------------------
    def zero_grad(self):
        self.inner.zero_grad(set_to_none=True)

    def finish_grad_sync(self):
        pass  # per-unit reduce-scatter completes at grad-ready hooks

    def step(self):
        norm = None
        if self.config.clip_grad and self.config.clip_grad > 0:
            norm = self.fsdp.clip_grad_norm(self.config.clip_grad)
        self.inner.step()
        self.fsdp.update_model_shards()
        self.step_count += 1
        return True, norm, None

    def reload_model_params(self):
        pass

    def state_dict(self):
        return {"inner": self.inner.state_dict(), "step": self.step_count}

    def load_state_dict(self, sd):
        self.inner.load_state_dict(sd["inner"])
        self.step_count = sd.get("step", 0)

    def set_lr(self, lr: float):
        for g in self.inner.param_groups:
            g["lr"] = lr

    def set_wd(self, wd: float):
        for g in self.inner.param_groups:
            g["weight_decay"] = wd

    def get_lr(self) -> float:
        return self.inner.param_groups[0]["lr"]

    @property
    def param_groups(self):
        return self.inner.param_groups


def setup_fsdp_model_and_optimizer(model_provider, config, opt_config,
                                   device: Optional[torch.device] = None):
    """--use-fsdp path: ZeRO-3 sharding instead of DDP + ZeRO-1."""
    from megatron_amd.distributed.fsdp import FullyShardedDataParallel

    if device is not None:
        with torch.device(device):
            m = model_provider(config, pre_process=True, post_process=True, vp_stage=None)
        m = m.to(device)
    else:
        m = model_provider(config, pre_process=True, post_process=True, vp_stage=None)
    fsdp = FullyShardedDataParallel(m)
    # train_step chunk surface
    fsdp.no_sync = fsdp.no_last_microbatch
    fsdp.start_grad_sync = lambda: None
    fsdp.finish_grad_sync = lambda: None
    fsdp.broadcast_params = lambda: None
    opt = FSDPTrainAdapter(fsdp, opt_config)
    config.finalize_model_grads_func = None
    return [fsdp], opt


def setup_model_and_optimizer(
    model_provider: Callable[..., torch.nn.Module],
    config: TransformerConfig,
    opt_config: OptimizerConfig,
    ddp_config: Optional[DDPConfig] = None,
    device: Optional[torch.device] = None,
):
    """Build (virtual-chunked) model list, wrap in DDP, build optimizer."""
    if ddp_config is None:
        ddp_config = DDPConfig(
            use_distributed_optimizer=opt_config.use_distributed_optimizer,
            grad_reduce_in_fp32=True,
        )
    vpp = config.virtual_pipeline_parallel_size
    chunks = []
    n_chunks = vpp if (vpp is not None and G.get_pipeline_model_parallel_world_size() > 1) else 1
    grid = G.get_grid() if G.grid_initialized() else None
    for vp in range(n_chunks):
        if grid is not None:
            grid.set_vpp_rank(vp if vpp else None)
        pre = G.grid_initialized() and grid.is_pipeline_first_stage() or not G.grid_initialized()
        post = G.grid_initialized() and grid.is_pipeline_last_stage() or not G.grid_initialized()
        if device is not None:
            with torch.device(device):  # init weights directly on the GPU
                m = model_provider(config, pre_process=pre, post_process=post, vp_stage=vp if vpp else None)
            m = m.to(device)
        else:
            m = model_provider(config, pre_process=pre, post_process=post, vp_stage=vp if vpp else None)
        chunks.append(m)
    if grid is not None:
        grid.set_vpp_rank(0 if vpp else None)

    ddp_chunks = [DistributedDataParallel(config, ddp_config, m) for m in chunks]
    for c in ddp_chunks:
        c.broadcast_params()
    _sync_embd_replicas(ddp_chunks, config)
    optimizer = get_optimizer(opt_config, ddp_chunks)

    # wire the schedule hooks (reference three-hook contract)
    config.finalize_model_grads_func = finalize_model_grads
    return ddp_chunks, optimizer


def _sync_embd_replicas(chunks, config):
    """First/last-PP-stage embedding replicas (tied output weight, MTP
    embedding) start from the first stage's values (reference
    setup_embeddings_and_output_layer).  The broadcast count is derived from
    config so every embd-group member issues matching collectives."""
    import torch.distributed as dist

    if not (G.grid_initialized() and dist.is_initialized()):
        return
    grid = G.get_grid()
    if grid.pp == 1:
        return
    group = grid.group("embd")
    ranks = grid.ranks("embd")
    if group is None or len(ranks) < 2:
        return
    tied = not config.untie_embeddings_and_output_weights
    mtp_untied = bool(getattr(config, "mtp_num_layers", 0)) and not tied
    for c in chunks:
        core = c.module if hasattr(c, "module") else c
        if not (getattr(core, "pre_process", False) or getattr(core, "post_process", False)):
            continue
        if tied:
            w = core.shared_embedding_or_output_weight()
            if w is not None:
                dist.broadcast(w.data, src=ranks[0], group=group)
        if mtp_untied:
            w = None
            if getattr(core, "pre_process", False) and getattr(core, "embedding", None) is not None:
                w = core.embedding.weight
            elif getattr(core, "mtp_embedding", None) is not None:
                w = core.mtp_embedding.weight
            if w is not None:
                dist.broadcast(w.data, src=ranks[0], group=group)


def train_step(
    forward_step_func,
    data_iterator,
    model_chunks: List,
    optimizer: ChainedOptimizer,
    config: TransformerConfig,
    num_microbatches: int,
    seq_length: int,
    micro_batch_size: int,
):
    """One optimizer step (reference training.py:2372). The rerun state
    machine wraps only the forward/backward (optimizer state is untouched on
    a replay, reference training.py:2395)."""
    rsm = get_rerun_state_machine()
    iters = data_iterator if isinstance(data_iterator, list) else [data_iterator]
    fb_func = get_forward_backward_func(config)
    losses = num_tokens = None
    while rsm.should_run_forward_backward(iters):
        for chunk in model_chunks:
            chunk.zero_grad_buffer()
        optimizer.zero_grad() if hasattr(optimizer, "zero_grad") else None
        losses, num_tokens = fb_func(
            forward_step_func=forward_step_func,
            data_iterator=iters,
            model=model_chunks,
            num_microbatches=num_microbatches,
            seq_length=seq_length,
            micro_batch_size=micro_batch_size,
        )
    exit_code = rsm.should_checkpoint_and_exit()
    if exit_code is not None:
        return {"lm_loss": float("nan"), "grad_norm": None, "skipped": True,
                "exit_code": exit_code}

    ok, grad_norm, _ = optimizer.step()

    # aggregate loss over DP(xCP) and microbatches: token-weighted average
    loss_sum = torch.zeros((), dtype=torch.float32, device=num_tokens.device)
    for m in losses:
        if "loss_sum" in m:
            loss_sum += m["loss_sum"]
    total_tokens = num_tokens.clone()
    if G.grid_initialized() and dist.is_initialized():
        grid = G.get_grid()
        group = grid.group("dp_cp")
        if group is not None and dist.get_world_size(group=group) > 1:
            dist.all_reduce(loss_sum, group=group)
            dist.all_reduce(total_tokens, group=group)
        # loss lives on the last PP stage: share it down the pipe for logging
        if grid.pp > 1 and grid.group("pp") is not None:
            dist.broadcast(loss_sum, src=grid.pipeline_last_rank(), group=grid.group("pp"))
            dist.broadcast(total_tokens, src=grid.pipeline_last_rank(), group=grid.group("pp"))
    mean_loss = loss_sum / total_tokens.clamp(min=1)
    return {"lm_loss": mean_loss.item(), "grad_norm": None if grad_norm is None else float(grad_norm), "skipped": not ok}
