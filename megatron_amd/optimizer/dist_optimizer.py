"""Distributed optimizer (ZeRO-1): optimizer state + fp32 main params sharded
over the DP group.

Capability analog of reference megatron/core/optimizer/distrib_optimizer.py:113.
Design follows the reference's param-boundary-ignorant contiguous sharding
(range maps distrib_optimizer.py:134-220) on top of our ParamAndGradBuffer:
each bucket [start, end) is cut into dp_size equal contiguous shards; this
rank owns shard dp_rank.  Grads arrive in the shard via the bucket
reduce-scatter (ddp.py); updated bf16 params leave via bucket all-gather.

The fp32 main params / Adam moments are flat per-(param ∩ shard ∩ wd-group)
segments — contiguous, so the multi-tensor AdamW kernel (K10) sees a handful
of large flat tensors.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

import torch
import torch.distributed as dist

from megatron_amd import ops
from megatron_amd.config import OptimizerConfig
from megatron_amd.optimizer.clip import (
    clip_grads_by_total_norm,
    get_grad_norm,
    param_is_not_tensor_parallel_duplicate,
)
from megatron_amd.optimizer.optimizer import _BaseOptimizer, _wd_group
from megatron_amd.parallel import grid as G


class _ShardSegment:
    """One param's intersection with this rank's shard of one bucket."""

    __slots__ = ("param", "main", "exp_avg", "exp_avg_sq", "model_view", "grad_view", "decay", "norm_ok", "key")

    def __init__(self, param, main, model_view, grad_view, decay, norm_ok, key):
        self.param = param
        self.main = main
        self.exp_avg = torch.zeros_like(main)
        self.exp_avg_sq = torch.zeros_like(main)
        self.model_view = model_view
        self.grad_view = grad_view
        self.decay = decay
        self.norm_ok = norm_ok
        self.key = key  # (buffer_idx, bucket_idx, param_offset) for checkpointing


class DistributedOptimizer(_BaseOptimizer):
    def __init__(self, config: OptimizerConfig, model_chunks: List):
        super().__init__(config, model_chunks)
        self.segments: List[_ShardSegment] = []
        param_names = {}
        for chunk in model_chunks:
            core = chunk.module if hasattr(chunk, "module") else chunk
            for name, p in core.named_parameters():
                param_names[p] = name
        for ci, chunk in enumerate(model_chunks):
            assert hasattr(chunk, "buffers"), "DistributedOptimizer requires DDP-wrapped chunks"
            for bi, buf in enumerate(chunk.buffers):
                assert buf.param_data is not None, (
                    "DDPConfig.use_distributed_optimizer must be set so params live in the buffer"
                )
                dp = buf.dp_size
                rank = dist.get_rank(group=buf.dp_group) if (dp > 1) else 0
                for bucket in buf.buckets:
                    shard_len = (bucket.end - bucket.start) // dp
                    s0 = bucket.start + rank * shard_len
                    s1 = s0 + shard_len
                    for p in bucket.params:
                        pstart, pend, _ = buf.param_index[p]
                        lo, hi = max(pstart, s0), min(pend, s1)
                        if lo >= hi:
                            continue
                        model_view = buf.param_data[lo:hi]
                        grad_view = buf.grad_data[lo:hi]
                        main = model_view.detach().float().clone()
                        self.segments.append(
                            _ShardSegment(
                                p, main, model_view, grad_view,
                                decay=_wd_group(p),
                                norm_ok=param_is_not_tensor_parallel_duplicate(p),
                                key=(param_names.get(p, f"buf{ci}.{bi}"), lo - pstart, hi - pstart),
                            )
                        )

    def reload_model_params(self):
        for seg in self.segments:
            seg.main.copy_(seg.model_view.float())

    @torch.no_grad()
    def step(self):
        self.finish_grad_sync()
        grads = [seg.grad_view.float() for seg in self.segments]
        dense_g, expert_g = [], []
        for seg, g in zip(self.segments, grads):
            if not seg.norm_ok:
                continue
            is_exp = getattr(seg.param, "is_expert_parallel", False)
            (expert_g if is_exp else dense_g).append(g)
        extra, eextra = [], []
        if G.grid_initialized():
            # shards are disjoint over the dense shard group (dp_cp, or the
            # intra-instance sub-group under multi-instance) / edp (expert);
            # the cross-instance replicas must NOT be re-counted
            dense_group = getattr(self.model_chunks[0], "dense_shard_group", None)
            extra.append(dense_group if dense_group is not None
                         else G.get_grid().group("dp_cp"))
            eextra.append(G.get_grid().group("expert_dp"))
        total_norm = get_grad_norm(
            dense_g, extra_groups=extra, expert_grads=expert_g, expert_extra_groups=eextra
        )
        if self.config.clip_grad > 0:
            clip_grads_by_total_norm(grads, self.config.clip_grad, total_norm)
        self.step_count += 1
        for apply_wd in (True, False):
            idx = [i for i, seg in enumerate(self.segments) if seg.decay == apply_wd]
            if not idx:
                continue
            ops.fused_adamw(
                [self.segments[i].main for i in idx],
                [grads[i] for i in idx],
                [self.segments[i].exp_avg for i in idx],
                [self.segments[i].exp_avg_sq for i in idx],
                self._lr,
                self.config.adam_beta1,
                self.config.adam_beta2,
                self.config.adam_eps,
                self._wd if apply_wd else 0.0,
                self.step_count,
                model_params_bf16=[self.segments[i].model_view for i in idx],
            )
        # propagate updated shards to all DP ranks (bucketed all-gather)
        for chunk in self.model_chunks:
            chunk.start_param_sync(async_op=self.config.overlap_param_gather)
            if not self.config.overlap_param_gather:
                chunk.finish_param_sync()
        return True, total_norm, None

    def start_param_sync(self):
        for chunk in self.model_chunks:
            chunk.start_param_sync(async_op=True)

    def state_dict(self):
        return {
            "step": self.step_count,
            "segments": {
                (seg.key): {"main": seg.main, "exp_avg": seg.exp_avg, "exp_avg_sq": seg.exp_avg_sq}
                for seg in self.segments
            },
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        for seg in self.segments:
            entry = sd["segments"][seg.key]
            seg.main.copy_(entry["main"])
            seg.exp_avg.copy_(entry["exp_avg"])
            seg.exp_avg_sq.copy_(entry["exp_avg_sq"])
            seg.model_view.copy_(seg.main.to(seg.model_view.dtype))
