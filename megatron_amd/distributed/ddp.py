"""Data-parallel gradient management: contiguous buffers, bucketed async
reduce, backward overlap.

Capability analog of reference megatron/core/distributed/
distributed_data_parallel.py:22 + param_and_grad_buffer.py
(_ParamAndGradBucket :92, _ParamAndGradBucketGroup :179, _ParamAndGradBuffer
:1005), redesigned for one 8-GPU xGMI node:

  * grads accumulate into ONE contiguous per-dtype ``grad_data`` buffer;
    each param's ``main_grad`` is a view into it (288 GB HBM3E -> big
    resident buffers, no per-step allocation).
  * buffer split into ~bucket_size-element buckets, padded so each bucket
    divides evenly by the DP size (reduce-scatter shards stay aligned).
  * bucket-ready triggers an async all-reduce (plain DDP) or reduce-scatter
    (distributed optimizer) on the DP group, overlapped with backward.
    xGMI is fully-connected single-hop: RCCL reduce-scatter engages all 7
    links, so fewer/larger buckets beat many small ones.
  * readiness signaling: non-fused params use
    ``register_post_accumulate_grad_hook``; params whose wgrad is fused
    straight into main_grad (tensor_parallel/layers.py here) call the
    ``_ddp_grad_ready_cb`` attribute from the autograd thread — no dummy
    grad tensors (the reference's workaround, layers.py:691-720).
"""

from __future__ import annotations

import math
from contextlib import contextmanager
from typing import Dict, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from megatron_amd.config import DDPConfig
from megatron_amd.parallel import grid as G


def _pad_to(x: int, align: int) -> int:
    return int(math.ceil(x / align) * align) if align > 1 else x


_GROUP_CACHE: Dict[tuple, object] = {}


def _cached_group(ranks: tuple):
    if ranks not in _GROUP_CACHE:
        _GROUP_CACHE[ranks] = dist.new_group(ranks=list(ranks))
    return _GROUP_CACHE[ranks]


def instance_groups(dp_group, n_instances: int):
    """Split the dp(_cp) group into n contiguous optimizer instances.
    Returns (intra, inter) for this rank: ``intra`` = this rank's instance
    (shard/all-gather domain), ``inter`` = same shard position across
    instances (grad replication all-reduce).  Every rank must call this with
    the same arguments (dist.new_group is collective); groups are cached."""
    ranks = dist.get_process_group_ranks(dp_group)
    n = len(ranks)
    assert n % n_instances == 0, (n, n_instances)
    per = n // n_instances
    me = dist.get_rank()
    intra = inter = None
    for i in range(n_instances):
        g = _cached_group(tuple(ranks[i * per : (i + 1) * per]))
        if me in ranks[i * per : (i + 1) * per]:
            intra = g
    for k in range(per):
        col = [ranks[i * per + k] for i in range(n_instances)]
        g = _cached_group(tuple(col))
        if me in col:
            inter = g
    return intra, inter


class _Bucket:
    def __init__(self, params: List[torch.nn.Parameter], start: int, end: int, index: int):
        self.params = params
        self.start = start
        self.end = end
        self.index = index
        self.params_with_grad = set()
        self.comm_handle = None
        self.grad_view: Optional[torch.Tensor] = None   # [end-start] of grad_data
        self.param_view: Optional[torch.Tensor] = None  # [end-start] of param_data
        self.local_grad_shard: Optional[torch.Tensor] = None  # this rank's RS shard


class ParamAndGradBuffer:
    """Contiguous param/grad storage + bucket bookkeeping for one dtype group."""

    def __init__(
        self,
        params: List[torch.nn.Parameter],
        ddp_config: DDPConfig,
        dp_group,
        param_dtype: torch.dtype,
        grad_dtype: torch.dtype,
        device: torch.device,
        grad_scale_denom: int = 0,
        inter_group=None,
    ):
        self.ddp_config = ddp_config
        self.dp_group = dp_group
        self.dp_size = dist.get_world_size(group=dp_group) if (dp_group is not None and dist.is_initialized()) else 1
        # Grads are averaged over this many ranks.  For the dense buffer this
        # equals the dp_cp group size (== dp_size); for the expert buffer the
        # reduce group is edp but the average must still be over dp_cp so
        # expert grads get the same effective scale as dense grads (reference
        # expert_gradient_scaling_factor = edp/dp_cp pre-scale + AVG over edp).
        self.grad_scale_denom = grad_scale_denom or self.dp_size
        self.param_names: List[tuple] = []  # [(name, param)], set by the DDP wrapper
        self.inter_group = inter_group  # cross-instance grad all-reduce (multi-instance dist-opt)
        self.grad_dtype = grad_dtype
        self.param_dtype = param_dtype
        self.device = device
        self.is_last_microbatch = True

        # --- assign offsets (params in reverse registration order: backward
        #     finishes roughly in that order, so early buckets fill first) ---
        align = 64 * self.dp_size  # keep shards 256B-aligned per rank
        bucket_target = ddp_config.bucket_size or 40_000_000
        self.buckets: List[_Bucket] = []
        self.param_index: Dict[torch.nn.Parameter, tuple] = {}  # param -> (start, end, bucket_idx)

        offset = 0
        cur_params: List[torch.nn.Parameter] = []
        cur_start = 0
        for p in params:
            n = p.data.nelement()
            self.param_index[p] = (offset, offset + n, len(self.buckets))
            cur_params.append(p)
            offset += n
            if offset - cur_start >= bucket_target:
                offset = _pad_to(offset, align)
                self.buckets.append(_Bucket(cur_params, cur_start, offset, len(self.buckets)))
                cur_params, cur_start = [], offset
        if cur_params:
            offset = _pad_to(offset, align)
            self.buckets.append(_Bucket(cur_params, cur_start, offset, len(self.buckets)))
        self.total_elements = offset

        # --- allocate storage (optionally inside an RCCL-registered pool,
        # reference nccl_allocator.py N2: zero-copy xGMI transports) ---
        import contextlib

        alloc_ctx = contextlib.nullcontext()
        self._comm_pool = None
        if getattr(ddp_config, "use_rccl_registered_buffers", False):
            from megatron_amd.distributed.rccl_allocator import RcclRegisteredPool
            from megatron_amd.parallel import grid as G

            self._comm_pool = RcclRegisteredPool(
                G.get_grid().group("dp_cp") if G.grid_initialized() else None)
            alloc_ctx = self._comm_pool.use()
        with alloc_ctx:
            self.grad_data = torch.zeros(self.total_elements, dtype=grad_dtype, device=device)
            self.param_data = None
            if ddp_config.use_distributed_optimizer:
                self.param_data = torch.empty(self.total_elements, dtype=param_dtype, device=device)

        for p in params:
            start, end, bidx = self.param_index[p]
            p.main_grad = self.grad_data[start:end].view(p.data.shape)
            if self.param_data is not None:
                with torch.no_grad():
                    self.param_data[start:end].view(p.data.shape).copy_(p.data)
                    new_data = self.param_data[start:end].view(p.data.shape)
                p.data = new_data

        for b in self.buckets:
            b.grad_view = self.grad_data[b.start : b.end]
            if self.param_data is not None:
                b.param_view = self.param_data[b.start : b.end]
                shard = (b.end - b.start) // self.dp_size
                rank = dist.get_rank(group=dp_group) if self.dp_size > 1 else 0
                b.local_grad_shard = self.grad_data[b.start + rank * shard : b.start + (rank + 1) * shard]

    # -- per-bucket collectives ---------------------------------------------

    def _launch_grad_reduce(self, bucket: _Bucket, async_op: bool):
        if self.ddp_config.check_for_nan_in_grad:
            # reference check_for_nan_in_grad: fail fast, per bucket, BEFORE
            # the collective so the faulty rank is identifiable
            if not torch.isfinite(bucket.grad_view.sum()):
                bad = [n for n, p in self.param_names
                       if any(p is q for q in bucket.params)]
                raise RuntimeError(
                    f"NaN/Inf grad in bucket {bucket.index} before reduce "
                    f"(rank {dist.get_rank() if dist.is_initialized() else 0}); "
                    f"params: {bad[:8]}")
        denom = self.grad_scale_denom
        if self.dp_size == 1:
            # no collective, but the average denominator may still exceed 1
            # (expert buffer with edp=1 while dp_cp>1)
            if denom > 1:
                bucket.grad_view.mul_(1.0 / denom)
            return
        # AVG happens in-collective on RCCL; gloo has no AVG -> pre-scale + SUM
        if self.ddp_config.average_in_collective and dist.get_backend(self.dp_group) == "nccl":
            if denom != self.dp_size:
                bucket.grad_view.mul_(self.dp_size / denom)
            op = dist.ReduceOp.AVG
        else:
            bucket.grad_view.mul_(1.0 / denom)
            op = dist.ReduceOp.SUM
        if self.ddp_config.use_distributed_optimizer:
            bucket.comm_handle = dist.reduce_scatter_tensor(
                bucket.local_grad_shard, bucket.grad_view, op=op, group=self.dp_group, async_op=async_op
            )
        else:
            bucket.comm_handle = dist.all_reduce(bucket.grad_view, op=op, group=self.dp_group, async_op=async_op)

    def start_grad_sync(self):
        for b in self.buckets:
            if b.comm_handle is None:
                self._launch_grad_reduce(b, async_op=True)

    def finish_grad_sync(self):
        self.start_grad_sync()
        for b in self.buckets:
            if b.comm_handle is not None and not isinstance(b.comm_handle, bool):
                b.comm_handle.wait()
            if self.inter_group is not None and b.comm_handle is not None:
                # multi-instance dist-opt: the intra-instance reduce-scatter
                # left each instance's partial sum in the shard; sum the
                # replicas across instances (reference two-level grad reduce)
                dist.all_reduce(b.local_grad_shard, group=self.inter_group)
            b.comm_handle = None
            b.params_with_grad = set()

    def start_param_sync(self, async_op: bool = False):
        """all-gather updated params (dist-opt) bucket by bucket."""
        if self.param_data is None or self.dp_size == 1:
            return
        self._param_handles = []
        for b in self.buckets:
            shard = (b.end - b.start) // self.dp_size
            rank = dist.get_rank(group=self.dp_group)
            local = b.param_view[rank * shard : (rank + 1) * shard]
            h = dist.all_gather_into_tensor(b.param_view, local.contiguous(), group=self.dp_group, async_op=async_op)
            if async_op:
                self._param_handles.append(h)

    def finish_param_sync(self):
        for h in getattr(self, "_param_handles", []):
            h.wait()
        self._param_handles = []

    def zero_grad(self):
        self.grad_data.zero_()
        for b in self.buckets:
            b.comm_handle = None
            b.params_with_grad = set()

    # -- readiness ------------------------------------------------------------

    def mark_ready(self, param: torch.nn.Parameter):
        if self.is_last_microbatch and self.ddp_config.overlap_grad_reduce:
            _, _, bidx = self.param_index[param]
            b = self.buckets[bidx]
            b.params_with_grad.add(param)
            if len(b.params_with_grad) == len(b.params) and b.comm_handle is None:
                self._launch_grad_reduce(b, async_op=True)


class DistributedDataParallel(nn.Module):
    """Wraps one model chunk (reference distributed_data_parallel.py:22)."""

    def __init__(self, config, ddp_config: DDPConfig, module: nn.Module):
        super().__init__()
        self.module = module
        self.config = config
        self.ddp_config = ddp_config
        self.dp_group = (
            G.get_data_parallel_group(with_context_parallel=True) if G.grid_initialized() else None
        )

        dense_params, expert_params = [], []
        for p in self.module.parameters():
            if not p.requires_grad:
                continue
            (expert_params if getattr(p, "is_expert_parallel", False) else dense_params).append(p)

        device = dense_params[0].device if dense_params else torch.device("cpu")
        grad_dtype = torch.float32 if ddp_config.grad_reduce_in_fp32 else (
            dense_params[0].dtype if dense_params else torch.float32
        )
        self.buffers: List[ParamAndGradBuffer] = []
        self.param_to_buffer: Dict[torch.nn.Parameter, ParamAndGradBuffer] = {}
        n_inst = getattr(ddp_config, "num_distributed_optimizer_instances", 1)
        dense_group, inter_group, dense_denom = self.dp_group, None, 0
        if (n_inst > 1 and ddp_config.use_distributed_optimizer
                and self.dp_group is not None and dist.is_initialized()):
            dense_group, inter_group = instance_groups(self.dp_group, n_inst)
            dense_denom = dist.get_world_size(group=self.dp_group)
        self.dense_shard_group = dense_group
        if dense_params:
            buf = ParamAndGradBuffer(
                list(reversed(dense_params)), ddp_config, dense_group,
                dense_params[0].dtype, grad_dtype, device,
                grad_scale_denom=dense_denom, inter_group=inter_group,
            )
            self.buffers.append(buf)
            for p in buf.param_index:
                self.param_to_buffer[p] = buf
        if expert_params:
            edp_group = G.get_grid().group("expert_dp") if G.grid_initialized() else None
            dp_cp_size = (
                dist.get_world_size(group=self.dp_group)
                if (self.dp_group is not None and dist.is_initialized()) else 1
            )
            ebuf = ParamAndGradBuffer(
                list(reversed(expert_params)), ddp_config, edp_group,
                expert_params[0].dtype, grad_dtype, device,
                grad_scale_denom=dp_cp_size,
            )
            self.buffers.append(ebuf)
            for p in ebuf.param_index:
                self.param_to_buffer[p] = ebuf

        name_of = {p: n for n, p in self.module.named_parameters()}
        for buf in self.buffers:
            buf.param_names = [(name_of.get(p, "?"), p) for p in buf.param_index]

        self._grad_hook_handles = []
        for p in self.param_to_buffer:
            p.grad_added_to_main_grad = False
            p._ddp_grad_ready_cb = self._make_ready_cb(p)  # fused wgrad path
            h = p.register_post_accumulate_grad_hook(self._make_post_acc_hook(p))
            self._grad_hook_handles.append(h)

    def _make_post_acc_hook(self, param):
        buffer = self.param_to_buffer[param]

        def hook(p):
            if p.grad is not None:
                param.main_grad.add_(p.grad.to(param.main_grad.dtype))
                p.grad = None
            buffer.mark_ready(p)

        return hook

    def _make_ready_cb(self, param):
        buffer = self.param_to_buffer[param]

        def cb():
            buffer.mark_ready(param)

        return cb

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    @contextmanager
    def no_sync(self):
        for buf in self.buffers:
            buf.is_last_microbatch = False
        try:
            yield
        finally:
            for buf in self.buffers:
                buf.is_last_microbatch = True

    def start_grad_sync(self):
        for buf in self.buffers:
            buf.start_grad_sync()

    def finish_grad_sync(self):
        for buf in self.buffers:
            buf.finish_grad_sync()

    def start_param_sync(self, async_op: bool = False, force_sync: bool = False):
        for buf in self.buffers:
            buf.start_param_sync(async_op=async_op)

    def finish_param_sync(self):
        for buf in self.buffers:
            buf.finish_param_sync()

    def zero_grad_buffer(self):
        for p in self.param_to_buffer:
            p.grad_added_to_main_grad = False
        for buf in self.buffers:
            buf.zero_grad()

    def broadcast_params(self):
        """Sync initial params across each buffer's OWN replica group: dense
        over dp, expert over edp.  Broadcasting experts over the dense dp
        group would overwrite different EP ranks' distinct experts."""
        if self.dp_group is None or not dist.is_initialized():
            return
        for buf in self.buffers:
            # multi-instance dense buffer shards over the intra-instance
            # group, but initial params must agree across ALL dp replicas
            group = self.dp_group if buf.inter_group is not None else buf.dp_group
            if group is None or dist.get_world_size(group=group) == 1:
                continue
            src = dist.get_process_group_ranks(group)[0]
            if buf.param_data is not None:
                dist.broadcast(buf.param_data, src=src, group=group)
            else:
                for p in buf.param_index:
                    dist.broadcast(p.data, src=src, group=group)

    def state_dict(self, *args, **kwargs):
        return self.module.state_dict(*args, **kwargs)

    def load_state_dict(self, *args, **kwargs):
        return self.module.load_state_dict(*args, **kwargs)
