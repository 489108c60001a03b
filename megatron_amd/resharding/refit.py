"""Online weight resharding (training -> inference refit).

Capability analog of reference megatron/core/resharding/ (planner.py,
transforms.py, copy services): after optimizer steps, an RL loop must push
updated weights from the training grid (TP x PP x DP sharding) into an
inference engine that may shard differently, without a round trip through
disk.

MI355X-native design: instead of the reference's NVSHMEM one-sided copy
kernels, we reuse the checkpoint layer's flat-atlas ShardedTensor maps
(checkpoint/state_dict.py) to describe both sides, assemble each global
tensor with a single summed all-reduce over the job (each element written by
exactly one owner rank, zeros elsewhere — on one node this rides RCCL over
all 7 xGMI links), then every rank slices its destination shards out
locally.  One global tensor is live at a time, so peak scratch is the
largest single weight, not the model.
"""

from __future__ import annotations

from typing import Dict, Iterable, List, Optional

import torch
import torch.distributed as dist

from megatron_amd.checkpoint.sharded import ShardedTensor
from megatron_amd.checkpoint.state_dict import model_sharded_state_dict


def _write_into_global(buf: torch.Tensor, st: ShardedTensor) -> None:
    """Place this rank's shard `st` into the zeros-initialized global buffer."""
    idx = tuple(slice(o, o + s) for o, s in zip(st.global_offset, st.local_shape))
    window = buf[idx]
    if st.flattened_range is not None:
        lo, hi = st.flattened_range
        window.reshape(-1)[lo:hi] = st.data.reshape(-1).to(buf.dtype)
    else:
        window.copy_(st.data.to(buf.dtype))


def _read_from_global(buf: torch.Tensor, st: ShardedTensor) -> None:
    """Fill this rank's destination shard `st` from the assembled global."""
    idx = tuple(slice(o, o + s) for o, s in zip(st.global_offset, st.local_shape))
    window = buf[idx]
    with torch.no_grad():
        if st.flattened_range is not None:
            lo, hi = st.flattened_range
            st.data.reshape(-1).copy_(window.reshape(-1)[lo:hi].to(st.data.dtype))
        else:
            st.data.copy_(window.to(st.data.dtype))


def assemble_global_tensors(
    shard_map: Dict[str, ShardedTensor],
    keys: Optional[Iterable[str]] = None,
    group=None,
    shapes: Optional[Dict[str, tuple]] = None,
    device=None,
) -> Dict[str, torch.Tensor]:
    """Reconstruct full global tensors from each rank's shards.

    Only shards with replica_id == 0 contribute (the unique owners by the
    state-dict convention), so the summed all-reduce writes every element
    exactly once.  All ranks must call with the same key set; ranks that
    hold no shard of a key need its global shape in `shapes` (e.g. under
    pipeline parallelism)."""
    by_key: Dict[str, List[ShardedTensor]] = {}
    for st in shard_map.values():
        by_key.setdefault(st.key, []).append(st)
    if keys is None:
        keys = sorted(by_key.keys())
    out: Dict[str, torch.Tensor] = {}
    for key in keys:
        shards = by_key.get(key, [])
        if shards:
            shape = shards[0].global_shape
            dev = shards[0].data.device
        elif shapes is not None and key in shapes:
            shape = shapes[key]
            dev = device or "cpu"
        else:
            raise KeyError(f"rank holds no shard metadata for {key!r}; "
                           "pass its global shape via `shapes`")
        buf = torch.zeros(shape, dtype=torch.float32, device=dev)
        for st in shards:
            if st.replica_id == 0:
                _write_into_global(buf, st)
        if dist.is_initialized() and dist.get_world_size(group) > 1:
            dist.all_reduce(buf, group=group)
        out[key] = buf
    return out


def refit_model(
    src_model: torch.nn.Module,
    dst_model: torch.nn.Module,
    group=None,
    prefix: str = "model.",
) -> int:
    """Copy src_model's weights into dst_model across (possibly different)
    shardings.  Both models must exist on every participating rank (their
    own shards of them).  Returns the number of global tensors moved."""
    src_map, _ = model_sharded_state_dict(src_model, prefix=prefix)
    dst_map, _ = model_sharded_state_dict(dst_model, prefix=prefix)

    dst_by_key: Dict[str, List[ShardedTensor]] = {}
    for st in dst_map.values():
        dst_by_key.setdefault(st.key, []).append(st)

    # every rank iterates the union of destination keys in sorted order so
    # the all-reduces line up even when a rank holds no shard of a key;
    # src global shapes travel with the key union (PP: a rank may need a
    # layer it doesn't own on either side)
    local_meta = {st.key: tuple(st.global_shape) for st in src_map.values()}
    local_keys = set(dst_by_key.keys())
    if dist.is_initialized() and dist.get_world_size(group) > 1:
        ws = dist.get_world_size(group)
        gathered: List[Optional[tuple]] = [None] * ws
        dist.all_gather_object(gathered, (local_keys, local_meta), group=group)
        all_keys = sorted(set().union(*(g[0] for g in gathered)))
        shapes: Dict[str, tuple] = {}
        for g in gathered:
            shapes.update(g[1])
    else:
        all_keys = sorted(local_keys)
        shapes = local_meta

    device = next(dst_model.parameters()).device
    moved = 0
    for key in all_keys:
        full = assemble_global_tensors(src_map, keys=[key], group=group,
                                       shapes=shapes, device=device)[key]
        for st in dst_by_key.get(key, []):
            _read_from_global(full, st)
        moved += 1
    return moved


# ---------------------------------------------------------------------------
# Copy planner (reference resharding/planner.py + transforms.py): instead of
# broadcasting whole global tensors, compute the exact (src shard x dst
# shard) intersections across ranks and move only those bytes with batched
# p2p — total traffic = sum of destination shard bytes, not world x model.
# ---------------------------------------------------------------------------


class CopyTask:
    """One rectangular intersection to move: src rank's slice -> dst rank."""

    __slots__ = ("key", "src_rank", "dst_rank", "box_off", "box_shape",
                 "src_off", "dst_off", "src_idx", "dst_idx")

    def __init__(self, key, src_rank, dst_rank, box_off, box_shape, src_off, dst_off,
                 src_idx, dst_idx):
        self.key = key
        self.src_rank = src_rank
        self.dst_rank = dst_rank
        self.box_off = box_off      # global offset of the intersection
        self.box_shape = box_shape
        self.src_off = src_off      # offset inside the src shard
        self.dst_off = dst_off      # offset inside the dst shard
        self.src_idx = src_idx      # position in src rank's send-meta list
        self.dst_idx = dst_idx      # position in dst rank's recv-meta list

    def sort_key(self):
        return (self.key, self.src_rank, self.dst_rank, self.src_idx, self.dst_idx,
                tuple(self.box_off))


def _intersect(off_a, shape_a, off_b, shape_b):
    lo = [max(a, b) for a, b in zip(off_a, off_b)]
    hi = [min(a + la, b + lb) for a, la, b, lb in zip(off_a, shape_a, off_b, shape_b)]
    if any(h <= l for l, h in zip(lo, hi)):
        return None
    return lo, [h - l for l, h in zip(lo, hi)]


def plan_refit(src_map: Dict[str, ShardedTensor], dst_map: Dict[str, ShardedTensor],
               group=None) -> List[CopyTask]:
    """Build the deterministic global copy plan (identical on every rank).

    Rectangular shards only (model weights); flattened optimizer shards go
    through the assemble path.  Only replica_id==0 src shards send."""
    rank = dist.get_rank(group) if dist.is_initialized() else 0
    ws = dist.get_world_size(group) if dist.is_initialized() else 1

    def metas(m, send_side):
        out = []
        for st in m.values():
            if st.flattened_range is not None:
                raise ValueError("planner handles rectangular shards only")
            if send_side and st.replica_id != 0:
                continue
            out.append((st.key, tuple(st.global_offset), tuple(st.local_shape)))
        return out

    local = (metas(src_map, True), metas(dst_map, False))
    if ws > 1:
        gathered: List = [None] * ws
        dist.all_gather_object(gathered, local, group=group)
    else:
        gathered = [local]

    tasks: List[CopyTask] = []
    for dr, (_, dsts) in enumerate(gathered):
        for di, (dkey, doff, dshape) in enumerate(dsts):
            for sr, (srcs, _) in enumerate(gathered):
                for si, (skey, soff, sshape) in enumerate(srcs):
                    if skey != dkey:
                        continue
                    hit = _intersect(soff, sshape, doff, dshape)
                    if hit is None:
                        continue
                    box_off, box_shape = hit
                    tasks.append(CopyTask(
                        dkey, sr, dr, box_off, box_shape,
                        [o - s for o, s in zip(box_off, soff)],
                        [o - d for o, d in zip(box_off, doff)],
                        si, di))
    tasks.sort(key=CopyTask.sort_key)
    return tasks


def execute_refit_plan(tasks: List[CopyTask], src_map: Dict[str, ShardedTensor],
                       dst_map: Dict[str, ShardedTensor], group=None) -> int:
    """Run the plan: local copies directly; remote pieces via batched
    isend/irecv (posting order = canonical plan order, so pairwise matching
    is deterministic)."""
    rank = dist.get_rank(group) if dist.is_initialized() else 0
    # index shards in the SAME order plan_refit's metas() enumerated them
    src_list = [st for st in src_map.values()
                if st.flattened_range is None and st.replica_id == 0]
    dst_list = [st for st in dst_map.values() if st.flattened_range is None]

    def narrow(t, off, shape):
        for d, (o, l) in enumerate(zip(off, shape)):
            t = t.narrow(d, o, l)
        return t

    ops_list = []
    recv_bufs = []
    moved = 0
    for t in tasks:
        if t.src_rank == rank and t.dst_rank == rank:
            s = src_list[t.src_idx]
            d = dst_list[t.dst_idx]
            with torch.no_grad():
                narrow(d.data.view(d.local_shape), t.dst_off, t.box_shape).copy_(
                    narrow(s.data.view(s.local_shape), t.src_off, t.box_shape).to(d.data.dtype))
            moved += 1
        elif t.src_rank == rank:
            s = src_list[t.src_idx]
            payload = narrow(s.data.view(s.local_shape), t.src_off, t.box_shape).contiguous().float()
            ops_list.append(dist.P2POp(dist.isend, payload, t.dst_rank, group=group))
        elif t.dst_rank == rank:
            buf = torch.empty(t.box_shape, dtype=torch.float32,
                              device=dst_list[t.dst_idx].data.device if dst_list else "cpu")
            recv_bufs.append((t, buf))
            ops_list.append(dist.P2POp(dist.irecv, buf, t.src_rank, group=group))
    if ops_list:
        for r in dist.batch_isend_irecv(ops_list):
            r.wait()
    for t, buf in recv_bufs:
        d = dst_list[t.dst_idx]
        with torch.no_grad():
            narrow(d.data.view(d.local_shape), t.dst_off, t.box_shape).copy_(buf.to(d.data.dtype))
        moved += 1
    return moved


def refit_model_planned(src_model, dst_model, group=None, prefix: str = "model.") -> int:
    """Planner-based refit (exact-intersection p2p); falls back to the
    assemble path if any shard is flattened."""
    src_map, _ = model_sharded_state_dict(src_model, prefix=prefix)
    dst_map, _ = model_sharded_state_dict(dst_model, prefix=prefix)
    try:
        tasks = plan_refit(src_map, dst_map, group=group)
    except ValueError:
        return refit_model(src_model, dst_model, group=group, prefix=prefix)
    return execute_refit_plan(tasks, src_map, dst_map, group=group)
