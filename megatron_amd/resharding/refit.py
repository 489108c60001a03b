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
