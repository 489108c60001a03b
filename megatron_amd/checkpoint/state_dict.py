"""Sharded-state-dict builders for model and optimizers.

Capability analog of reference gpt_model.py:873 sharded_state_dict +
distrib_optimizer.py sharded_state_dict ("fully_reshardable") +
ShardedTensorFactory (mapping.py:438 — optimizer state shards exactly like
its param).

Core abstraction: a per-param *flat atlas* — an ordered list of sub-shards
covering the param's local flat element space, each mapping a flat interval
to (global key, global offset, sub-shard local shape, sub-shard flat
offset).  Most params have one entry; gated fc1 weights split into .gate /
.up global tensors (the per-TP-rank [gate;up] layout is not dim-0
reshardable); stacked expert gated weights split per (expert, half).
Model params AND the distributed optimizer's param-boundary-ignorant flat
shards both serialize by intersecting their flat ranges with the atlas.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Tuple

import torch

from megatron_amd.checkpoint.sharded import ShardedTensor
from megatron_amd.parallel import grid as G


@dataclass
class AtlasEntry:
    key: str
    global_shape: Tuple[int, ...]
    global_offset: Tuple[int, ...]
    local_shape: Tuple[int, ...]  # this rank's sub-shard of `key`
    flat_start: int               # interval in the param's local flat space
    flat_end: int
    sub_flat_start: int           # position of flat_start inside the sub-shard's flat space


def _prod(shape):
    n = 1
    for s in shape:
        n *= s
    return n


def param_atlas(p: torch.nn.Parameter, gname: str, grid) -> List[AtlasEntry]:
    tp = grid.size("tp")
    tp_rank = grid.rank_in("tp")
    if getattr(p, "is_expert_parallel", False):
        ep_rank = grid.rank_in("ep")
        e_local = p.shape[0]
        n_e = e_local * grid.ep
        if getattr(p, "is_gated_fc1", False):
            # [E_local, 2*ffn_pp, h] -> keys .gate/.up, global [E, ffn, h]
            ffn_pp = p.shape[1] // 2
            h = p.shape[2]
            gshape = (n_e, ffn_pp * tp, h)
            blk = ffn_pp * h  # elements per (expert, half)
            entries = []
            for e in range(e_local):
                for half, suffix in ((0, ".gate"), (1, ".up")):
                    entries.append(AtlasEntry(
                        key=gname + suffix, global_shape=gshape,
                        global_offset=(ep_rank * e_local, tp_rank * ffn_pp, 0),
                        local_shape=(e_local, ffn_pp, h),
                        flat_start=(e * 2 + half) * blk, flat_end=(e * 2 + half + 1) * blk,
                        sub_flat_start=e * blk,
                    ))
            return entries
        if gname.endswith("weight1"):
            gshape = (n_e, p.shape[1] * tp, p.shape[2])
            goff = (ep_rank * e_local, tp_rank * p.shape[1], 0)
        else:  # weight2 [E_local, h, ffn_pp]
            gshape = (n_e, p.shape[1], p.shape[2] * tp)
            goff = (ep_rank * e_local, 0, tp_rank * p.shape[2])
        return [AtlasEntry(gname, gshape, goff, tuple(p.shape), 0, p.numel(), 0)]

    if getattr(p, "is_gated_fc1", False):
        # [2*ffn_pp, h] -> .gate/.up, global [ffn, h]
        ffn_pp = p.shape[0] // 2
        h = p.shape[1]
        gshape = (ffn_pp * tp, h)
        blk = ffn_pp * h
        return [
            AtlasEntry(gname + ".gate", gshape, (tp_rank * ffn_pp, 0), (ffn_pp, h), 0, blk, 0),
            AtlasEntry(gname + ".up", gshape, (tp_rank * ffn_pp, 0), (ffn_pp, h), blk, 2 * blk, 0),
        ]

    dim = getattr(p, "partition_dim", None) if getattr(p, "tensor_parallel", False) else None
    if dim is not None and tp > 1:
        gshape = tuple(s * tp if d == dim else s for d, s in enumerate(p.shape))
        goff = tuple(tp_rank * p.shape[dim] if d == dim else 0 for d in range(p.dim()))
    else:
        gshape = tuple(p.shape)
        goff = tuple(0 for _ in p.shape)
    return [AtlasEntry(gname, gshape, goff, tuple(p.shape), 0, p.numel(), 0)]


def _tp_owns(p, grid) -> bool:
    if getattr(p, "is_expert_parallel", False):
        return True
    if getattr(p, "tensor_parallel", False):
        return True
    return grid.rank_in("tp") == 0 or grid.size("tp") == 1


def _dup_rank(p, grid) -> int:
    """replica index over the data-parallel axis for this param."""
    if getattr(p, "is_expert_parallel", False):
        return grid.rank_in("expert_dp")
    return grid.rank_in("dp_cp")


def _emit_pieces(atlas: List[AtlasEntry], flat_tensor: torch.Tensor, lo: int, hi: int,
                 key_prefix: str, key_suffix: str, replica: int, out: Dict[str, ShardedTensor]):
    """Intersect [lo, hi) of the param's local flat space with the atlas and
    emit ShardedTensors for each overlap.  flat_tensor is indexed in [lo, hi)
    coordinates (i.e. flat_tensor[0] is element `lo` of the param)."""
    for ent in atlas:
        olo, ohi = max(lo, ent.flat_start), min(hi, ent.flat_end)
        if olo >= ohi:
            continue
        sub_lo = ent.sub_flat_start + (olo - ent.flat_start)
        sub_hi = sub_lo + (ohi - olo)
        key = key_prefix + ent.key + key_suffix
        full_cover = (olo - lo) == 0 and (ohi - olo) == flat_tensor.numel() and \
            sub_lo == 0 and (sub_hi - sub_lo) == _prod(ent.local_shape)
        data = flat_tensor if full_cover and flat_tensor.shape == torch.Size(ent.local_shape) \
            else flat_tensor.view(-1)[olo - lo: ohi - lo]
        out[f"{key}@{sub_lo}"] = ShardedTensor(
            key=key, data=data, global_shape=ent.global_shape,
            global_offset=ent.global_offset, local_shape=ent.local_shape,
            flattened_range=None if (full_cover and data.dim() > 1) else (sub_lo, sub_hi),
            replica_id=replica,
        )


def model_sharded_state_dict(model, prefix: str = "model."):
    """Returns (key->ShardedTensor, param->(gname, atlas, tp_owns))."""
    core = model.module if hasattr(model, "module") else model
    grid = G.get_grid()
    ren = {}
    for i, layer in enumerate(core.decoder.layers):
        ren[f"decoder.layers.{i}."] = f"decoder.layers.{layer.layer_number}."
    out: Dict[str, ShardedTensor] = {}
    param_map = {}
    for name, p in core.named_parameters():
        gname = name
        for loc, glob in ren.items():
            if gname.startswith(loc):
                gname = glob + gname[len(loc):]
                break
        gname = prefix + gname
        atlas = param_atlas(p, gname, grid)
        tp_owns = _tp_owns(p, grid)
        replica = _dup_rank(p, grid) + (0 if tp_owns else 1000 + grid.rank_in("tp"))
        _emit_pieces(atlas, p.data, 0, p.numel(), "", "", replica, out)
        param_map[p] = (gname, atlas, tp_owns)
    # router buffers (expert_bias) and other persistent buffers
    for name, buf in core.named_buffers():
        gname = name
        for loc, glob in ren.items():
            if gname.startswith(loc):
                gname = glob + gname[len(loc):]
                break
        gname = prefix + gname
        replica = grid.rank_in("dp_cp") + (0 if grid.rank_in("tp") == 0 else 1000)
        out[gname] = ShardedTensor(
            key=gname, data=buf.data, global_shape=tuple(buf.shape),
            global_offset=tuple(0 for _ in buf.shape), replica_id=replica,
        )
    return out, param_map


def optimizer_sharded_state_dict(optimizer, param_maps: List[dict]):
    from megatron_amd.optimizer.dist_optimizer import DistributedOptimizer

    merged = {}
    for pm in param_maps:
        merged.update(pm)
    grid = G.get_grid()
    out: Dict[str, ShardedTensor] = {}
    for sub in optimizer.chained_optimizers:
        if isinstance(sub, DistributedOptimizer):
            for seg in sub.segments:
                if seg.param not in merged:
                    continue
                gname, atlas, tp_owns = merged[seg.param]
                lo, hi = seg.key[1], seg.key[2]
                for sname, tensor in (("main", seg.main), ("exp_avg", seg.exp_avg),
                                      ("exp_avg_sq", seg.exp_avg_sq)):
                    _emit_pieces(atlas, tensor, lo, hi, "optimizer.", f".{sname}",
                                 0 if tp_owns else 1, out)
        else:
            dp_rank = grid.rank_in("dp_cp")
            params = sub.params
            mains = getattr(sub, "main_params", [None] * len(params))
            for p, main, ea, eas in zip(params, mains, sub.exp_avg, sub.exp_avg_sq):
                if p not in merged:
                    continue
                gname, atlas, tp_owns = merged[p]
                states = [("exp_avg", ea), ("exp_avg_sq", eas)]
                if main is not None:
                    states.append(("main", main))
                replica = dp_rank * 2 + (0 if tp_owns else 1)
                for sname, tensor in states:
                    _emit_pieces(atlas, tensor, 0, p.numel(), "optimizer.", f".{sname}",
                                 replica, out)
    return out
