"""Sharded, reshardable distributed checkpoint format.

Capability analog of reference megatron/core/dist_checkpointing/
(ShardedTensor mapping.py:52-91, save/load serialization.py:341/:69,
fully-parallel dedup via replica_id, reshard-on-load validation.py).

Format (directory):
  common.pt          — rank-0 non-sharded objects (iteration, args echo, ...)
  shards_r<R>.pt     — this rank's shard payloads {key: [piece, ...]}
  metadata.json      — global index: key -> global_shape/dtype + piece list

Each piece carries (global_offset, local_shape, flattened_range) so any
(TP, PP, EP, DP) layout can be reassembled and re-sliced on load — the
distributed optimizer's param-boundary-ignorant flat shards round-trip via
``flattened_range`` over the local (TP-)shard (reference
distrib_optimizer.py "fully_reshardable" format).
"""

from __future__ import annotations

import dataclasses
import json
import os
import threading
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist


@dataclass
class ShardedTensor:
    key: str
    data: torch.Tensor                       # local piece (flattened if flattened_range)
    global_shape: Tuple[int, ...]
    global_offset: Tuple[int, ...]           # element offset of this rank's shard
    local_shape: Optional[Tuple[int, ...]] = None  # shape of the (unflattened) shard
    flattened_range: Optional[Tuple[int, int]] = None  # [start, end) within the flat shard
    replica_id: int = 0                      # >0 -> duplicate, skipped at save

    def __post_init__(self):
        if self.local_shape is None:
            self.local_shape = tuple(self.data.shape)


def _is_main_writer(st: ShardedTensor) -> bool:
    return st.replica_id == 0


def _async_payload_writer(payload, dest):
    """Runs in a spawned background process (async save)."""
    torch.save(payload, dest)


def save(sharded_state: Dict[str, ShardedTensor], common_state: dict, path: str,
         async_save: bool = False) -> Optional[threading.Thread]:
    """Collective save.  Every rank contributes its pieces; rank 0 writes
    common.pt + metadata.json (metadata gathered over the world group)."""
    rank = dist.get_rank() if dist.is_initialized() else 0
    world = dist.get_world_size() if dist.is_initialized() else 1
    os.makedirs(path, exist_ok=True)

    payload: Dict[str, list] = {}
    meta_local: Dict[str, dict] = {}
    for key, st in sharded_state.items():
        if not _is_main_writer(st):
            continue
        piece_meta = {
            "global_offset": list(st.global_offset),
            "local_shape": list(st.local_shape),
            "flattened_range": list(st.flattened_range) if st.flattened_range else None,
            "file": f"shards_r{rank}.pt",
        }
        meta_local.setdefault(st.key, {
            "global_shape": list(st.global_shape),
            "dtype": str(st.data.dtype).replace("torch.", ""),
            "pieces": [],
        })["pieces"].append(piece_meta)
        payload.setdefault(st.key, []).append(st.data.detach().cpu().clone())

    def _write_payload():
        torch.save(payload, os.path.join(path, f"shards_r{rank}.pt"))

    if dist.is_initialized() and world > 1:
        all_meta = [None] * world
        dist.all_gather_object(all_meta, meta_local)
    else:
        all_meta = [meta_local]

    writer = None
    if async_save:
        # background PROCESS (reference strategies/async_utils.py): torch.save
        # serialization + disk IO run outside this process's GIL.  Payload
        # tensors are moved to shared memory so the spawn pickling passes FD
        # handles, not copies.
        import torch.multiprocessing as tmp

        for blobs in payload.values():
            for t in blobs:
                t.share_memory_()
        ctx = tmp.get_context("spawn")
        writer = ctx.Process(
            target=_async_payload_writer,
            args=(payload, os.path.join(path, f"shards_r{rank}.pt")),
            daemon=False,
        )
        writer.start()
    else:
        _write_payload()

    if rank == 0:
        merged: Dict[str, dict] = {}
        for m in all_meta:
            for key, entry in m.items():
                tgt = merged.setdefault(key, {
                    "global_shape": entry["global_shape"],
                    "dtype": entry["dtype"],
                    "pieces": [],
                })
                assert tgt["global_shape"] == entry["global_shape"], f"shape mismatch for {key}"
                tgt["pieces"].extend(entry["pieces"])
        with open(os.path.join(path, "metadata.json"), "w") as f:
            json.dump(merged, f)
        torch.save(common_state, os.path.join(path, "common.pt"))
    if dist.is_initialized() and world > 1 and not async_save:
        dist.barrier()
    return writer


class _ShardReader:
    def __init__(self, path: str):
        self.path = path
        with open(os.path.join(path, "metadata.json")) as f:
            self.meta = json.load(f)
        self._files: Dict[str, dict] = {}

    def _file(self, name: str) -> dict:
        if name not in self._files:
            self._files[name] = torch.load(os.path.join(self.path, name), map_location="cpu",
                                           weights_only=False)
        return self._files[name]

    def assemble(self, key: str) -> torch.Tensor:
        """Reconstruct the FULL global tensor for a key."""
        entry = self.meta[key]
        dtype = getattr(torch, entry["dtype"])
        out = torch.empty(entry["global_shape"], dtype=dtype)
        consumed: Dict[str, int] = {}
        for piece in entry["pieces"]:
            fname = piece["file"]
            blobs = self._file(fname)[key]
            idx = consumed.get((fname, key), 0)
            # pieces from the same file were appended in save order
            data = blobs[idx]
            consumed[(fname, key)] = idx + 1
            off = piece["global_offset"]
            lshape = piece["local_shape"]
            view = out
            for d, (o, l) in enumerate(zip(off, lshape)):
                view = view.narrow(d, o, l)
            if piece["flattened_range"]:
                s0, s1 = piece["flattened_range"]
                # view may be non-contiguous (narrowed region): read-modify-write
                flat = view.reshape(-1)  # copy if non-contiguous
                flat[s0:s1] = data.reshape(-1)
                if flat.data_ptr() != view.data_ptr():
                    view.copy_(flat.view(view.shape))
            else:
                view.copy_(data.view(lshape))
        return out


def _boxes_overlap(off_a, shape_a, off_b, shape_b):
    for oa, la, ob, lb in zip(off_a, shape_a, off_b, shape_b):
        if oa + la <= ob or ob + lb <= oa:
            return False
    return True


def _apply_box_piece(piece, data, st):
    """Rectangular piece -> copy its intersection with st's box into st.data
    (st may itself carry a flattened_range: read-modify-write its box)."""
    poff, pshape = piece["global_offset"], piece["local_shape"]
    soff, sshape = st.global_offset, st.local_shape
    # intersection in global coords
    ioff = [max(po, so) for po, so in zip(poff, soff)]
    iend = [min(po + pl, so + sl) for po, pl, so, sl in zip(poff, pshape, soff, sshape)]
    ishape = [e - o for o, e in zip(ioff, iend)]
    pview = data.view(pshape)
    for d, (o, l) in enumerate(zip(ioff, ishape)):
        pview = pview.narrow(d, o - poff[d], l)
    if st.flattened_range is None:
        sview = st.data.view(st.local_shape)
        for d, (o, l) in enumerate(zip(ioff, ishape)):
            sview = sview.narrow(d, o - soff[d], l)
        sview.copy_(pview.to(st.data.dtype))
    else:
        # st.data is flat [t0, t1) over its box: rmw via a box-sized scratch
        t0, t1 = st.flattened_range
        scratch = torch.empty(sshape, dtype=st.data.dtype)
        flat = scratch.reshape(-1)
        flat[t0:t1] = st.data.reshape(-1)
        sview = scratch
        for d, (o, l) in enumerate(zip(ioff, ishape)):
            sview = sview.narrow(d, o - soff[d], l)
        sview.copy_(pview.to(st.data.dtype))
        st.data.copy_(scratch.reshape(-1)[t0:t1].view(st.data.shape))


def load(sharded_state: Dict[str, ShardedTensor], path: str) -> dict:
    """Fill each requested ShardedTensor's .data in place from the checkpoint,
    resharding as needed.  Returns the common state.

    Windowed (reference fully_parallel.py:522 / exchange_utils): shard files
    are visited ONE AT A TIME and only pieces overlapping a request are
    copied, so peak host memory is ~(one shard file + one scratch box), not
    the full global model.  Keys whose source pieces carry flattened_range
    over a DIFFERENT box than the request (cross-layout optimizer reshard)
    fall back to full-key assembly — bounded by that key's param size."""
    reader = _ShardReader(path)
    requests: Dict[str, List[ShardedTensor]] = {}
    for _, st in sharded_state.items():
        if st.key not in reader.meta:
            raise KeyError(f"checkpoint missing {st.key}")
        requests.setdefault(st.key, []).append(st)

    # classify keys; build per-file work lists with per-(file,key) blob index
    assemble_keys = set()
    file_work: Dict[str, list] = {}
    for key, sts in requests.items():
        entry = reader.meta[key]
        per_file_idx: Dict[str, int] = {}
        for piece in entry["pieces"]:
            fname = piece["file"]
            idx = per_file_idx.get(fname, 0)
            per_file_idx[fname] = idx + 1
            if piece["flattened_range"]:
                same_box_ok = all(
                    st.flattened_range is not None
                    and list(st.global_offset) == piece["global_offset"]
                    and list(st.local_shape) == piece["local_shape"]
                    for st in sts
                    if _boxes_overlap(piece["global_offset"], piece["local_shape"],
                                      st.global_offset, st.local_shape)
                )
                if not same_box_ok:
                    assemble_keys.add(key)
            file_work.setdefault(fname, []).append((key, idx, piece))

    for fname in sorted(file_work):
        blobs = reader._file(fname)
        for key, idx, piece in file_work[fname]:
            if key in assemble_keys:
                continue
            data = blobs[key][idx]
            for st in requests[key]:
                if not _boxes_overlap(piece["global_offset"], piece["local_shape"],
                                      st.global_offset, st.local_shape):
                    continue
                if piece["flattened_range"]:
                    # same box guaranteed by classification: flat-segment copy
                    s0, s1 = piece["flattened_range"]
                    t0, t1 = st.flattened_range
                    lo, hi = max(s0, t0), min(s1, t1)
                    if lo < hi:
                        st.data.reshape(-1)[lo - t0 : hi - t0] = (
                            data.reshape(-1)[lo - s0 : hi - s0].to(st.data.dtype))
                else:
                    _apply_box_piece(piece, data, st)
        reader._files.clear()  # one shard file resident at a time

    for key in assemble_keys:
        full = reader.assemble(key)
        for st in requests[key]:
            view = full
            for d, (o, l) in enumerate(zip(st.global_offset, st.local_shape)):
                view = view.narrow(d, o, l)
            if st.flattened_range:
                s0, s1 = st.flattened_range
                src = view.reshape(-1)[s0:s1]
                st.data.copy_(src.view(st.data.shape).to(st.data.dtype))
            else:
                st.data.copy_(view.to(st.data.dtype))
        reader._files.clear()

    common = torch.load(os.path.join(path, "common.pt"), map_location="cpu", weights_only=False)
    return common
