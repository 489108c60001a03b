"""Context parallelism: ring attention over RCCL p2p (xGMI is fully connected,
so the ring sends are single-hop) and Ulysses-style head-scatter all-to-all.

Capability analog of the reference's CP support (parallel_state.py:124-131 CP
groups, extensions/transformer_engine.py:2086-2113 cp_comm_type p2p/a2a,
core/utils.py get_batch_on_this_cp_rank, rope_utils.py:48 freq slicing) — but
the ring exchange and online-softmax merge are implemented here on top of our
own flash kernel instead of inside a vendor library.

Load-balanced causal sharding (p2p mode): the sequence is cut into 2*cp
chunks; rank r holds chunks (r, 2cp-1-r), so every rank does the same amount
of causal-attention work. Chunk ids order the masking: a q-chunk attends a
kv-chunk fully if q_id > kv_id, causally if equal, not at all if less.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.distributed as dist

from megatron_amd import ops
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.mappings import all_to_all


# ---------------------------------------------------------------------------
# batch / position sharding
# ---------------------------------------------------------------------------

def cp_chunk_ids(cp_rank: int, cp_size: int):
    return (cp_rank, 2 * cp_size - 1 - cp_rank)


def slice_for_cp_rank(x: torch.Tensor, cp_rank: int, cp_size: int, seq_dim: int = 1,
                      mode: str = "p2p") -> torch.Tensor:
    """Slice a full-sequence tensor to this CP rank's shard.

    p2p (ring): load-balanced 2-chunk layout; a2a (Ulysses): contiguous."""
    if cp_size == 1:
        return x
    s = x.size(seq_dim)
    if mode == "a2a":
        assert s % cp_size == 0
        return x.narrow(seq_dim, cp_rank * (s // cp_size), s // cp_size)
    assert s % (2 * cp_size) == 0, f"seq {s} must divide 2*cp={2*cp_size}"
    L = s // (2 * cp_size)
    c0, c1 = cp_chunk_ids(cp_rank, cp_size)
    return torch.cat([x.narrow(seq_dim, c0 * L, L), x.narrow(seq_dim, c1 * L, L)], dim=seq_dim)


def cp_rope_positions(seq_len_global: int, cp_rank: int, cp_size: int,
                      device, mode: str = "p2p") -> torch.Tensor:
    """Global positions of this rank's tokens (for RoPE table indexing)."""
    if cp_size == 1:
        return torch.arange(seq_len_global, device=device)
    if mode == "a2a":
        L = seq_len_global // cp_size
        return torch.arange(cp_rank * L, (cp_rank + 1) * L, device=device)
    L = seq_len_global // (2 * cp_size)
    c0, c1 = cp_chunk_ids(cp_rank, cp_size)
    return torch.cat([torch.arange(c0 * L, (c0 + 1) * L, device=device),
                      torch.arange(c1 * L, (c1 + 1) * L, device=device)])


class _CPGatherSeq(torch.autograd.Function):
    """All-gather CP sequence shards back into NATURAL order.

    Forward: all-gather + chunk reorder (inverts slice_for_cp_rank's
    load-balanced layout).  Backward: sum the full-sequence grads over the
    CP group (every rank consumed the full sequence) and slice this rank's
    shard back out — the gather/reduce-scatter adjoint pair, done as
    all-reduce+slice so it also runs on gloo."""

    @staticmethod
    def forward(ctx, x, seq_dim, mode):
        cp = G.get_context_parallel_world_size()
        ctx.seq_dim, ctx.mode, ctx.cp = seq_dim, mode, cp
        if cp == 1:
            return x
        group = G.get_grid().group("cp")
        shards = [torch.empty_like(x) for _ in range(cp)]
        dist.all_gather(shards, x.contiguous(), group=group)
        if mode == "a2a":
            return torch.cat(shards, dim=seq_dim)
        L = x.size(seq_dim) // 2
        slots = [None] * (2 * cp)
        for r in range(cp):
            c0, c1 = cp_chunk_ids(r, cp)
            slots[c0] = shards[r].narrow(seq_dim, 0, L)
            slots[c1] = shards[r].narrow(seq_dim, L, L)
        return torch.cat(slots, dim=seq_dim)

    @staticmethod
    def backward(ctx, g):
        if ctx.cp == 1:
            return g, None, None
        group = G.get_grid().group("cp")
        g = g.contiguous()
        dist.all_reduce(g, group=group)
        r = G.get_context_parallel_rank()
        return slice_for_cp_rank(g, r, ctx.cp, seq_dim=ctx.seq_dim, mode=ctx.mode), None, None


class _CPSeqToChannel(torch.autograd.Function):
    """CP all-to-all turning a sequence-sharded full-channel tensor into a
    full-sequence channel-sharded one (Ulysses-style, for the Mamba scan:
    each rank scans the WHOLE sequence for 1/cp of the channels instead of
    redundantly scanning everything).

    forward:  [s/cp, b, C] -> [s, b, C/cp] in NATURAL sequence order
              (inverts the zigzag load-balanced layout for mode "p2p")
    backward: exact inverse all-to-all.
    """

    @staticmethod
    def forward(ctx, x, mode):
        cp = G.get_context_parallel_world_size()
        ctx.mode, ctx.cp = mode, cp
        if cp == 1:
            return x
        group = G.get_grid().group("cp")
        s_loc, b, C = x.shape
        ctx.shape_in = (s_loc, b, C)
        send = x.view(s_loc, b, cp, C // cp).permute(2, 0, 1, 3).contiguous()
        recv = torch.empty_like(send)
        dist.all_to_all_single(recv, send, group=group)
        # recv[r] = rank r's seq shard of my channel chunk (rank order)
        if mode == "a2a":
            return recv.reshape(cp * s_loc, b, C // cp)
        L = s_loc // 2
        slots = [None] * (2 * cp)
        for r in range(cp):
            c0, c1 = cp_chunk_ids(r, cp)
            slots[c0] = recv[r].narrow(0, 0, L)
            slots[c1] = recv[r].narrow(0, L, L)
        return torch.cat(slots, dim=0)

    @staticmethod
    def backward(ctx, g):
        if ctx.cp == 1:
            return g, None
        group = G.get_grid().group("cp")
        s_loc, b, C = ctx.shape_in
        cp = ctx.cp
        g = g.contiguous()
        if ctx.mode == "a2a":
            send = g.view(cp, s_loc, b, C // cp)
        else:
            L = s_loc // 2
            parts = []
            for r in range(cp):
                c0, c1 = cp_chunk_ids(r, cp)
                parts.append(torch.cat([g.narrow(0, c0 * L, L), g.narrow(0, c1 * L, L)], dim=0))
            send = torch.stack(parts, dim=0)
        recv = torch.empty_like(send)
        dist.all_to_all_single(recv, send.contiguous(), group=group)
        out = recv.permute(1, 2, 0, 3).reshape(s_loc, b, C)
        return out, None


class _CPChannelToSeq(torch.autograd.Function):
    """Inverse of _CPSeqToChannel: [s, b, C/cp] -> [s/cp, b, C]."""

    @staticmethod
    def forward(ctx, y, mode):
        cp = G.get_context_parallel_world_size()
        ctx.mode, ctx.cp = mode, cp
        if cp == 1:
            return y
        group = G.get_grid().group("cp")
        s_full, b, Cc = y.shape
        s_loc = s_full // cp
        ctx.shape_out = (s_loc, b, Cc)
        if mode == "a2a":
            send = y.contiguous().view(cp, s_loc, b, Cc)
        else:
            L = s_loc // 2
            parts = []
            for r in range(cp):
                c0, c1 = cp_chunk_ids(r, cp)
                parts.append(torch.cat([y.narrow(0, c0 * L, L), y.narrow(0, c1 * L, L)], dim=0))
            send = torch.stack(parts, dim=0)
        recv = torch.empty_like(send)
        dist.all_to_all_single(recv, send.contiguous(), group=group)
        return recv.permute(1, 2, 0, 3).reshape(s_loc, b, cp * Cc)

    @staticmethod
    def backward(ctx, g):
        if ctx.cp == 1:
            return g, None
        group = G.get_grid().group("cp")
        s_loc, b, Cc = ctx.shape_out
        cp = ctx.cp
        send = g.contiguous().view(s_loc, b, cp, Cc).permute(2, 0, 1, 3).contiguous()
        recv = torch.empty_like(send)
        dist.all_to_all_single(recv, send, group=group)
        if ctx.mode == "a2a":
            return recv.reshape(cp * s_loc, b, Cc), None
        L = s_loc // 2
        slots = [None] * (2 * cp)
        for r in range(cp):
            c0, c1 = cp_chunk_ids(r, cp)
            slots[c0] = recv[r].narrow(0, 0, L)
            slots[c1] = recv[r].narrow(0, L, L)
        return torch.cat(slots, dim=0), None


def cp_seq_to_channel(x: torch.Tensor, mode: str = "p2p") -> torch.Tensor:
    return _CPSeqToChannel.apply(x, mode)


def cp_channel_to_seq(y: torch.Tensor, mode: str = "p2p") -> torch.Tensor:
    return _CPChannelToSeq.apply(y, mode)


def gather_cp_sequence(x: torch.Tensor, seq_dim: int = 0, mode: str = "p2p") -> torch.Tensor:
    """Differentiable shard -> full-sequence gather (natural order)."""
    return _CPGatherSeq.apply(x, seq_dim, mode)


def get_batch_on_this_cp_rank(batch: dict, mode: str = "p2p") -> dict:
    cp = G.get_context_parallel_world_size()
    if cp == 1:
        return batch
    r = G.get_context_parallel_rank()
    return {k: (slice_for_cp_rank(v, r, cp, seq_dim=1, mode=mode)
                if v.dim() >= 2 else v) for k, v in batch.items()}


# ---------------------------------------------------------------------------
# attention partials (native flash kernel on GPU, fp32 torch on CPU)
# ---------------------------------------------------------------------------

def _pair_mask(s, skv, device, causal, pos_off, window):
    """[s, skv] bool valid-mask for one chunk pair: query i attends key j iff
    0 <= (pos_off + i - j) (causal) and < window (if set).  pos_off is the
    global q-minus-kv position offset of the chunk pair (skv - s for the
    plain single-chunk case)."""
    m = None
    if causal:
        m = torch.ones(s, skv, dtype=torch.bool, device=device).tril_(pos_off)
    if window is not None and window > 0:
        w = torch.ones(s, skv, dtype=torch.bool, device=device).triu_(pos_off - window + 1)
        m = w if m is None else (m & w)
    return m


def _fwd_partial(q, k, v, causal: bool, scale: float, window=None, pos_off=None):
    """returns (out [s,b,hq,d] same dtype, lse [b,hq,s] fp32)."""
    if pos_off is None:
        pos_off = k.shape[0] - q.shape[0]
    if (ops.has_native() and q.is_cuda and window is None
            and pos_off == k.shape[0] - q.shape[0]):
        return ops._C.attn_fwd(q.contiguous(), k.contiguous(), v.contiguous(),
                               causal, scale, 0)
    s, b, hq, d = q.shape
    hkv = k.shape[2]
    rep = hq // hkv
    qf = q.permute(1, 2, 0, 3).float()
    kf = k.permute(1, 2, 0, 3).float().repeat_interleave(rep, dim=1)
    vf = v.permute(1, 2, 0, 3).float().repeat_interleave(rep, dim=1)
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale  # [b,hq,s,skv]
    mask = _pair_mask(s, k.shape[0], q.device, causal, pos_off, window)
    if mask is not None:
        scores = scores.masked_fill(~mask, float("-inf"))
    lse = torch.logsumexp(scores, dim=-1)  # [b,hq,s]
    out = torch.matmul(torch.softmax(scores, dim=-1), vf)
    out = torch.nan_to_num(out)  # fully-masked rows (out-of-window) -> 0
    return out.permute(2, 0, 1, 3).to(q.dtype), lse


def _bwd_partial(dout, q, k, v, out, lse, causal: bool, scale: float,
                 window=None, pos_off=None):
    """FA2-style manual backward of one partial (global out/lse): returns
    (dq, dk, dv) in fp32."""
    if pos_off is None:
        pos_off = k.shape[0] - q.shape[0]
    if (ops.has_native() and q.is_cuda and window is None
            and pos_off == k.shape[0] - q.shape[0]):
        dq, dk, dv = ops._C.attn_bwd(dout.contiguous(), q.contiguous(), k.contiguous(),
                                     v.contiguous(), out.contiguous(), lse.contiguous(),
                                     causal, scale, 0)
        return dq.float(), dk.float(), dv.float()
    s, b, hq, d = q.shape
    hkv = k.shape[2]
    rep = hq // hkv
    qf = q.permute(1, 2, 0, 3).float()
    kf = k.permute(1, 2, 0, 3).float().repeat_interleave(rep, dim=1)
    vf = v.permute(1, 2, 0, 3).float().repeat_interleave(rep, dim=1)
    dof = dout.permute(1, 2, 0, 3).float()
    of = out.permute(1, 2, 0, 3).float()
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    p = torch.exp(scores - lse.unsqueeze(-1))  # uses the GLOBAL lse
    mask = _pair_mask(s, k.shape[0], q.device, causal, pos_off, window)
    if mask is not None:
        p = p * mask
    dvf = torch.matmul(p.transpose(-1, -2), dof)
    dp = torch.matmul(dof, vf.transpose(-1, -2))
    drow = (dof * of).sum(-1, keepdim=True)
    ds = p * (dp - drow) * scale
    dqf = torch.matmul(ds, kf)
    dkf = torch.matmul(ds.transpose(-1, -2), qf)
    if rep > 1:
        dkf = dkf.view(b, hkv, rep, *dkf.shape[2:]).sum(2)
        dvf = dvf.view(b, hkv, rep, *dvf.shape[2:]).sum(2)
    perm = lambda t: t.permute(2, 0, 1, 3)
    return perm(dqf), perm(dkf), perm(dvf)


def _merge(o_run, lse_run, o_new, lse_new):
    """online-softmax merge of two partials (fp32)."""
    lse_max = torch.maximum(lse_run, lse_new)
    a = torch.exp(lse_run - lse_max)
    bexp = torch.exp(lse_new - lse_max)
    lse_out = lse_max + torch.log(a + bexp)
    w1 = torch.exp(lse_run - lse_out).permute(2, 0, 1).unsqueeze(-1)  # [s,b,h,1]
    w2 = torch.exp(lse_new - lse_out).permute(2, 0, 1).unsqueeze(-1)
    return o_run * w1 + o_new.float() * w2, lse_out


def _ring_peers(group):
    cp = dist.get_world_size(group)
    ranks = dist.get_process_group_ranks(group)
    r = ranks.index(dist.get_rank())
    return cp, r, ranks[(r + 1) % cp], ranks[(r - 1) % cp], ranks


def _ring_sendrecv_begin(send: torch.Tensor, group):
    """start one ring step: send to next, receive from prev (batched p2p).
    Returns (recv_buffer, work_handles) - call _ring_wait before reading."""
    cp, r, nxt, prv, _ = _ring_peers(group)
    recv = torch.empty_like(send)
    ops_ = [dist.P2POp(dist.isend, send.contiguous(), nxt, group=group),
            dist.P2POp(dist.irecv, recv, prv, group=group)]
    return recv, dist.batch_isend_irecv(ops_)


def _ring_wait(works):
    for w in works:
        w.wait()


def _ring_sendrecv(send: torch.Tensor, group) -> torch.Tensor:
    recv, works = _ring_sendrecv_begin(send, group)
    _ring_wait(works)
    return recv


class _RingAttention(torch.autograd.Function):
    """Causal ring flash attention over the CP group (C14/K2).

    q/k/v: [2L, b, h, d] in the load-balanced 2-chunk layout. KV rotates
    around the ring; partials merge by LSE. Backward re-rotates KV and
    accumulates dK/dV in a buffer that travels with them.
    """

    @staticmethod
    def forward(ctx, q, k, v, scale, group, window=None):
        cp, r, *_ = _ring_peers(group)
        L = q.shape[0] // 2
        dk_dim, dv_dim = k.shape[-1], v.shape[-1]  # may differ (MLA)
        my_chunks = cp_chunk_ids(r, cp)
        o = torch.zeros((*q.shape[:3], dv_dim), dtype=torch.float32, device=q.device)
        lse = torch.full((q.shape[1], q.shape[2], q.shape[0]), float("-inf"),
                         dtype=torch.float32, device=q.device)
        kv = torch.cat([k, v], dim=-1)  # one travel tensor even when dk != dv
        for step in range(cp):
            src = (r - step) % cp
            src_chunks = cp_chunk_ids(src, cp)
            # start the exchange BEFORE compute so it overlaps the partials
            works = None
            if step < cp - 1:
                kv_next, works = _ring_sendrecv_begin(kv, group)
            for qi in range(2):
                for ki in range(2):
                    if my_chunks[qi] < src_chunks[ki]:
                        continue
                    delta = (my_chunks[qi] - src_chunks[ki]) * L
                    if window is not None and window > 0 and delta - (L - 1) >= window:
                        continue  # whole pair outside the window
                    causal = my_chunks[qi] == src_chunks[ki]
                    qc = q[qi * L:(qi + 1) * L]
                    kc = kv[ki * L:(ki + 1) * L, ..., :dk_dim]
                    vc = kv[ki * L:(ki + 1) * L, ..., dk_dim:]
                    oc, lsec = _fwd_partial(qc, kc, vc, True, scale,
                                            window=window, pos_off=delta)
                    sl = slice(qi * L, (qi + 1) * L)
                    o_m, lse_m = _merge(o[sl], lse[..., sl], oc, lsec)
                    o[sl] = o_m
                    lse[..., sl] = lse_m
            if works is not None:
                _ring_wait(works)
                kv = kv_next
        out = o.to(q.dtype)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.scale, ctx.group, ctx.window = scale, group, window
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        scale, group, window = ctx.scale, ctx.group, ctx.window
        cp, r, *_ = _ring_peers(group)
        L = q.shape[0] // 2
        dk_dim = k.shape[-1]
        my_chunks = cp_chunk_ids(r, cp)
        dq = torch.zeros(q.shape, dtype=torch.float32, device=q.device)
        # kv and its gradient accumulator travel together around the ring;
        # after cp steps the accumulator is back at the kv owner.
        kv = torch.cat([k, v], dim=-1)
        dkv = torch.zeros(kv.shape, dtype=torch.float32, device=q.device)
        for step in range(cp):
            src = (r - step) % cp
            src_chunks = cp_chunk_ids(src, cp)
            for qi in range(2):
                for ki in range(2):
                    if my_chunks[qi] < src_chunks[ki]:
                        continue
                    delta = (my_chunks[qi] - src_chunks[ki]) * L
                    if window is not None and window > 0 and delta - (L - 1) >= window:
                        continue
                    sl = slice(qi * L, (qi + 1) * L)
                    kl = slice(ki * L, (ki + 1) * L)
                    dqc, dkc, dvc = _bwd_partial(
                        dout[sl], q[sl], kv[kl, ..., :dk_dim], kv[kl, ..., dk_dim:],
                        out[sl], lse[..., sl], True, scale,
                        window=window, pos_off=delta)
                    dq[sl] += dqc
                    dkv[kl, ..., :dk_dim] += dkc
                    dkv[kl, ..., dk_dim:] += dvc
            if step < cp - 1:
                kv = _ring_sendrecv(kv, group)
                dkv = _ring_sendrecv(dkv, group)
        # final rotation returns dkv to its owner (cp-1 hops done, one more)
        if cp > 1:
            dkv = _ring_sendrecv(dkv, group)
        return (dq.to(q.dtype), dkv[..., :dk_dim].to(q.dtype),
                dkv[..., dk_dim:].to(q.dtype), None, None, None)


def ring_attention(q, k, v, scale: Optional[float] = None, group=None, window=None):
    """q [2L,b,hq,d], k/v [2L,b,hkv,d] in CP 2-chunk layout -> out [2L,b,hq,d].
    `window`: causal sliding window in GLOBAL positions (masking is applied
    per chunk pair with the pair's position offset)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    group = group if group is not None else G.get_context_parallel_group()
    if group is None or dist.get_world_size(group) == 1:
        return ops.flash_attention(q, k, v, causal=True, scale=scale, window=window)
    return _RingAttention.apply(q, k, v, scale, group, window)


def ulysses_attention(q, k, v, scale: Optional[float] = None, group=None, window=None):
    """Ulysses a2a CP: scatter heads / gather sequence around full attention.

    q [s/cp, b, hq, d] (contiguous slicing) -> out [s/cp, b, hq, d]. Needs
    hq and hkv divisible by cp. Single-hop on fully-connected xGMI.
    """
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    group = group if group is not None else G.get_context_parallel_group()
    cp = dist.get_world_size(group) if group is not None else 1
    if cp == 1:
        return ops.flash_attention(q, k, v, causal=True, scale=scale, window=window)
    sl, b, hq, d = q.shape
    hkv = k.shape[2]
    assert hq % cp == 0 and hkv % cp == 0, "Ulysses needs heads divisible by cp"

    def sp2hp(x):
        s_, b_, h_, d_ = x.shape
        xt = x.reshape(s_, b_, cp, h_ // cp, d_).permute(2, 0, 1, 3, 4).contiguous()
        out = all_to_all(group, xt.view(cp * s_, b_, h_ // cp, d_))
        return out  # [s_full, b, h/cp, d]

    def hp2sp(x):
        s_full, d_ = x.shape[0], x.shape[-1]
        out = all_to_all(group, x.contiguous())
        s_ = s_full // cp
        return out.view(cp, s_, b, x.shape[2], d_).permute(1, 2, 0, 3, 4).reshape(s_, b, x.shape[2] * cp, d_)

    qh, kh, vh = sp2hp(q), sp2hp(k), sp2hp(v)
    if v.shape[-1] == d and d in (64, 128) and q.is_cuda:
        oh = ops.flash_attention(qh, kh, vh, causal=True, scale=scale, window=window)
    else:
        from megatron_amd.ops import reference as _ref

        oh = _ref.attention(qh, kh, vh, causal=True, scale=scale, window=window)
    return hp2sp(oh)
