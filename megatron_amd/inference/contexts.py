"""Inference KV-cache contexts.

Capability analog of reference megatron/core/inference/contexts/
(StaticInferenceContext; DynamicInferenceContext dynamic_context.py:297 with
KVBlockAllocator kv_block_allocator.py): a context owns the KV memory and the
attention over it. The attention module calls ``context.attend(layer, q, k, v)``
so each context picks its kernel:

* prefill  — contiguous flash attention (hand-written MFMA kernel), new KV
             appended to the cache first,
* decode   — batched masked attention over per-request lengths (paged gather
             for the dynamic context).

MI355X note: 288 GB HBM3E makes KV capacity cheap — default block size is
large (256 tokens) to keep block tables small and gathers coalesced.
"""

from __future__ import annotations

import math
from typing import List, Optional, Tuple

import torch

from megatron_amd import ops


def _masked_decode_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                             context_lens: torch.Tensor, scale: float) -> torch.Tensor:
    """One-token-per-request attention with per-request KV lengths.

    q: [b, hq, d]; k/v: [b, L, hkv, d] (rows beyond context_lens[i] are
    garbage); returns [b, hq, d]. fp32 softmax.
    """
    b, hq, d = q.shape
    hkv = k.shape[2]
    rep = hq // hkv
    qf = q.float().view(b, hkv, rep, d)
    kf = k.float().permute(0, 2, 1, 3)  # [b, hkv, L, d]
    vf = v.float().permute(0, 2, 1, 3)
    scores = torch.einsum("bgrd,bgld->bgrl", qf, kf) * scale  # [b, hkv, rep, L]
    L = k.shape[1]
    mask = torch.arange(L, device=q.device)[None, :] >= context_lens[:, None]  # [b, L]
    scores = scores.masked_fill(mask[:, None, None, :], float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    out = torch.einsum("bgrl,bgld->bgrd", probs, vf)
    return out.reshape(b, hq, d).to(q.dtype)


class StaticInferenceContext:
    """Preallocated contiguous KV cache for a fixed batch (static engine).

    Cache layout per layer: [max_seq, max_batch, hkv, d] (seq-major matches
    the flash kernel's [s, b, h, d] operand layout — appends are slice writes,
    no transpose).
    """

    def __init__(self, num_layers: int, max_batch: int, max_seq: int,
                 num_kv_heads: int, head_dim: int, dtype=torch.bfloat16, device="cuda"):
        self.max_batch, self.max_seq = max_batch, max_seq
        # zeros not empty: masked attention still computes 0*v for padded
        # slots, so inf/NaN garbage in reused memory would poison the output
        self.k_cache = [torch.zeros(max_seq, max_batch, num_kv_heads, head_dim,
                                    dtype=dtype, device=device) for _ in range(num_layers)]
        self.v_cache = [torch.zeros_like(self.k_cache[0]) for _ in range(num_layers)]
        # per-row filled length; uniform during same-length prefill, ragged after
        self.context_lens = torch.zeros(max_batch, dtype=torch.long, device=device)
        self.batch_size = 0

    def reset(self, batch_size: int):
        self.batch_size = batch_size
        self.context_lens.zero_()

    def set_prompt_lens(self, lens: List[int]):
        self.context_lens[: len(lens)] = torch.as_tensor(lens, device=self.context_lens.device)

    def rope_positions(self, s: int) -> torch.Tensor:
        """Positions of the s tokens being fed this step: [s] (uniform) or
        [s, b] (ragged decode)."""
        lens = self.context_lens[: self.batch_size]
        if s > 1 or bool((lens == lens[0]).all()):
            base = int(lens[0])
            return torch.arange(base, base + s, device=lens.device)
        return lens.view(1, -1)  # [1, b]: each row decodes at its own position

    def attend(self, layer: int, q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
               scale: float, window=None) -> torch.Tensor:
        s, b = q.shape[0], q.shape[1]
        lens = self.context_lens[:b]
        kc, vc = self.k_cache[layer], self.v_cache[layer]
        if s > 1:
            # prefill (uniform offset across rows)
            off = int(lens[0])
            kc[off:off + s, :b] = k
            vc[off:off + s, :b] = v
            return ops.flash_attention(q, kc[:off + s, :b], vc[:off + s, :b],
                                       causal=True, scale=scale, window=window)
        # ragged decode: scatter new kv at each row's length
        idx = lens  # [b]
        ar = torch.arange(b, device=q.device)
        kc[idx, ar] = k[0]
        vc[idx, ar] = v[0]
        new_lens = lens + 1
        Lmax = int(new_lens.max())
        out = _masked_decode_attention(
            q[0], kc[:Lmax, :b].permute(1, 0, 2, 3), vc[:Lmax, :b].permute(1, 0, 2, 3),
            new_lens, scale)
        return out.unsqueeze(0)

    def advance(self, n: int = 1):
        self.context_lens[: self.batch_size] += n


class KVBlockAllocator:
    """Fixed-size KV block free-list (reference kv_block_allocator.py)."""

    def __init__(self, num_blocks: int, first_id: int = 0):
        self.num_blocks = num_blocks
        self._free = list(range(first_id + num_blocks - 1, first_id - 1, -1))

    @property
    def num_free(self) -> int:
        return len(self._free)

    def allocate(self, n: int) -> List[int]:
        if n > len(self._free):
            raise RuntimeError(f"out of KV blocks: need {n}, have {len(self._free)}")
        out = [self._free.pop() for _ in range(n)]
        return out

    def free(self, blocks: List[int]):
        self._free.extend(blocks)


class PrefixCachingAllocator(KVBlockAllocator):
    """Refcounted block allocator with content-hash prefix reuse (reference
    analog: vLLM-style automatic prefix caching adapted to the paged
    context; the reference's engines gained this via its kv reuse work).

    Only FULL prompt blocks are registered (keyed by the running hash of the
    token chain), so shared blocks are immutable and no copy-on-write is
    needed: decode always appends into an unshared tail block.  Freed
    cached blocks keep their KV and move to an LRU; allocation evicts LRU
    entries only when the free list is empty."""

    def __init__(self, num_blocks: int, first_id: int = 0):
        super().__init__(num_blocks, first_id)
        self._hash_to_block: dict = {}
        self._block_to_hash: dict = {}
        self._ref: dict = {}
        self._lru: dict = {}   # block -> tick (ref==0 cached blocks)
        self._tick = 0
        self.hits = 0
        self.misses = 0

    @property
    def num_free(self) -> int:
        return len(self._free) + len(self._lru)

    def allocate(self, n: int) -> List[int]:
        out = []
        for _ in range(n):
            if self._free:
                out.append(self._free.pop())
            elif self._lru:
                blk = min(self._lru, key=self._lru.get)  # evict oldest
                del self._lru[blk]
                h = self._block_to_hash.pop(blk, None)
                if h is not None:
                    self._hash_to_block.pop(h, None)
                self._ref.pop(blk, None)
                out.append(blk)
            else:
                self._free.extend(out)
                raise RuntimeError(f"out of KV blocks: need {n}")
        for b in out:
            self._ref[b] = 1
        return out

    def free(self, blocks: List[int]):
        for b in blocks:
            r = self._ref.get(b, 1) - 1
            if r > 0:
                self._ref[b] = r
                continue
            self._ref.pop(b, None)
            if b in self._block_to_hash:
                self._tick += 1
                self._lru[b] = self._tick   # keep KV for reuse
            else:
                self._free.append(b)

    # -- prefix reuse -----------------------------------------------------

    def lookup(self, chain_hash) -> Optional[int]:
        blk = self._hash_to_block.get(chain_hash)
        if blk is None:
            self.misses += 1
            return None
        self.hits += 1
        if blk in self._lru:        # revive from the evictable pool
            del self._lru[blk]
            self._ref[blk] = 1
        else:
            self._ref[blk] = self._ref.get(blk, 0) + 1
        return blk

    def register(self, chain_hash, block: int):
        if chain_hash in self._hash_to_block:
            return
        self._hash_to_block[chain_hash] = block
        self._block_to_hash[block] = chain_hash


def prompt_block_hashes(prompt: List[int], block_size: int) -> List[int]:
    """Running content hash per FULL block of the prompt."""
    out = []
    h = 0
    for i in range(len(prompt) // block_size):
        h = hash((h, tuple(prompt[i * block_size : (i + 1) * block_size])))
        out.append(h)
    return out


class DynamicInferenceContext:
    """Paged KV cache for continuous batching (reference dynamic_context.py:297).

    Cache per layer: [num_blocks, block_size, hkv, d]. Each active request has
    a block table; the engine assigns a dense row index per step.
    """

    def __init__(self, num_layers: int, num_kv_heads: int, head_dim: int,
                 num_blocks: int = 512, block_size: int = 256,
                 dtype=torch.bfloat16, device="cuda", kv_cache_dtype=None):
        self.block_size = block_size
        self.num_layers = num_layers
        # physical block 0 is a scratch target for padded rows in hipGraph
        # decode replays; the allocator only hands out ids 1..num_blocks
        self.allocator = KVBlockAllocator(num_blocks, first_id=1)
        # kv_cache_dtype="fp8": halve cache memory by storing e4m3 values with
        # one fp32 scale per (slot, head) — quantize on write, dequantize on
        # the paged gather (reference kv-cache quantization role)
        self.kv_fp8 = kv_cache_dtype in ("fp8", torch.float8_e4m3fn)
        store_dtype = torch.float8_e4m3fn if self.kv_fp8 else dtype
        self.compute_dtype = dtype
        self.k_cache = [torch.zeros(num_blocks + 1, block_size, num_kv_heads, head_dim,
                                    dtype=store_dtype, device=device) for _ in range(num_layers)]
        self.v_cache = [torch.zeros_like(self.k_cache[0]) for _ in range(num_layers)]
        if self.kv_fp8:
            self.k_scale = [torch.ones((num_blocks + 1) * block_size, num_kv_heads,
                                       dtype=torch.float32, device=device) for _ in range(num_layers)]
            self.v_scale = [torch.ones_like(self.k_scale[0]) for _ in range(num_layers)]
        self.device = device
        self._static = False
        # step state, set by the engine before each forward
        self._mode: str = "decode"  # or "prefill"
        self._block_tables: Optional[torch.Tensor] = None  # [b, max_blocks] int64
        self._context_lens: Optional[torch.Tensor] = None  # [b] lengths BEFORE this step
        self._prefill_table: Optional[torch.Tensor] = None  # [nblocks] for the one prefill req

    # -- engine-facing step setup -------------------------------------------

    def begin_prefill(self, block_table: List[int], prior_len: int):
        self._static = False
        self._mode = "prefill"
        self._prefill_table = torch.as_tensor(block_table, dtype=torch.long, device=self.device)
        self._prior_len = prior_len

    def begin_decode_static(self, tables_buf: torch.Tensor, lens_buf: torch.Tensor):
        """Graph-capture decode mode: fixed-shape persistent buffers; the
        gather always spans the full (padded) table width so no host-side
        data-dependent shapes appear inside the captured region."""
        self._mode = "decode"
        self._static = True
        self._block_tables = tables_buf
        self._context_lens = lens_buf

    def begin_decode(self, block_tables: List[List[int]], context_lens: List[int]):
        self._static = False
        self._mode = "decode"
        maxb = max(len(t) for t in block_tables)
        bt = torch.zeros(len(block_tables), maxb, dtype=torch.long, device=self.device)
        for i, t in enumerate(block_tables):
            bt[i, : len(t)] = torch.as_tensor(t, dtype=torch.long, device=self.device)
        self._block_tables = bt
        self._context_lens = torch.as_tensor(context_lens, dtype=torch.long, device=self.device)

    def rope_positions(self, s: int) -> torch.Tensor:
        if self._mode == "prefill":
            return torch.arange(self._prior_len, self._prior_len + s, device=self.device)
        return self._context_lens.view(1, -1)  # [1, b]

    # -- attention ------------------------------------------------------------

    def _slot_index(self, positions: torch.Tensor, table: torch.Tensor) -> torch.Tensor:
        """positions [n] within a request -> flat slot ids block*bs+off."""
        blk = table[torch.div(positions, self.block_size, rounding_mode="floor")]
        return blk * self.block_size + positions % self.block_size

    def _quantize(self, x):
        """x [..., hkv, d] -> (fp8 payload, fp32 scale [..., hkv])."""
        amax = x.detach().float().abs().amax(dim=-1).clamp(min=1e-8)
        sc = amax / 448.0
        qx = (x.float() / sc.unsqueeze(-1)).clamp(-448, 448).to(torch.float8_e4m3fn)
        return qx, sc

    def _store(self, layer, flat_k, flat_v, slots, k, v):
        if not self.kv_fp8:
            flat_k[slots] = k
            flat_v[slots] = v
            return
        qk, sk = self._quantize(k)
        qv, sv = self._quantize(v)
        flat_k[slots] = qk
        flat_v[slots] = qv
        self.k_scale[layer][slots] = sk
        self.v_scale[layer][slots] = sv

    def _load(self, layer, flat, scales, slots, kind):
        x = flat[slots]
        if not self.kv_fp8:
            return x
        return (x.float() * scales[slots].unsqueeze(-1)).to(self.compute_dtype)

    def attend(self, layer: int, q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
               scale: float, window=None) -> torch.Tensor:
        kc, vc = self.k_cache[layer], self.v_cache[layer]
        bs = self.block_size
        flat_k = kc.view(-1, *kc.shape[2:])  # [num_blocks*bs, hkv, d]
        flat_v = vc.view(-1, *vc.shape[2:])
        if self._mode == "prefill":
            s = q.shape[0]
            assert q.shape[1] == 1, "prefill is one request at a time"
            pos = torch.arange(self._prior_len, self._prior_len + s, device=self.device)
            slots = self._slot_index(pos, self._prefill_table)
            self._store(layer, flat_k, flat_v, slots, k[:, 0], v[:, 0])
            if self._prior_len == 0 and not self.kv_fp8:
                return ops.flash_attention(q, k, v, causal=True, scale=scale, window=window)
            # chunked prefill: gather prior + current contiguous KV.  (fp8
            # mode reads back even fresh writes so prefill and decode see the
            # SAME quantized values — keeps the two paths consistent.)
            all_pos = torch.arange(0, self._prior_len + s, device=self.device)
            all_slots = self._slot_index(all_pos, self._prefill_table)
            k_full = self._load(layer, flat_k, self.k_scale[layer] if self.kv_fp8 else None,
                                all_slots, "k").unsqueeze(1)  # [L, 1, hkv, d]
            v_full = self._load(layer, flat_v, self.v_scale[layer] if self.kv_fp8 else None,
                                all_slots, "v").unsqueeze(1)
            return ops.flash_attention(q, k_full, v_full, causal=True, scale=scale, window=window)
        # decode: scatter the new token, then paged gather + masked attention
        b = q.shape[1]
        lens = self._context_lens
        rows = torch.arange(b, device=self.device)
        slots = (self._block_tables[rows, torch.div(lens, bs, rounding_mode="floor")] * bs
                 + lens % bs)
        self._store(layer, flat_k, flat_v, slots, k[0], v[0])
        new_lens = lens + 1
        if self._static:
            tables = self._block_tables  # fixed width under graph capture
        else:
            max_blocks_needed = int(torch.div(new_lens.max() + bs - 1, bs, rounding_mode="floor"))
            tables = self._block_tables[:, :max_blocks_needed]  # [b, nb]
        # gather [b, nb*bs, hkv, d] via block-id expansion
        slot_grid = (tables.unsqueeze(-1) * bs +
                     torch.arange(bs, device=self.device).view(1, 1, bs)).reshape(b, -1)
        k_gath = self._load(layer, flat_k, self.k_scale[layer] if self.kv_fp8 else None,
                            slot_grid, "k")  # [b, L, hkv, d]
        v_gath = self._load(layer, flat_v, self.v_scale[layer] if self.kv_fp8 else None,
                            slot_grid, "v")
        out = _masked_decode_attention(q[0], k_gath, v_gath, new_lens, scale)
        return out.unsqueeze(0)
