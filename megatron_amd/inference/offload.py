"""KV-cache host offload (request preemption under memory pressure).

Capability analog of reference megatron/core/inference/unified_memory.py:
when the paged KV pool runs dry mid-decode, instead of failing, the engine
preempts the youngest requests — their KV blocks are swapped to (pinned)
host memory and the device blocks freed — and swaps them back in when
blocks free up.  On MI355X the swap rides the PCIe/host path, so the
policy is strictly LIFO-preempt / FIFO-restore to minimize thrash; with
288 GB HBM3E per GPU the pool is big, so this is a long-context safety
valve rather than a steady-state mechanism.
"""

from __future__ import annotations

import itertools
from typing import Dict, List, Tuple

import torch


class KVHostOffloader:
    def __init__(self, context):
        self.context = context
        self._store: Dict[int, Tuple[torch.Tensor, torch.Tensor]] = {}
        self._ids = itertools.count()

    def swap_out(self, block_table: List[int]) -> int:
        """Copy the blocks' K/V (all layers, plus any MLA latent pools) to
        host and free them.  Returns a handle for swap_in."""
        ctx = self.context
        idx = torch.as_tensor(block_table, dtype=torch.long, device=ctx.device)
        k = torch.stack([ctx.k_cache[l][idx] for l in range(ctx.num_layers)])
        v = torch.stack([ctx.v_cache[l][idx] for l in range(ctx.num_layers)])
        k_host, v_host = k.cpu(), v.cpu()
        if torch.cuda.is_available() and k.is_cuda:
            k_host, v_host = k_host.pin_memory(), v_host.pin_memory()
        latents = None
        pools = getattr(ctx, "mla_latent_pool", None)
        if pools:
            latents = {layer: pool[idx].cpu() for layer, pool in pools.items()}
        scales = None
        if getattr(ctx, "kv_fp8", False):
            # fp8 cache: the per-slot scales must travel with the payload
            bs = ctx.block_size
            slot = (idx.unsqueeze(-1) * bs
                    + torch.arange(bs, device=idx.device)).reshape(-1)
            scales = (torch.stack([ctx.k_scale[l][slot] for l in range(ctx.num_layers)]).cpu(),
                      torch.stack([ctx.v_scale[l][slot] for l in range(ctx.num_layers)]).cpu())
        handle = next(self._ids)
        self._store[handle] = (k_host, v_host, latents, scales)
        ctx.allocator.free(list(block_table))
        return handle

    def drop(self, handle: int) -> None:
        """Discard a swapped-out request's host copy (aborted requests)."""
        self._store.pop(handle, None)

    def num_blocks_of(self, handle: int) -> int:
        return self._store[handle][0].shape[1]

    def swap_in(self, handle: int) -> List[int]:
        """Re-allocate device blocks and restore the saved K/V into them.
        Returns the new block table."""
        ctx = self.context
        k_host, v_host, latents, scales = self._store.pop(handle)
        n = k_host.shape[1]
        blocks = ctx.allocator.allocate(n)
        idx = torch.as_tensor(blocks, dtype=torch.long, device=ctx.device)
        for l in range(ctx.num_layers):
            ctx.k_cache[l][idx] = k_host[l].to(ctx.device, non_blocking=True)
            ctx.v_cache[l][idx] = v_host[l].to(ctx.device, non_blocking=True)
        if latents:
            pools = getattr(ctx, "mla_latent_pool", {})
            for layer, lat in latents.items():
                if layer in pools:
                    pools[layer][idx] = lat.to(ctx.device, non_blocking=True)
        if scales is not None:
            bs = ctx.block_size
            slot = (idx.unsqueeze(-1) * bs
                    + torch.arange(bs, device=idx.device)).reshape(-1)
            for l in range(ctx.num_layers):
                ctx.k_scale[l][slot] = scales[0][l].to(ctx.device, non_blocking=True)
                ctx.v_scale[l][slot] = scales[1][l].to(ctx.device, non_blocking=True)
        if torch.cuda.is_available() and ctx.k_cache[0].is_cuda:
            torch.cuda.current_stream().synchronize()
        return blocks
