"""Disaggregated prefill/decode serving.

Capability analog of reference megatron/core/inference/disaggregation/:
prefill (compute-bound, fills whole KV blocks) and decode (latency-bound,
one token per step) run on SEPARATE engine instances — in production on
separate GPUs sized independently — with the prompt's KV cache handed off
after prefill.

The KV package rides the host-offload path (offload.py): the prefill
engine's blocks are swapped out of its pool into (pinned) host memory and
swapped into the decode engine's pool.  Between two MI355X GPUs the same
package could move over xGMI peer copies instead; the host hop is the
portable baseline and is what a cross-node deployment uses anyway.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional, Sequence

import torch

from megatron_amd.inference.engine import DynamicInferenceEngine, GenerationResult, _full_logits, _Request
from megatron_amd.inference.offload import KVHostOffloader
from megatron_amd.inference.sampling import SamplingParams


@dataclass
class KVPackage:
    """A prefilled request's state, detached from any engine."""

    prompt: List[int]
    first_token: int
    k_host: torch.Tensor  # [L, n_blocks, bs, hkv, d]
    v_host: torch.Tensor
    block_size: int
    latents: Optional[dict] = None  # MLA latent-pool blocks per layer


class PrefillWorker:
    """Runs chunked prefill only; emits KVPackages."""

    def __init__(self, engine: DynamicInferenceEngine):
        self.engine = engine

    @torch.no_grad()
    def prefill(self, prompt: Sequence[int], params: SamplingParams = SamplingParams()) -> KVPackage:
        eng = self.engine
        ctx = eng.context
        prompt = list(prompt)
        need = eng._blocks_for(len(prompt) + 1)
        table = ctx.allocator.allocate(need)
        cached = 0
        logits_tp = None
        while cached < len(prompt):
            chunk = min(eng.max_prefill_tokens, len(prompt) - cached)
            ctx.begin_prefill(table, cached)
            toks = torch.as_tensor(prompt[cached:cached + chunk], device=eng.device).view(1, -1)
            logits_tp = eng.model(toks, inference_context=ctx)
            cached += chunk
        first = int(_full_logits(logits_tp[-1, 0]).float().argmax()) if params.greedy else None
        if first is None:
            from megatron_amd.inference.sampling import sample

            first = int(sample(_full_logits(logits_tp[-1, 0]).float().unsqueeze(0), params)[0])
        # package the blocks and free them from the prefill pool
        idx = torch.as_tensor(table, dtype=torch.long, device=ctx.device)
        k = torch.stack([ctx.k_cache[l][idx] for l in range(ctx.num_layers)]).cpu()
        v = torch.stack([ctx.v_cache[l][idx] for l in range(ctx.num_layers)]).cpu()
        pools = getattr(ctx, "mla_latent_pool", None)
        latents = {l: p[idx].cpu() for l, p in pools.items()} if pools else None
        ctx.allocator.free(table)
        return KVPackage(prompt=prompt, first_token=first, k_host=k, v_host=v,
                         block_size=ctx.block_size, latents=latents)


class DecodeWorker:
    """Adopts KVPackages into its own paged pool and continues decoding."""

    def __init__(self, engine: DynamicInferenceEngine):
        self.engine = engine

    def adopt(self, pkg: KVPackage, params: SamplingParams = SamplingParams()) -> int:
        eng = self.engine
        ctx = eng.context
        assert ctx.block_size == pkg.block_size, "pool block sizes must match"
        n = pkg.k_host.shape[1]
        blocks = ctx.allocator.allocate(n)
        idx = torch.as_tensor(blocks, dtype=torch.long, device=ctx.device)
        for l in range(ctx.num_layers):
            ctx.k_cache[l][idx] = pkg.k_host[l].to(ctx.device)
            ctx.v_cache[l][idx] = pkg.v_host[l].to(ctx.device)
        if pkg.latents:
            pools = getattr(ctx, "mla_latent_pool", None)
            if pools is None:
                pools = {}
                ctx.mla_latent_pool = pools
            for l, lat in pkg.latents.items():
                if l not in pools:  # seed the decode engine's pool lazily
                    pools[l] = torch.zeros(ctx.k_cache[0].shape[0], ctx.block_size,
                                           lat.shape[-1], dtype=torch.float32,
                                           device=ctx.device)
                pools[l][idx] = lat.to(ctx.device)
        # register as an in-flight request that already emitted first_token
        rid = next(eng._ids)
        req = _Request(rid, list(pkg.prompt), params,
                       GenerationResult(rid, list(pkg.prompt)))
        req.cached = len(pkg.prompt)
        req.block_table = blocks
        req.result.output_tokens.append(pkg.first_token)
        req.next_input = pkg.first_token
        if params.stop_on_eod and pkg.first_token == eng.eod:
            req.result.output_tokens.pop()
            eng._finish(req)
        elif params.max_tokens <= 1:
            eng._finish(req)
        else:
            eng.active.append(req)
        return rid

    def run(self) -> None:
        while self.engine.has_work():
            self.engine.step()


def disaggregated_generate(prefill_engine: DynamicInferenceEngine,
                           decode_engine: DynamicInferenceEngine,
                           prompts: Sequence[Sequence[int]],
                           params: SamplingParams = SamplingParams()) -> List[GenerationResult]:
    """Convenience driver: prefill each prompt on one engine, decode on the
    other, preserving input order."""
    pw = PrefillWorker(prefill_engine)
    dw = DecodeWorker(decode_engine)
    rids = [dw.adopt(pw.prefill(p, params), params) for p in prompts]
    dw.run()
    return [decode_engine.finished.pop(r) for r in rids]
