"""Inference engines.

Capability analog of reference megatron/core/inference/engines/
(StaticInferenceEngine static_engine.py; DynamicInferenceEngine
dynamic_engine.py — continuous batching, paged KV, chunked prefill).

Static engine: fixed batch, contiguous KV. Prefill runs the whole padded
batch in one flash-attention pass (causal masking makes pad positions inert;
their KV slots are overwritten during decode). Dynamic engine: request queue,
paged KV blocks, one prefill or one batched decode step per engine step.
"""

from __future__ import annotations

import itertools
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

import torch
import torch.distributed as dist

from megatron_amd.inference.contexts import DynamicInferenceContext, StaticInferenceContext
from megatron_amd.inference.sampling import SamplingParams, log_prob_of, sample
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.mappings import gather_from_tensor_model_parallel_region


class _DecodeGraphRunner:
    """hipGraph-captured decode steps (reference: per-step CUDA-graph decode,
    dynamic_engine.py). One graph per (batch-bucket, table-width-bucket);
    inputs are copied into persistent buffers, padded rows write their KV into
    the reserved scratch block 0. Decode is launch-bound at small batch - the
    whole multi-layer step replays as one hipGraph."""

    def __init__(self, model, context, device):
        self.model, self.context, self.device = model, context, device
        self.graphs = {}

    @staticmethod
    def _bucket(n: int, lo: int = 8) -> int:
        b = lo
        while b < n:
            b *= 2
        return b

    def _build(self, bb: int, nbb: int):
        toks = torch.zeros(bb, 1, dtype=torch.long, device=self.device)
        tables = torch.zeros(bb, nbb, dtype=torch.long, device=self.device)
        lens = torch.ones(bb, dtype=torch.long, device=self.device)
        self.context.begin_decode_static(tables, lens)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):  # warmup per graph-capture rules
                self.model(toks, inference_context=self.context)
        torch.cuda.current_stream().wait_stream(side)
        g = torch.cuda.CUDAGraph()
        self.context.begin_decode_static(tables, lens)
        with torch.cuda.graph(g):
            out = self.model(toks, inference_context=self.context)
        return {"graph": g, "toks": toks, "tables": tables, "lens": lens, "out": out}

    def run(self, tok_list, tables_list, lens_list) -> torch.Tensor:
        b = len(tok_list)
        nb = max(len(t) for t in tables_list)
        key = (self._bucket(b), self._bucket(nb, lo=1))
        entry = self.graphs.get(key)
        if entry is None:
            entry = self._build(*key)
            self.graphs[key] = entry
        entry["toks"].zero_()
        entry["tables"].zero_()
        entry["lens"].fill_(1)
        entry["toks"][:b, 0] = torch.as_tensor(tok_list, device=self.device)
        for i, t in enumerate(tables_list):
            entry["tables"][i, : len(t)] = torch.as_tensor(t, dtype=torch.long, device=self.device)
        entry["lens"][:b] = torch.as_tensor(lens_list, device=self.device)
        entry["graph"].replay()
        return entry["out"][0, :b]


@dataclass
class GenerationResult:
    request_id: int
    prompt_tokens: List[int]
    output_tokens: List[int] = field(default_factory=list)
    log_probs: List[float] = field(default_factory=list)
    # per position: [(token, logprob) x top_n] when params.top_n_logprobs > 0
    top_logprobs: List[list] = field(default_factory=list)
    text: Optional[str] = None
    finished: bool = False
    aborted: bool = False
    # log p(prompt[i] | prompt[:i]) for i >= 1 (prompt_logprobs=True)
    prompt_log_probs: Optional[List[float]] = None


def _full_logits(logits_tp: torch.Tensor) -> torch.Tensor:
    """[.., V/tp] -> [.., V] (all-gather over TP; identity at TP=1)."""
    if G.get_tensor_model_parallel_world_size() == 1:
        return logits_tp
    return gather_from_tensor_model_parallel_region(logits_tp)


class StaticInferenceEngine:
    def __init__(self, model, tokenizer=None, max_batch: int = 8, max_seq: int = 2048,
                 device=None):
        self.model = model.eval()
        self.tokenizer = tokenizer
        cfg = model.config
        device = device or next(model.parameters()).device
        self.device = device
        self.eod = tokenizer.eod if tokenizer is not None else cfg.vocab_size - 1
        tp = G.get_tensor_model_parallel_world_size()
        self.context = StaticInferenceContext(
            cfg.num_layers, max_batch, max_seq,
            (cfg.num_query_groups or cfg.num_attention_heads) // tp,
            cfg.kv_channels, dtype=cfg.params_dtype, device=device)
        self.max_seq = max_seq

    def _pp_forward(self, tokens: torch.Tensor) -> Optional[torch.Tensor]:
        """Pipeline-parallel forward for inference (reference inference PP
        communication): stages relay hidden states over the PP group; logits
        exist on the last stage only, which returns them (None elsewhere).
        All ranks must call with the same tokens."""
        grid = G.get_grid()
        if grid.pp == 1:
            return self.model(tokens, inference_context=self.context)
        s = tokens.shape[1]
        b = tokens.shape[0]
        h = self.model.config.hidden_size
        dtype = self.model.config.params_dtype
        group = grid.group("pp")
        if not grid.is_pipeline_first_stage(ignore_virtual=True):
            buf = torch.empty(s, b, h, dtype=dtype, device=self.device)
            dist.recv(buf, src=grid.pipeline_prev_rank(), group=group)
            self.model.set_input_tensor(buf)
        out = self.model(tokens, inference_context=self.context)
        if not grid.is_pipeline_last_stage(ignore_virtual=True):
            dist.send(out.contiguous(), dst=grid.pipeline_next_rank(), group=group)
            return None
        return out

    def _pp_broadcast_tokens(self, toks: Optional[torch.Tensor], b: int) -> torch.Tensor:
        """Sampled tokens live on the last stage; share them with the rest."""
        grid = G.get_grid()
        if grid.pp == 1:
            return toks
        if toks is None:
            toks = torch.empty(b, dtype=torch.long, device=self.device)
        src = grid.ranks("pp")[-1]
        dist.broadcast(toks, src=src, group=grid.group("pp"))
        return toks

    @torch.no_grad()
    def generate(self, prompts: Sequence, params: SamplingParams = SamplingParams()) -> List[GenerationResult]:
        if self.tokenizer is not None and isinstance(prompts[0], str):
            prompts = [self.tokenizer.tokenize(p) for p in prompts]
        prompts = [list(p) for p in prompts]
        b = len(prompts)
        assert b <= self.context.max_batch
        lens = [len(p) for p in prompts]
        Lmax = max(lens)
        assert Lmax + params.max_tokens <= self.max_seq
        gen = None
        if params.seed is not None:
            gen = torch.Generator(device="cpu").manual_seed(params.seed)

        tokens = torch.full((b, Lmax), self.eod, dtype=torch.long, device=self.device)
        for i, p in enumerate(prompts):
            tokens[i, : len(p)] = torch.as_tensor(p, device=self.device)

        self.context.reset(b)
        pp_last = G.get_grid().is_pipeline_last_stage(ignore_virtual=True) if G.grid_initialized() else True
        logits_tp = self._pp_forward(tokens)  # [Lmax, b, V/tp] on the last stage
        self.context.set_prompt_lens(lens)
        if pp_last:
            last_pos = torch.as_tensor(lens, device=self.device) - 1
            last = logits_tp[last_pos, torch.arange(b, device=self.device)]  # [b, V/tp]
            logits = _full_logits(last).float()
        else:
            logits = None

        results = [GenerationResult(i, p) for i, p in enumerate(prompts)]
        if params.prompt_logprobs and pp_last:
            # log p(prompt[i] | prompt[:i]): logits row i-1 scores token i
            full = _full_logits(logits_tp).float().log_softmax(dim=-1)  # [L, b, V]
            for i, p in enumerate(prompts):
                ids = torch.as_tensor(p[1:], device=self.device)
                rows = full[: len(p) - 1, i]
                results[i].prompt_log_probs = rows.gather(
                    -1, ids.unsqueeze(-1)).squeeze(-1).tolist()
        finished = torch.zeros(b, dtype=torch.bool, device=self.device)
        for _ in range(params.max_tokens):
            if pp_last:
                next_tok = sample(logits.cpu() if gen is not None else logits, params, gen,
                                  prev_tokens=[r.output_tokens for r in results]).to(self.device)
                if params.return_log_probs:
                    lp = log_prob_of(logits, next_tok)
            else:
                next_tok = None
            next_tok = self._pp_broadcast_tokens(next_tok, b)
            for i in range(b):
                if not bool(finished[i]):
                    results[i].output_tokens.append(int(next_tok[i]))
                    if params.return_log_probs and pp_last:
                        results[i].log_probs.append(float(lp[i]))
            if params.stop_on_eod:
                finished |= next_tok == self.eod
            if bool(finished.all()):
                break
            logits_tp = self._pp_forward(next_tok.view(b, 1))
            self.context.advance(1)
            if pp_last:
                logits = _full_logits(logits_tp[0]).float()

        for r in results:
            if params.stop_on_eod and r.output_tokens and r.output_tokens[-1] == self.eod:
                r.output_tokens = r.output_tokens[:-1]
            r.finished = True
            if self.tokenizer is not None:
                r.text = self.tokenizer.detokenize(r.output_tokens)
        return results

    @torch.no_grad()
    def generate_speculative(self, prompts: Sequence, params: SamplingParams = SamplingParams(),
                             draft_fn=None, num_draft: int = 2) -> List[GenerationResult]:
        """Draft-verify speculative decoding (greedy, pp=1): each step feeds
        the committed token plus ``num_draft`` drafted tokens as one chunk;
        the model's own argmax verifies the drafts, accepting the longest
        matching prefix, so outputs are token-identical to plain greedy
        decode while accepted drafts cost one forward for several tokens.

        ``draft_fn(tokens) -> List[int]`` proposes continuations of the full
        context; default is prompt-lookup (n-gram match against the context
        — training-free, strongest on repetitive text).  Rejected draft KV
        entries are left in place and overwritten by the next chunk at the
        same offsets (the static cache masks by tracked length)."""
        assert params.greedy, "speculative decoding is exact for greedy only"
        assert not params.stop_strings and not params.return_log_probs, (
            "speculative decode: stop_strings/logprobs unsupported — use generate()")
        if not G.grid_initialized() or G.get_grid().pp == 1:
            pass
        else:
            raise NotImplementedError("speculative decode: pp=1 only")
        if self.tokenizer is not None and isinstance(prompts[0], str):
            prompts = [self.tokenizer.tokenize(p) for p in prompts]
        mtp_drafter = None
        if draft_fn == "mtp":
            mtp_drafter = MTPDrafter(self.model)
            draft_fn = None
        if draft_fn is None and mtp_drafter is None:
            draft_fn = lambda toks: _prompt_lookup_draft(toks, num_draft)
        try:
            return self._generate_speculative_impl(prompts, params, draft_fn,
                                                   mtp_drafter, num_draft)
        finally:
            if mtp_drafter is not None:
                mtp_drafter.remove()

    def _generate_speculative_impl(self, prompts, params, draft_fn, mtp_drafter,
                                   num_draft):

        results = []
        for rid, prompt in enumerate(prompts):
            prompt = list(prompt)
            assert len(prompt) + params.max_tokens + num_draft <= self.max_seq
            self.context.reset(1)
            toks = torch.as_tensor([prompt], device=self.device)
            logits_tp = self.model(toks, inference_context=self.context)
            self.context.set_prompt_lens([len(prompt)])
            cur = int(_full_logits(logits_tp[-1]).float().argmax(dim=-1)[0])
            out = [cur]
            cur_hidden_idx = len(prompt) - 1
            while len(out) < params.max_tokens and not (params.stop_on_eod and cur == self.eod):
                if mtp_drafter is not None:
                    pos = len(prompt) + len(out) - 1
                    drafts = mtp_drafter(mtp_drafter.hidden[cur_hidden_idx, 0], cur, pos)
                else:
                    drafts = [int(d) for d in (draft_fn(prompt + out) or [])]
                drafts = [int(d) for d in drafts][:num_draft]
                chunk = torch.as_tensor([[cur] + drafts], device=self.device)
                logits_tp = self.model(chunk, inference_context=self.context)
                preds = _full_logits(logits_tp[:, 0]).float().argmax(dim=-1)  # [1+D]
                accepted = 0
                for j, d in enumerate(drafts):
                    if int(preds[j]) == d:
                        accepted += 1
                    else:
                        break
                new = drafts[:accepted] + [int(preds[accepted])]
                if params.stop_on_eod:
                    cut = next((i + 1 for i, t in enumerate(new) if t == self.eod), len(new))
                    new = new[:cut]
                out.extend(new)
                cur = out[-1]
                cur_hidden_idx = accepted  # chunk position whose pred we committed
                self.context.advance(len(new))
            out = out[: params.max_tokens]
            if mtp_drafter is not None:
                mtp_drafter.hidden = None
            r = GenerationResult(rid, prompt)
            r.output_tokens = out
            if params.stop_on_eod and r.output_tokens and r.output_tokens[-1] == self.eod:
                r.output_tokens = r.output_tokens[:-1]
            r.finished = True
            if self.tokenizer is not None:
                r.text = self.tokenizer.detokenize(r.output_tokens)
            results.append(r)
        return results


class MTPDrafter:
    """Model-based drafter: runs the model's MTP head(s) statelessly on the
    last committed position (DeepSeek-V3-style self-speculation).  The head
    chain proposes one token per depth; the verify pass keeps outputs exact,
    so the single-position approximation (no cross-token attention inside
    the head) only affects the acceptance rate, never correctness."""

    def __init__(self, model):
        core = model.module if hasattr(model, "module") else model
        assert core.mtp is not None, "MTPDrafter needs a model built with mtp_num_layers"
        self.core = core
        self.hidden = None  # [s, 1, h] from the latest forward
        self._hook = core.decoder.register_forward_hook(
            lambda mod, args, out: setattr(self, "hidden", out.detach()))

    def __call__(self, position_hidden: torch.Tensor, next_token: int,
                 position: int) -> List[int]:
        core = self.core
        table = core._rotary_freqs(core.config.max_position_embeddings,
                                   position_hidden.device)
        drafts = []
        h_k = position_hidden.view(1, 1, -1)
        tok = next_token
        for head in core.mtp.heads:
            emb = core.embedding(torch.tensor([[tok]], device=h_k.device))
            fused = torch.cat([head.norm_hidden(h_k), head.norm_embed(emb)], dim=-1)
            h_k = head.layer(head.proj(fused),
                             rotary_freqs=table[position : position + 1])
            logits, _ = core.output_layer(h_k)
            tok = int(_full_logits(logits[0, 0]).float().argmax())
            drafts.append(tok)
        return drafts

    def remove(self):
        self._hook.remove()


def _prompt_lookup_draft(tokens: List[int], num_draft: int) -> List[int]:
    """Prompt-lookup drafting: find the most recent earlier occurrence of the
    trailing 2-gram and propose the tokens that followed it."""
    if len(tokens) < 3:
        return []
    key = tokens[-2:]
    for i in range(len(tokens) - 3, -1, -1):
        if tokens[i : i + 2] == key:
            return tokens[i + 2 : i + 2 + num_draft]
    return []


@dataclass
class _Request:
    rid: int
    prompt: List[int]
    params: SamplingParams
    result: GenerationResult
    block_table: List[int] = field(default_factory=list)
    cached: int = 0          # prompt tokens already in the KV cache
    next_input: Optional[int] = None  # token to feed at the next decode step
    gen: Optional[torch.Generator] = None


class DynamicInferenceEngine:
    """Continuous batching over a paged KV cache."""

    def __init__(self, model, tokenizer=None, num_blocks: int = 512, block_size: int = 256,
                 max_batch: int = 64, max_prefill_tokens: int = 8192, device=None,
                 use_hip_graphs: bool = True, enable_prefix_caching: bool = True,
                 kv_cache_dtype=None, scheduling_policy: str = "fcfs"):
        self.model = model.eval()
        self.tokenizer = tokenizer
        cfg = model.config
        device = device or next(model.parameters()).device
        self.device = device
        self.eod = tokenizer.eod if tokenizer is not None else cfg.vocab_size - 1
        tp = G.get_tensor_model_parallel_world_size()
        self.context = DynamicInferenceContext(
            cfg.num_layers,
            (cfg.num_query_groups or cfg.num_attention_heads) // tp,
            cfg.kv_channels, num_blocks=num_blocks, block_size=block_size,
            dtype=cfg.params_dtype, device=device, kv_cache_dtype=kv_cache_dtype)
        self.max_batch = max_batch
        self.max_prefill_tokens = max_prefill_tokens
        assert scheduling_policy in ("fcfs", "priority", "sjf")
        self.scheduling_policy = scheduling_policy
        self._graphs = None
        pp = G.get_grid().pp if G.grid_initialized() else 1
        if (use_hip_graphs and torch.cuda.is_available() and device.type == "cuda"
                and tp == 1 and pp == 1):
            self._graphs = _DecodeGraphRunner(self.model, self.context, device)
        self._ids = itertools.count()
        self.waiting: List[_Request] = []
        self.active: List[_Request] = []
        self.finished: Dict[int, GenerationResult] = {}
        # host-offload preemption (LIFO preempt, FIFO restore)
        from megatron_amd.inference.offload import KVHostOffloader

        self.offloader = KVHostOffloader(self.context)
        self.preempted: List[tuple] = []  # (req, handle)
        self.prefix_caching = enable_prefix_caching
        if enable_prefix_caching:
            from megatron_amd.inference.contexts import PrefixCachingAllocator

            # swap in the refcounted reuse allocator (same id space)
            self.context.allocator = PrefixCachingAllocator(num_blocks, first_id=1)

    def add_request(self, prompt, params: SamplingParams = SamplingParams(),
                    priority: int = 0) -> int:
        """Lower `priority` values schedule first (ties: FCFS).  With
        scheduling_policy="sjf", shorter prompts are prefilled first inside
        a priority class."""
        if self.tokenizer is not None and isinstance(prompt, str):
            prompt = self.tokenizer.tokenize(prompt)
        rid = next(self._ids)
        gen = torch.Generator(device="cpu").manual_seed(params.seed) if params.seed is not None else None
        req = _Request(rid, list(prompt), params, GenerationResult(rid, list(prompt)), gen=gen)
        req.priority = priority
        self.waiting.append(req)
        return rid

    def _next_waiting(self):
        """Pick the next prefill candidate per the scheduling policy."""
        if self.scheduling_policy == "fcfs":
            return self.waiting[0]
        if self.scheduling_policy == "sjf":
            return min(self.waiting,
                       key=lambda r: (getattr(r, "priority", 0), len(r.prompt), r.rid))
        return min(self.waiting, key=lambda r: (getattr(r, "priority", 0), r.rid))

    def has_work(self) -> bool:
        return bool(self.waiting or self.active or self.preempted)

    def abort(self, rid: int) -> bool:
        """Cancel a request in any state (waiting / active / preempted):
        its blocks (or host swap) are released and a finished-with-abort
        result is published.  Returns False for unknown/finished ids."""
        for q in (self.waiting, self.active):
            for req in q:
                if req.rid == rid:
                    q.remove(req)
                    self.context.allocator.free(req.block_table)
                    req.block_table = []
                    req.result.finished = True
                    req.result.aborted = True
                    self.finished[rid] = req.result
                    return True
        for i, (req, handle) in enumerate(self.preempted):
            if req.rid == rid:
                self.preempted.pop(i)
                self.offloader.drop(handle)
                req.result.finished = True
                req.result.aborted = True
                self.finished[rid] = req.result
                return True
        return False

    def _preempt_one(self, exclude=None) -> bool:
        """Swap the youngest active request's KV to host, freeing its blocks."""
        for victim in reversed(self.active):
            if victim is exclude:
                continue
            self.active.remove(victim)
            handle = self.offloader.swap_out(victim.block_table)
            victim.block_table = []
            self.preempted.append((victim, handle))
            return True
        return False

    def _restore_preempted(self):
        while self.preempted and len(self.active) < self.max_batch:
            req, handle = self.preempted[0]
            need = self.offloader.num_blocks_of(handle) + 1  # +1 headroom to decode
            if self.context.allocator.num_free < need:
                break
            self.preempted.pop(0)
            req.block_table = self.offloader.swap_in(handle)
            self.active.append(req)

    def _pp(self):
        grid = G.get_grid() if G.grid_initialized() else None
        return grid if (grid is not None and grid.pp > 1) else None

    def _model_forward(self, toks: torch.Tensor):
        """Model forward with pipeline-parallel hidden relay (logits on the
        last stage only; None elsewhere).  Mirrors the static engine."""
        grid = self._pp()
        if grid is None:
            return self.model(toks, inference_context=self.context)
        s, b = toks.shape[1], toks.shape[0]
        h = self.model.config.hidden_size
        group = grid.group("pp")
        if not grid.is_pipeline_first_stage(ignore_virtual=True):
            buf = torch.empty(s, b, h, dtype=self.model.config.params_dtype,
                              device=self.device)
            dist.recv(buf, src=grid.pipeline_prev_rank(), group=group)
            self.model.set_input_tensor(buf)
        out = self.model(toks, inference_context=self.context)
        if not grid.is_pipeline_last_stage(ignore_virtual=True):
            dist.send(out.contiguous(), dst=grid.pipeline_next_rank(), group=group)
            return None
        return out

    def _pp_share_tokens(self, toks: Optional[List[int]], n: int) -> List[int]:
        grid = self._pp()
        if grid is None:
            return toks
        t = (torch.as_tensor(toks, dtype=torch.long, device=self.device)
             if toks is not None else torch.empty(n, dtype=torch.long, device=self.device))
        dist.broadcast(t, src=grid.ranks("pp")[-1], group=grid.group("pp"))
        return t.tolist()

    def _blocks_for(self, n_tokens: int) -> int:
        bs = self.context.block_size
        return (n_tokens + bs - 1) // bs

    def _ensure_blocks(self, req: _Request, total_tokens: int) -> bool:
        need = self._blocks_for(total_tokens) - len(req.block_table)
        if need <= 0:
            return True
        if need > self.context.allocator.num_free:
            return False
        req.block_table.extend(self.context.allocator.allocate(need))
        return True

    def _reuse_prefix(self, req: _Request):
        """Adopt cached KV blocks for the longest full-block prompt prefix.
        The reused tokens are marked cached so chunked prefill starts after
        them (they are never recomputed).  At least one token is always
        prefilled so the first sampled logits exist."""
        from megatron_amd.inference.contexts import prompt_block_hashes

        bs = self.context.block_size
        hashes = prompt_block_hashes(req.prompt, bs)
        req._hashes = hashes
        usable = len(hashes)
        if usable * bs == len(req.prompt):
            usable -= 1  # keep >= 1 token to prefill
        for h in hashes[:usable]:
            blk = self.context.allocator.lookup(h)
            if blk is None:
                break
            req.block_table.append(blk)
            req.cached += bs

    def _register_prefix_blocks(self, req: _Request):
        """After prefill progress: publish newly completed full prompt
        blocks for reuse by later requests."""
        from megatron_amd.inference.contexts import prompt_block_hashes

        bs = self.context.block_size
        hashes = getattr(req, "_hashes", None)
        if hashes is None:
            hashes = prompt_block_hashes(req.prompt, bs)
            req._hashes = hashes
        full = min(req.cached // bs, len(hashes))
        for i in range(full):
            self.context.allocator.register(hashes[i], req.block_table[i])

    def _hit_stop_string(self, req: _Request) -> bool:
        if not req.params.stop_strings or self.tokenizer is None:
            return False
        text = self.tokenizer.detokenize(req.result.output_tokens)
        return any(text.endswith(ss) for ss in req.params.stop_strings)

    def _finish(self, req: _Request):
        self.context.allocator.free(req.block_table)
        req.block_table = []
        req.result.finished = True
        if self.tokenizer is not None:
            req.result.text = self.tokenizer.detokenize(req.result.output_tokens)
        self.finished[req.rid] = req.result

    def _sample_row(self, logits_row: torch.Tensor, req: _Request) -> int:
        logits_row = logits_row.unsqueeze(0)
        if req.gen is not None:
            logits_row = logits_row.cpu()
        tok = int(sample(logits_row, req.params, req.gen,
                         prev_tokens=[req.result.output_tokens])[0])
        if req.params.return_log_probs:
            req.result.log_probs.append(float(log_prob_of(logits_row, torch.tensor([tok]))[0]))
        if req.params.top_n_logprobs > 0:
            lp = torch.log_softmax(logits_row[0].float(), dim=-1)
            v, i = lp.topk(req.params.top_n_logprobs)
            req.result.top_logprobs.append(list(zip(i.tolist(), v.tolist())))
        return tok

    @torch.no_grad()
    def step(self) -> None:
        """One engine step: chunked prefill of the next waiting request, or one
        batched decode step over all active requests."""
        self._restore_preempted()
        if self.waiting and len(self.active) < self.max_batch:
            req = self._next_waiting()
            if self.prefix_caching and req.cached == 0 and not req.block_table:
                self._reuse_prefix(req)
            chunk = min(self.max_prefill_tokens, len(req.prompt) - req.cached)
            if not self._ensure_blocks(req, req.cached + chunk):
                if not self.active:
                    raise RuntimeError("KV pool too small for a single prompt chunk")
            else:
                self.context.begin_prefill(req.block_table, req.cached)
                toks = torch.as_tensor(req.prompt[req.cached:req.cached + chunk],
                                       device=self.device).view(1, -1)
                logits_tp = self._model_forward(toks)  # [chunk, 1, V/tp] on pp-last
                req.cached += chunk
                if self.prefix_caching:
                    self._register_prefix_blocks(req)
                if req.cached == len(req.prompt):
                    if logits_tp is not None:
                        logits = _full_logits(logits_tp[-1, 0]).float()
                        tok = self._sample_row(logits, req)
                    else:
                        tok = None
                    tok = self._pp_share_tokens([tok] if tok is not None else None, 1)[0]
                    req.result.output_tokens.append(tok)
                    self.waiting.remove(req)
                    if req.params.stop_on_eod and tok == self.eod:
                        req.result.output_tokens.pop()
                        self._finish(req)
                    elif len(req.result.output_tokens) >= req.params.max_tokens:
                        self._finish(req)
                    else:
                        req.next_input = tok
                        self.active.append(req)
                return

        if not self.active:
            return
        # batched decode; on pool exhaustion preempt youngest requests to host
        for req in list(self.active):
            if req not in self.active:  # already preempted this pass
                continue
            total = len(req.prompt) + len(req.result.output_tokens)
            while not self._ensure_blocks(req, total + 1):
                if not self._preempt_one(exclude=req):
                    raise RuntimeError("KV pool exhausted and nothing left to preempt")
        batch = self.active
        tables = [r.block_table for r in batch]
        lens = [len(r.prompt) + len(r.result.output_tokens) - 1 for r in batch]  # cached so far
        if self._graphs is not None:
            logits_row = self._graphs.run([r.next_input for r in batch], tables, lens)
            logits = _full_logits(logits_row).float()
        else:
            self.context.begin_decode(tables, lens)
            toks = torch.as_tensor([r.next_input for r in batch], device=self.device).view(-1, 1)
            logits_tp = self._model_forward(toks)  # [1, b, V/tp] on pp-last
            logits = _full_logits(logits_tp[0]).float() if logits_tp is not None else None
        pp_last = logits is not None
        # batched sampling fast path: all requests share one params object
        batch_toks = None
        p0 = batch[0].params
        if pp_last and all((r.params is p0) and r.gen is None for r in batch) and p0.top_n_logprobs == 0:
            batch_toks = sample(logits, p0,
                                prev_tokens=[r.result.output_tokens for r in batch]).tolist()
            if p0.return_log_probs:
                lps = log_prob_of(logits, torch.as_tensor(batch_toks, device=logits.device)).tolist()
        if self._pp() is not None:
            if batch_toks is None and pp_last:
                batch_toks = [self._sample_row(logits[i], r) for i, r in enumerate(batch)]
            batch_toks = self._pp_share_tokens(batch_toks, len(batch))
        still = []
        for i, req in enumerate(batch):
            if batch_toks is not None:
                tok = int(batch_toks[i])
                if pp_last and p0.return_log_probs and req.params is p0 and req.gen is None                         and self._pp() is not None:
                    pass  # per-row _sample_row already recorded probs on pp-last
                elif pp_last and batch_toks is not None and p0.return_log_probs                         and self._pp() is None:
                    req.result.log_probs.append(float(lps[i]))
            else:
                tok = self._sample_row(logits[i], req)
            req.result.output_tokens.append(tok)
            if req.params.stop_on_eod and tok == self.eod:
                req.result.output_tokens.pop()
                self._finish(req)
            elif self._hit_stop_string(req):
                self._finish(req)
            elif len(req.result.output_tokens) >= req.params.max_tokens:
                self._finish(req)
            else:
                req.next_input = tok
                still.append(req)
        self.active = still

    @torch.no_grad()
    def generate(self, prompts: Sequence, params: SamplingParams = SamplingParams()) -> List[GenerationResult]:
        ids = [self.add_request(p, params) for p in prompts]
        while self.has_work():
            self.step()
        return [self.finished.pop(i) for i in ids]
