"""Token sampling (capability analog of reference inference text-generation
controller sampling: greedy / temperature / top-k / top-p, log-probs)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch


@dataclass
class SamplingParams:
    max_tokens: int = 64
    temperature: float = 1.0
    top_k: int = 0          # 0 = off
    top_p: float = 0.0      # 0 = off
    greedy: bool = False
    stop_on_eod: bool = True
    return_log_probs: bool = False
    seed: Optional[int] = None
    # generation stops (token excluded) when the detokenized output ends
    # with any of these (requires a tokenizer on the engine)
    stop_strings: tuple = ()
    # multiplicative penalty on logits of already-generated tokens (>1
    # discourages repeats; HF convention: divide positive / multiply negative)
    repetition_penalty: float = 1.0
    # attach the top-N (token, logprob) alternatives per generated position
    top_n_logprobs: int = 0
    # keep only tokens with prob >= min_p * max_prob (0 = off)
    min_p: float = 0.0
    # additive per-token-id logit bias {token_id: bias} (OpenAI logit_bias)
    logit_bias: Optional[dict] = None
    # also return log p(prompt[i] | prompt[:i]) from prefill (RL / eval scoring)
    prompt_logprobs: bool = False


def filter_logits(logits: torch.Tensor, top_k: int = 0, top_p: float = 0.0,
                  min_p: float = 0.0) -> torch.Tensor:
    """Mask logits outside top-k / nucleus top-p / min-p to -inf. [b, V]."""
    if min_p > 0.0:
        probs = torch.softmax(logits.float(), dim=-1)
        keep = probs >= min_p * probs.amax(dim=-1, keepdim=True)
        logits = logits.masked_fill(~keep, float("-inf"))
    if top_k > 0:
        kth = torch.topk(logits, min(top_k, logits.size(-1)), dim=-1).values[..., -1, None]
        logits = logits.masked_fill(logits < kth, float("-inf"))
    if top_p > 0.0:
        sorted_logits, sorted_idx = torch.sort(logits, descending=True, dim=-1)
        probs = torch.softmax(sorted_logits, dim=-1)
        cum = torch.cumsum(probs, dim=-1)
        # keep the smallest prefix with cumulative prob >= top_p (always >= 1 token)
        drop = cum - probs >= top_p
        sorted_logits = sorted_logits.masked_fill(drop, float("-inf"))
        logits = torch.full_like(logits, float("-inf")).scatter_(-1, sorted_idx, sorted_logits)
    return logits


def apply_repetition_penalty(logits: torch.Tensor, prev_tokens,
                             penalty: float) -> torch.Tensor:
    """HF-convention repetition penalty: for each already-seen token id,
    positive logits are divided by `penalty`, negative multiplied.
    logits [b, V]; prev_tokens: per-row sequences of ids."""
    if penalty == 1.0:
        return logits
    logits = logits.clone()
    for i, toks in enumerate(prev_tokens):
        if not len(toks):
            continue
        ids = torch.as_tensor(list(set(toks)), device=logits.device, dtype=torch.long)
        row = logits[i, ids]
        logits[i, ids] = torch.where(row > 0, row / penalty, row * penalty)
    return logits


def sample(logits: torch.Tensor, params: SamplingParams,
           generator: Optional[torch.Generator] = None,
           prev_tokens=None) -> torch.Tensor:
    """logits: [b, V] (full vocab, fp32) -> next token ids [b]."""
    if params.repetition_penalty != 1.0 and prev_tokens is not None:
        logits = apply_repetition_penalty(logits, prev_tokens, params.repetition_penalty)
    if params.logit_bias:
        logits = logits.clone()
        for tid, bias in params.logit_bias.items():
            logits[:, int(tid)] += float(bias)
    if params.greedy or params.temperature == 0.0:
        return logits.argmax(dim=-1)
    logits = logits / max(params.temperature, 1e-6)
    logits = filter_logits(logits, params.top_k, params.top_p, params.min_p)
    probs = torch.softmax(logits.float(), dim=-1)
    return torch.multinomial(probs, 1, generator=generator).squeeze(-1)


def log_prob_of(logits: torch.Tensor, tokens: torch.Tensor) -> torch.Tensor:
    """log p(token) under logits. logits [b, V], tokens [b] -> [b]."""
    logp = torch.log_softmax(logits.float(), dim=-1)
    return logp.gather(-1, tokens.unsqueeze(-1)).squeeze(-1)


@torch.no_grad()
def beam_search(model, prompt, beam_width: int = 4, max_new_tokens: int = 16,
                eod: Optional[int] = None, length_penalty: float = 1.0):
    """Beam search decode (reference legacy text-generation beam search).

    Full-forward scoring (no KV reuse — correctness-first reference path;
    the cached engines cover throughput).  Returns a list of
    (tokens, score) sorted best-first, score = sum logp / len^length_penalty
    over the generated part."""
    device = next(model.parameters()).device
    beams = [(list(prompt), 0.0, False)]
    for _ in range(max_new_tokens):
        alive = [b for b in beams if not b[2]]
        if not alive:
            break
        batch = torch.tensor([b[0] for b in alive], device=device)
        logits = model(batch)  # [s, b, V]
        logp = torch.log_softmax(logits[-1].float(), dim=-1)  # [b, V]
        cands = [b for b in beams if b[2]]
        for (toks, score, _), row in zip(alive, logp):
            topv, topi = row.topk(min(beam_width, row.numel()))
            for v, i in zip(topv.tolist(), topi.tolist()):
                done = eod is not None and i == eod
                cands.append((toks + [i], score + v, done))
        cands.sort(key=lambda b: b[1], reverse=True)
        beams = cands[:beam_width]
    n_prompt = len(prompt)

    def final_score(b):
        gen_len = max(len(b[0]) - n_prompt, 1)
        return b[1] / (gen_len ** length_penalty)

    return [(b[0], final_score(b)) for b in sorted(beams, key=final_score, reverse=True)]
