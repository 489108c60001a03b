"""GPU inference tests: KV-cache engines on MI355X must match the no-cache
oracle through the native flash-attention kernel (bf16 tolerance: greedy token
agreement)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from megatron_amd.config import TransformerConfig  # noqa: E402
from megatron_amd.inference import (  # noqa: E402
    DynamicInferenceEngine,
    SamplingParams,
    StaticInferenceEngine,
)
from megatron_amd.models.gpt import GPTModel  # noqa: E402
from megatron_amd.parallel import grid as G  # noqa: E402
from megatron_amd.parallel.random import model_parallel_seed  # noqa: E402


@pytest.fixture()
def gpu_model():
    from megatron_amd import ops

    assert ops.has_native(), "native extension required on GPU"
    G.destroy_model_parallel()
    G.initialize_model_parallel(world_size=1, rank=0)
    model_parallel_seed(77)
    cfg = TransformerConfig(
        num_layers=2, hidden_size=256, num_attention_heads=4, num_query_groups=2,
        ffn_hidden_size=512, vocab_size=512, max_position_embeddings=512, bf16=True,
    )
    model = GPTModel(cfg).cuda().eval()
    yield model
    G.destroy_model_parallel()


def _oracle_greedy(model, prompt, n_new):
    toks = list(prompt)
    out = []
    for _ in range(n_new):
        with torch.no_grad():
            logits = model(torch.tensor([toks], device="cuda"))
        out.append(int(logits[-1, 0].float().argmax()))
        toks.append(out[-1])
    return out


def test_static_engine_gpu_matches_oracle(gpu_model):
    prompts = [[3, 7, 11, 2, 9, 100, 42], [5, 1]]
    eng = StaticInferenceEngine(gpu_model, max_batch=4, max_seq=256)
    res = eng.generate(prompts, SamplingParams(max_tokens=8, greedy=True, stop_on_eod=False))
    for p, r in zip(prompts, res):
        assert r.output_tokens == _oracle_greedy(gpu_model, p, 8)


def test_dynamic_engine_gpu_matches_static(gpu_model):
    prompts = [[3, 7, 11, 2, 9], [5, 1], list(range(40, 90))]
    params = SamplingParams(max_tokens=6, greedy=True, stop_on_eod=False)
    static = StaticInferenceEngine(gpu_model, max_batch=4, max_seq=256).generate(prompts, params)
    dynamic = DynamicInferenceEngine(gpu_model, num_blocks=32, block_size=16,
                                     max_prefill_tokens=16).generate(prompts, params)
    for a, b in zip(static, dynamic):
        assert a.output_tokens == b.output_tokens


def test_graph_decode_matches_eager(gpu_model):
    prompts = [[3, 7, 11, 2, 9], [5, 1], list(range(40, 80))]
    params = SamplingParams(max_tokens=8, greedy=True, stop_on_eod=False)
    eager = DynamicInferenceEngine(gpu_model, num_blocks=32, block_size=16,
                                   use_hip_graphs=False).generate(prompts, params)
    graphed = DynamicInferenceEngine(gpu_model, num_blocks=32, block_size=16,
                                     use_hip_graphs=True).generate(prompts, params)
    for a, b in zip(eager, graphed):
        assert a.output_tokens == b.output_tokens


def test_speculative_decode_gpu_matches_plain(gpu_model):
    """Draft-verify speculative decoding on MI355X: exact greedy equality
    through the native flash prefill kernels, for oracle and wrong drafts."""
    eng = StaticInferenceEngine(gpu_model, max_batch=2, max_seq=256)
    prompts = [[3, 7, 11, 2, 9], [5, 1, 4]]
    params = SamplingParams(max_tokens=10, greedy=True, stop_on_eod=False)
    plain = eng.generate(prompts, params)
    oracle = {tuple(p): r.output_tokens for p, r in zip(prompts, plain)}

    def oracle_draft(toks):
        for p, out in oracle.items():
            if tuple(toks[: len(p)]) == p and toks[len(p):] == out[: len(toks) - len(p)]:
                done = len(toks) - len(p)
                return out[done:done + 3]
        return []

    for draft_fn in (oracle_draft, lambda t: [0, 0], None):
        got = eng.generate_speculative(prompts, params, draft_fn=draft_fn, num_draft=3)
        for a, b in zip(plain, got):
            assert a.output_tokens == b.output_tokens
