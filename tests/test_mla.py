"""Multi-latent attention tests (reference multi_latent_attention.py)."""

import torch

from megatron_amd.config import TransformerConfig
from megatron_amd.models.gpt import GPTModel
from megatron_amd.parallel.random import model_parallel_seed
from tests.utils import init_single


def _cfg(**kw):
    return TransformerConfig(num_layers=2, hidden_size=64, num_attention_heads=4,
                             num_query_groups=4, ffn_hidden_size=128, vocab_size=128,
                             max_position_embeddings=64, multi_latent_attention=True,
                             kv_lora_rank=32, qk_nope_head_dim=16, qk_rope_head_dim=16,
                             v_head_dim=16, **kw)


def test_mla_forward_backward_and_trains():
    init_single()
    model_parallel_seed(11)
    m = GPTModel(_cfg(q_lora_rank=48))
    tokens = torch.randint(0, 128, (2, 32))
    labels = torch.randint(0, 128, (2, 32))
    opt = torch.optim.AdamW(m.parameters(), lr=1e-3)
    losses = []
    for _ in range(6):
        loss = m(tokens, labels=labels).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0]
    attn = m.decoder.layers[0].self_attention
    assert attn.linear_kv_down.weight.shape == (32 + 16, 64)


def test_mla_is_causal():
    init_single()
    model_parallel_seed(11)
    m = GPTModel(_cfg()).eval()
    t1 = torch.randint(0, 128, (1, 16))
    t2 = t1.clone()
    t2[0, -1] = (t2[0, -1] + 1) % 128
    with torch.no_grad():
        l1, l2 = m(t1), m(t2)
    torch.testing.assert_close(l1[:-1], l2[:-1], rtol=1e-4, atol=1e-5)
    assert not torch.allclose(l1[-1], l2[-1])


def test_mla_nonflash_dims_path():
    # dv != dqk exercises the torch core-attention fallback
    init_single()
    model_parallel_seed(11)
    cfg = TransformerConfig(num_layers=1, hidden_size=64, num_attention_heads=4,
                            num_query_groups=4, ffn_hidden_size=128, vocab_size=128,
                            max_position_embeddings=64, multi_latent_attention=True,
                            kv_lora_rank=32, qk_nope_head_dim=24, qk_rope_head_dim=8,
                            v_head_dim=16)
    m = GPTModel(cfg)
    tokens = torch.randint(0, 128, (2, 16))
    loss = m(tokens, labels=tokens)
    loss.sum().backward()
    attn = m.decoder.layers[0].self_attention
    assert attn.dqk == 32 and attn.dv == 16
    assert attn.linear_kv_up.weight.grad is not None


def test_mla_absorbed_inference_matches_oracle():
    """Static-engine generation through the absorbed latent-cache path must
    equal no-cache greedy generation (full forward each step)."""
    from megatron_amd.inference import SamplingParams, StaticInferenceEngine

    init_single()
    model_parallel_seed(21)
    m = GPTModel(_cfg(q_lora_rank=48)).eval()
    prompts = [[3, 7, 11, 2, 9], [5, 1]]
    params = SamplingParams(max_tokens=8, greedy=True, stop_on_eod=False)
    eng = StaticInferenceEngine(m, max_batch=4, max_seq=64)
    res = eng.generate(prompts, params)

    for p, r in zip(prompts, res):
        toks = list(p)
        expect = []
        for _ in range(8):
            with torch.no_grad():
                logits = m(torch.tensor([toks]))
            tok = int(logits[-1, 0].float().argmax())
            expect.append(tok)
            toks.append(tok)
        assert r.output_tokens == expect, (r.output_tokens, expect)


def test_mla_absorbed_prefill_matches_training_forward():
    """The absorbed path's prefill logits equal the training-path forward."""
    from megatron_amd.inference.contexts import StaticInferenceContext

    init_single()
    model_parallel_seed(5)
    m = GPTModel(_cfg()).eval()
    toks = torch.randint(0, 128, (2, 12))
    with torch.no_grad():
        train_logits = m(toks)
        ctx = StaticInferenceContext(num_layers=2, max_batch=2, max_seq=32,
                                     num_kv_heads=4, head_dim=16,
                                     dtype=torch.float32, device="cpu")
        ctx.reset(2)
        inf_logits = m(toks, inference_context=ctx)
    assert torch.allclose(train_logits, inf_logits, atol=1e-4), \
        (train_logits - inf_logits).abs().max()


def test_mla_dynamic_engine_matches_oracle():
    """MLA through the dynamic (paged latent pool) engine: continuous
    batching + chunked prefill must equal no-cache greedy generation."""
    from megatron_amd.inference import DynamicInferenceEngine, SamplingParams

    init_single()
    model_parallel_seed(31)
    m = GPTModel(_cfg()).eval()
    prompts = [[3, 7, 11, 2, 9], [5, 1], [8, 8, 4, 2, 1, 0, 9]]
    params = SamplingParams(max_tokens=6, greedy=True, stop_on_eod=False)
    eng = DynamicInferenceEngine(m, num_blocks=16, block_size=4, max_prefill_tokens=4)
    res = eng.generate(prompts, params)
    for p, r in zip(prompts, res):
        toks = list(p)
        expect = []
        for _ in range(6):
            with torch.no_grad():
                logits = m(torch.tensor([toks]))
            tok = int(logits[-1, 0].float().argmax())
            expect.append(tok)
            toks.append(tok)
        assert r.output_tokens == expect, (r.output_tokens, expect)


def test_mla_preemption_equivalence():
    """MLA + tight pool: preemption must swap the latent pool too (greedy
    outputs equal the big-pool run)."""
    from megatron_amd.inference import DynamicInferenceEngine, SamplingParams

    init_single()
    model_parallel_seed(41)
    m = GPTModel(_cfg()).eval()
    prompts = [[3, 7, 11, 2, 9], [5, 1], [8, 8, 4, 2], [12, 13, 14]]
    params = SamplingParams(max_tokens=8, greedy=True, stop_on_eod=False)
    big = DynamicInferenceEngine(m, num_blocks=64, block_size=4)
    expected = big.generate(prompts, params)
    small = DynamicInferenceEngine(m, num_blocks=9, block_size=4)
    got = small.generate(prompts, params)
    for a, b in zip(expected, got):
        assert a.output_tokens == b.output_tokens
    assert small.offloader._ids.__reduce__()[1][0] > 0  # preemption occurred


def test_mla_disaggregated_matches_single():
    from megatron_amd.inference import DynamicInferenceEngine, SamplingParams
    from megatron_amd.inference.disaggregation import disaggregated_generate

    init_single()
    model_parallel_seed(51)
    m = GPTModel(_cfg()).eval()
    prompts = [[3, 7, 11, 2, 9], [5, 1]]
    params = SamplingParams(max_tokens=6, greedy=True, stop_on_eod=False)
    single = DynamicInferenceEngine(m, num_blocks=32, block_size=4)
    expected = single.generate(prompts, params)
    pre = DynamicInferenceEngine(m, num_blocks=16, block_size=4)
    dec = DynamicInferenceEngine(m, num_blocks=32, block_size=4)
    got = disaggregated_generate(pre, dec, prompts, params)
    for a, b in zip(expected, got):
        assert a.output_tokens == b.output_tokens


def test_mla_packed_sequences_match_separate():
    """Two documents packed into one [T, 1] stream attend block-diagonally
    with per-document RoPE restart: each document's output equals running it
    alone (closes the round-1 'MLA packed: round 2' limitation)."""
    from megatron_amd.transformer.packed_seq import PackedSeqParams

    from megatron_amd.transformer.multi_latent_attention import MLASelfAttention

    init_single()
    torch.manual_seed(0)
    cfg = _cfg()
    att = MLASelfAttention(cfg, layer_number=1).eval()
    l1, l2 = 12, 8
    h1 = torch.randn(l1, 1, cfg.hidden_size)
    h2 = torch.randn(l2, 1, cfg.hidden_size)
    with torch.no_grad():
        o1 = att(h1)
        o2 = att(h2)
        packed = torch.cat([h1, h2], dim=0)
        psp = PackedSeqParams.from_lengths([l1, l2])
        op = att(packed, packed_seq_params=psp)
    torch.testing.assert_close(op[:l1], o1, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(op[l1:], o2, rtol=1e-4, atol=1e-5)
