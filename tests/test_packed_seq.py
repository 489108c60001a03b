"""Packed (THD) sequences: varlen attention, packing, model equivalence."""

import torch

from megatron_amd.config import TransformerConfig
from megatron_amd.models.gpt import GPTModel
from megatron_amd.ops import reference as ref
from megatron_amd.transformer.packed_seq import PackedSeqParams, pack_sequences

from tests.utils import assert_close, init_single


def test_attention_varlen_matches_per_segment():
    torch.manual_seed(0)
    lens = [5, 3, 7]
    t, hq, hkv, d = sum(lens), 4, 2, 8
    q = torch.randn(t, 1, hq, d)
    k = torch.randn(t, 1, hkv, d)
    v = torch.randn(t, 1, hkv, d)
    p = PackedSeqParams.from_lengths(lens)
    out = ref.attention_varlen(q, k, v, p.cu_seqlens, causal=True)
    # each segment independently must equal plain causal attention on it
    for i in range(len(lens)):
        a, b = int(p.cu_seqlens[i]), int(p.cu_seqlens[i + 1])
        seg = ref.attention(q[a:b], k[a:b], v[a:b], causal=True)
        assert_close(out[a:b], seg, rtol=1e-5, atol=1e-6)


def test_packed_params_positions_and_segments():
    p = PackedSeqParams.from_lengths([3, 2, 4])
    assert p.total_tokens == 9
    assert p.segment_ids().tolist() == [0, 0, 0, 1, 1, 2, 2, 2, 2]
    assert p.positions().tolist() == [0, 1, 2, 0, 1, 0, 1, 2, 3]


def test_pack_sequences_first_fit():
    docs = [torch.arange(1, 6), torch.arange(10, 13), torch.arange(20, 24), torch.arange(30, 32)]
    rows = pack_sequences(docs, seq_length=8, pad_id=0)
    total_tokens = sum(d.numel() for d in docs)
    packed_tokens = sum(int((r["cu_seqlens"][-2] if r["cu_seqlens"][-1] == 8 and r["loss_mask"][-1] == 0
                             else r["cu_seqlens"][-1])) for r in rows)
    # all documents present exactly once
    flat = torch.cat([r["tokens"][r["tokens"] != 0] for r in rows]).sort().values
    expect = torch.cat(docs).sort().values
    assert torch.equal(flat, expect)
    for r in rows:
        # labels are next-token within the pack, boundary tokens masked
        cu = r["cu_seqlens"]
        for j in range(len(cu) - 1):
            end = int(cu[j + 1]) - 1
            if end < 8:
                assert r["loss_mask"][end] == 0.0


def test_gpt_packed_forward_matches_separate():
    """A packed forward of two documents must equal the two separate
    forwards (block-diagonal attention + per-document RoPE restart)."""
    init_single()
    cfg = TransformerConfig(
        num_layers=2, hidden_size=32, num_attention_heads=4, num_query_groups=2,
        vocab_size=64, ffn_hidden_size=48, gradient_accumulation_fusion=False,
        max_position_embeddings=64)
    torch.manual_seed(4)
    model = GPTModel(cfg).eval()
    d1 = torch.randint(0, 64, (1, 7))
    d2 = torch.randint(0, 64, (1, 5))
    packed = torch.cat([d1, d2], dim=1)  # [1, 12]
    p = PackedSeqParams.from_lengths([7, 5])
    with torch.no_grad():
        out_p = model(packed, position_ids=None, attention_mask=None, packed_seq_params=p)
        out_1 = model(d1, position_ids=None, attention_mask=None)
        out_2 = model(d2, position_ids=None, attention_mask=None)
    assert out_p.shape == (12, 1, 64)
    assert_close(out_p[:7], out_1, rtol=1e-5, atol=1e-5)
    assert_close(out_p[7:], out_2, rtol=1e-5, atol=1e-5)


def test_pack_sequences_fuzz_conservation():
    """Property: packing never loses or duplicates tokens and every row's
    cu_seqlens is consistent, across random document mixes."""
    import random

    rng = random.Random(0)
    for trial in range(10):
        n_docs = rng.randint(1, 12)
        docs = [torch.randint(1, 999, (rng.randint(1, 20),)) for _ in range(n_docs)]
        L = rng.choice([16, 24, 32])
        rows = pack_sequences(docs, seq_length=L, pad_id=0)
        flat = torch.cat([r["tokens"][r["tokens"] != 0] for r in rows]).sort().values
        expect = torch.cat([d[:L] for d in docs]).sort().values
        assert torch.equal(flat, expect), trial
        for r in rows:
            cu = r["cu_seqlens"]
            assert cu[0] == 0 and int(cu[-1]) <= L
            assert (cu[1:] > cu[:-1]).all() or int(cu[-1]) == L
            assert r["tokens"].shape == (L,)


def test_eod_boundaries_fuzz_roundtrip():
    """Fuzz: random EOD layouts -> cu boundaries are sorted, end at seq_len,
    split only after EODs, and PackedSeqParams positions restart per doc."""
    import numpy as np

    from megatron_amd.datasets.gpt_dataset import eod_boundaries
    from megatron_amd.transformer.packed_seq import PackedSeqParams

    rng = np.random.RandomState(0)
    for trial in range(50):
        s = int(rng.randint(4, 64))
        eod = 0
        toks = torch.from_numpy(rng.randint(0, 5, size=s)).long()
        cu = eod_boundaries(toks, eod, s, max_docs=16)
        vals = [int(x) for x in cu if x > 0]
        assert vals == sorted(set(vals)) and vals[-1] == s
        for c in vals[:-1]:
            assert int(toks[c - 1]) == eod
        lengths = [b - a for a, b in zip([0] + vals, vals)]
        psp = PackedSeqParams.from_lengths(lengths)
        pos = psp.positions()
        off = 0
        for L in lengths:
            assert pos[off] == 0 and int(pos[off + L - 1]) == L - 1
            off += L
