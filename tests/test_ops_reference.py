"""Reference-op sanity (the same oracles the GPU numerics tests compare against)."""

import math

import torch
import torch.nn.functional as F

from megatron_amd.ops import reference as ref
from tests.utils import assert_close


def test_rms_norm_matches_manual():
    x = torch.randn(4, 8, 64)
    w = torch.randn(64)
    out = ref.rms_norm(x, w, 1e-5)
    manual = x / torch.sqrt(x.pow(2).mean(-1, keepdim=True) + 1e-5) * w
    assert_close(out, manual, rtol=1e-5, atol=1e-5)


def test_swiglu():
    x = torch.randn(10, 32)
    out = ref.swiglu(x)
    x1, x2 = x.chunk(2, -1)
    assert_close(out, F.silu(x1) * x2, rtol=1e-5, atol=1e-5)


def test_rope_preserves_norm_and_zero_rotation():
    s, b, h, d = 12, 2, 4, 16
    x = torch.randn(s, b, h, d)
    freqs = ref.rope_freqs(s, d, base=10000.0)
    out = ref.rope_apply(x, freqs)
    assert out.shape == x.shape
    # norms preserved per (pair) rotation
    assert_close(out.norm(dim=-1), x.norm(dim=-1), rtol=1e-4, atol=1e-4)
    # position 0 has zero angle -> identity
    assert_close(out[0], x[0], rtol=1e-6, atol=1e-6)


def test_attention_vs_sdpa_causal():
    torch.manual_seed(0)
    s, b, hq, hkv, d = 16, 2, 4, 2, 8
    q = torch.randn(s, b, hq, d)
    k = torch.randn(s, b, hkv, d)
    v = torch.randn(s, b, hkv, d)
    out = ref.attention(q, k, v, causal=True)
    qs = q.permute(1, 2, 0, 3)
    ks = k.permute(1, 2, 0, 3).repeat_interleave(2, dim=1)
    vs = v.permute(1, 2, 0, 3).repeat_interleave(2, dim=1)
    sdpa = F.scaled_dot_product_attention(qs, ks, vs, is_causal=True)
    assert_close(out, sdpa.permute(2, 0, 1, 3), rtol=1e-4, atol=1e-4)


def test_attention_sliding_window():
    torch.manual_seed(0)
    s, b, h, d = 16, 1, 2, 8
    q = torch.randn(s, b, h, d)
    k = torch.randn(s, b, h, d)
    v = torch.randn(s, b, h, d)
    out = ref.attention(q, k, v, causal=True, window=4)
    # manual
    scale = 1 / math.sqrt(d)
    scores = torch.einsum("sbhd,tbhd->bhst", q, k) * scale
    mask = torch.ones(s, s, dtype=torch.bool).tril_() & torch.ones(s, s, dtype=torch.bool).triu_(-3)
    scores = scores.masked_fill(~mask, float("-inf"))
    manual = torch.einsum("bhst,tbhd->sbhd", torch.softmax(scores, -1), v)
    assert_close(out, manual, rtol=1e-4, atol=1e-4)


def test_llama3_rope_scaling_matches_hf_formula():
    import math

    import torch

    from megatron_amd.ops.reference import apply_llama3_rope_scaling, rope_freqs

    dim, base = 128, 500000.0
    inv = 1.0 / (base ** (torch.arange(0, dim, 2, dtype=torch.float32) / dim))
    factor, lo, hi, orig = 8.0, 1.0, 4.0, 8192
    got = apply_llama3_rope_scaling(inv, factor, lo, hi, orig)
    # HF transformers _compute_llama3_parameters reference computation
    low_wl = orig / lo
    high_wl = orig / hi
    expect = []
    for f in inv.tolist():
        wl = 2 * math.pi / f
        if wl < high_wl:
            expect.append(f)
        elif wl > low_wl:
            expect.append(f / factor)
        else:
            smooth = (orig / wl - lo) / (hi - lo)
            expect.append((1 - smooth) * f / factor + smooth * f)
    assert torch.allclose(got, torch.tensor(expect), rtol=1e-6, atol=0)
    # high-freq (small wavelength) channels unchanged; lowest-freq divided by 8
    assert got[0] == inv[0]
    assert abs(got[-1] - inv[-1] / factor) < 1e-9
    # end-to-end: freqs table differs once scaling is on
    f0 = rope_freqs(16, dim, base=base)
    f1 = rope_freqs(16, dim, base=base, rope_scaling={"type": "llama3"})
    assert not torch.allclose(f0, f1)


def test_mrope_reduces_to_rope_for_text():
    import torch

    from megatron_amd.ops.reference import mrope_freqs, rope_freqs

    s, dim = 12, 128
    # text-only: all three position rows identical -> plain RoPE
    pos = torch.arange(s).unsqueeze(0).repeat(3, 1)
    got = mrope_freqs(pos, dim, base=10000.0, mrope_section=(16, 24, 24))
    expect = rope_freqs(s, dim, base=10000.0)
    assert torch.allclose(got, expect, atol=1e-6)


def test_mrope_sections_use_their_position_rows():
    import torch

    from megatron_amd.ops.reference import mrope_freqs

    dim = 64  # n_half = 32
    sec = (8, 12, 12)
    pos = torch.stack([torch.zeros(4), torch.ones(4) * 2, torch.ones(4) * 5]).long()
    f = mrope_freqs(pos, dim, mrope_section=sec)
    # temporal section: position 0 -> zero angles
    assert torch.all(f[:, :8] == 0)
    # height/width sections scale with their own positions
    assert torch.all(f[:, 8:20] > 0) and torch.all(f[:, 20:] > 0)
    f2 = mrope_freqs(pos * 2, dim, mrope_section=sec)
    assert torch.allclose(f2[:, 8:], f[:, 8:] * 2)


def test_ce_native_tp_merge_math():
    """CPU check of the fused-CE TP merge algebra (cross_entropy.py native
    path): per-shard online (max, sumexp, raw-target-logit) stats merged via
    MAX + rebased SUM must reproduce full-softmax cross entropy."""
    import torch
    import torch.nn.functional as F

    torch.manual_seed(0)
    T, V = 64, 256
    logits = torch.randn(T, V) * 4
    target = torch.randint(0, V, (T,))
    ms, ss, ps = [], [], []
    for r, lg in enumerate(logits.chunk(2, dim=-1)):
        vp = lg.shape[-1]
        m_loc = lg.max(-1).values
        s_loc = torch.exp(lg - m_loc[:, None]).sum(-1)
        mt = target - r * vp
        in_sh = (mt >= 0) & (mt < vp)
        pred = torch.where(
            in_sh, lg.gather(-1, mt.clamp(0, vp - 1)[:, None]).squeeze(-1), torch.zeros(T)
        )
        ms.append(m_loc); ss.append(s_loc); ps.append(pred)
    m = torch.maximum(ms[0], ms[1])                       # all-reduce MAX
    s = ss[0] * torch.exp(ms[0] - m) + ss[1] * torch.exp(ms[1] - m)  # rebased SUM
    pred_raw = ps[0] + ps[1]                              # SUM
    loss = torch.log(s) - (pred_raw - m)
    ref = F.cross_entropy(logits, target, reduction="none")
    assert torch.allclose(loss, ref, rtol=1e-5, atol=1e-5)


def test_yarn_rope_scaling():
    """YaRN NTK-by-parts: high-frequency dims keep their rotation rate,
    low-frequency dims interpolate by 1/factor, the band ramps; the
    attention temperature squares into softmax_scale."""
    import math

    from megatron_amd.ops.reference import rope_freqs, yarn_mscale

    base = rope_freqs(8, 64, base=10000.0)
    yarn = rope_freqs(8, 64, base=10000.0,
                      rope_scaling={"type": "yarn", "factor": 8.0,
                                    "original_max_position_embeddings": 512})
    # angles at position 1 = inv_freq directly
    f0, f1 = base[1], yarn[1]
    assert torch.allclose(f1[0], f0[0])            # fastest dim untouched
    assert torch.allclose(f1[-1], f0[-1] / 8.0)    # slowest dim fully scaled
    ratio = (f1 / f0)
    assert bool((ratio[1:] <= ratio[:-1] + 1e-6).all())  # monotone ramp
    assert abs(yarn_mscale(8.0) - (0.1 * math.log(8.0) + 1.0)) < 1e-9

    from megatron_amd.config import TransformerConfig
    from megatron_amd.transformer.attention import SelfAttention
    from tests.utils import init_single

    init_single()
    cfg = TransformerConfig(num_layers=1, hidden_size=64, num_attention_heads=4,
                            num_query_groups=4, vocab_size=64, ffn_hidden_size=64,
                            rope_scaling={"type": "yarn", "factor": 8.0})
    att = SelfAttention(cfg, layer_number=1)
    expect = (1.0 / math.sqrt(att.kv_channels)) * yarn_mscale(8.0) ** 2
    assert abs(att.softmax_scale - expect) < 1e-9
