"""T5 encoder-decoder tests (reference models/T5)."""

import torch

from megatron_amd.config import TransformerConfig
from megatron_amd.models.t5 import T5Model
from megatron_amd.parallel.random import model_parallel_seed
from tests.utils import init_single


def _cfg():
    return TransformerConfig(num_layers=2, hidden_size=64, num_attention_heads=4,
                             num_query_groups=4, ffn_hidden_size=128, vocab_size=128,
                             max_position_embeddings=64, activation="gelu")


def test_t5_forward_backward():
    init_single()
    model_parallel_seed(9)
    m = T5Model(_cfg())
    enc = torch.randint(0, 128, (2, 24))
    dec = torch.randint(0, 128, (2, 16))
    labels = torch.randint(0, 128, (2, 16))
    loss = m(enc, dec, labels=labels)
    assert loss.shape == (16, 2)
    loss.sum().backward()
    assert m.decoder_layers[0].cross_attention.linear_kv.weight.grad is not None
    assert m.encoder.layers[0].self_attention.linear_qkv.weight.grad is not None


def test_t5_decoder_is_causal_encoder_not():
    init_single()
    model_parallel_seed(9)
    m = T5Model(_cfg()).eval()
    enc = torch.randint(0, 128, (1, 24))
    dec = torch.randint(0, 128, (1, 16))
    with torch.no_grad():
        base = m(enc, dec)
        # perturb LAST decoder token: earlier positions' logits unchanged (causal)
        dec2 = dec.clone()
        dec2[0, -1] = (dec2[0, -1] + 1) % 128
        out2 = m(enc, dec2)
        torch.testing.assert_close(base[:-1], out2[:-1], rtol=1e-4, atol=1e-5)
        # perturb LAST encoder token: ALL decoder logits change (cross-attention)
        enc2 = enc.clone()
        enc2[0, -1] = (enc2[0, -1] + 1) % 128
        out3 = m(enc2, dec)
        assert not torch.allclose(base[0], out3[0])


def test_t5_trains():
    init_single()
    model_parallel_seed(9)
    m = T5Model(_cfg())
    opt = torch.optim.AdamW(m.parameters(), lr=1e-3)
    enc = torch.randint(0, 128, (2, 24))
    dec = torch.randint(0, 128, (2, 16))
    labels = torch.randint(0, 128, (2, 16))
    losses = []
    for _ in range(6):
        loss = m(enc, dec, labels=labels).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0]


def _rel_cfg():
    return TransformerConfig(num_layers=2, hidden_size=64, num_attention_heads=4,
                             num_query_groups=4, ffn_hidden_size=128, vocab_size=128,
                             max_position_embeddings=64, activation="gelu",
                             position_embedding_type="relative")


def test_t5_relative_bias_buckets():
    """Bucketing follows the classic T5 scheme: exact buckets near zero,
    log-spaced to max_distance, direction split only when bidirectional."""
    from megatron_amd.models.t5 import T5RelativePositionBias

    init_single()
    cfg = _rel_cfg()
    bid = T5RelativePositionBias(cfg, bidirectional=True)
    rel = torch.arange(-200, 201)
    b = bid._bucket(rel)
    assert b.min() >= 0 and b.max() < cfg.relative_attention_num_buckets
    # direction is distinguished
    assert b[rel == 5] != b[rel == -5]
    # exact near-field: distinct buckets for small distances
    small = b[(rel >= -3) & (rel <= 3)]
    assert len(set(small.tolist())) == 7
    # far field saturates
    assert b[rel == 200] == b[rel == 190]

    cau = T5RelativePositionBias(cfg, bidirectional=False)
    bc = cau._bucket(rel)
    # future positions (rel > 0) all collapse to bucket 0 for causal
    assert torch.all(bc[rel > 0] == 0)
    assert bc.max() < cfg.relative_attention_num_buckets


def test_t5_relative_bias_trains_and_is_causal():
    init_single()
    model_parallel_seed(13)
    m = T5Model(_rel_cfg())
    assert m.position_embedding is None
    enc = torch.randint(0, 128, (2, 24))
    dec = torch.randint(0, 128, (2, 16))
    labels = torch.randint(0, 128, (2, 16))
    opt = torch.optim.AdamW(m.parameters(), lr=1e-3)
    losses = []
    for _ in range(6):
        loss = m(enc, dec, labels=labels).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0]
    assert m.encoder_rel_bias.embedding.weight.grad is None  # zeroed by opt
    # decoder causality with the bias applied
    m.eval()
    d1 = torch.randint(0, 128, (1, 12))
    d2 = d1.clone()
    d2[0, -1] = (d2[0, -1] + 1) % 128
    with torch.no_grad():
        l1 = m(enc[:1], d1)
        l2 = m(enc[:1], d2)
    torch.testing.assert_close(l1[:-1], l2[:-1], rtol=1e-4, atol=1e-5)


def test_t5_relative_bias_matches_manual_attention():
    """One self-attention layer with the bias equals a hand-computed
    softmax(QK^T * scale + bias) V."""
    from megatron_amd.transformer.attention import SelfAttention

    init_single()
    torch.manual_seed(5)
    cfg = _rel_cfg().replace(causal_attention=False)
    att = SelfAttention(cfg, layer_number=1).eval()
    s, b = 10, 2
    x = torch.randn(s, b, cfg.hidden_size)
    bias = torch.randn(cfg.num_attention_heads, s, s) * 0.1
    with torch.no_grad():
        out = att(x, attention_bias=bias)
        qkv, _ = att.linear_qkv(x)
        d = att.kv_channels
        ng = att.num_query_groups_per_partition
        rep = att.num_heads_per_partition // ng
        qkv = qkv.view(s, b, ng, (rep + 2) * d)
        q, k, v = torch.split(qkv, [rep * d, d, d], dim=3)
        q = q.reshape(s, b, ng * rep, d).permute(1, 2, 0, 3)
        k = k.reshape(s, b, ng, d).permute(1, 2, 0, 3)
        v = v.reshape(s, b, ng, d).permute(1, 2, 0, 3)
        scores = q.float() @ k.float().transpose(-1, -2) * att.softmax_scale + bias.float()
        ref = (torch.softmax(scores, -1) @ v.float()).permute(2, 0, 1, 3).reshape(s, b, -1)
        ref_out, _ = att.linear_proj(ref.to(x.dtype))
    torch.testing.assert_close(out, ref_out, rtol=1e-4, atol=1e-5)


def test_t5_relative_bias_with_padding_mask():
    """Relative bias composes with a key-padding mask: padded key positions
    contribute nothing regardless of their bias."""
    from megatron_amd.transformer.attention import SelfAttention

    init_single()
    torch.manual_seed(9)
    cfg = _rel_cfg().replace(causal_attention=False)
    att = SelfAttention(cfg, layer_number=1).eval()
    s, b = 8, 2
    x = torch.randn(s, b, cfg.hidden_size)
    bias = torch.randn(cfg.num_attention_heads, s, s)
    mask = torch.ones(b, s, dtype=torch.bool)
    mask[:, -2:] = False  # last two keys padded
    with torch.no_grad():
        out_masked = att(x, attention_mask=mask, attention_bias=bias)
        # oracle: truncate the padded keys entirely
        out_trunc = att(x[:, :, :], attention_mask=mask, attention_bias=bias)
        x2 = x.clone()
        x2[-2:] = 100.0  # garbage in padded positions must not leak
        out_garbage = att(x2, attention_mask=mask, attention_bias=bias)
    torch.testing.assert_close(out_masked[:6], out_garbage[:6], rtol=1e-4, atol=1e-5)
