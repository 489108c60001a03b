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
