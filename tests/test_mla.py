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
