"""Multi-token prediction tests (reference transformer/multi_token_prediction.py)."""

import torch

from megatron_amd.config import TransformerConfig
from megatron_amd.models.gpt import GPTModel
from megatron_amd.parallel.random import model_parallel_seed
from tests.utils import init_single

KW = dict(num_layers=2, hidden_size=64, num_attention_heads=4, num_query_groups=2,
          ffn_hidden_size=128, vocab_size=128, max_position_embeddings=64)


def test_mtp_zero_matches_baseline():
    init_single()
    model_parallel_seed(5)
    base = GPTModel(TransformerConfig(**KW))
    tokens = torch.randint(0, 128, (2, 32))
    labels = torch.randint(0, 128, (2, 32))
    loss0 = base(tokens, labels=labels)
    init_single()
    model_parallel_seed(5)
    same = GPTModel(TransformerConfig(**KW, mtp_num_layers=0))
    torch.testing.assert_close(loss0, same(tokens, labels=labels))


def test_mtp_adds_scaled_loss_and_trains():
    init_single()
    model_parallel_seed(5)
    cfg = TransformerConfig(**KW, mtp_num_layers=2)
    model = GPTModel(cfg)
    tokens = torch.randint(0, 128, (2, 32))
    labels = torch.randint(0, 128, (2, 32))
    loss = model(tokens, labels=labels)
    assert loss.shape == (32, 2)
    loss.sum().backward()
    # MTP head params get gradients
    g = model.mtp.heads[0].proj.weight.grad
    assert g is not None and torch.isfinite(g).all()
    # loss exceeds the main-only loss (extra positive terms)
    with torch.no_grad():
        model2 = model
        model2.mtp = None
        main_only = model2(tokens, labels=labels)
    assert float(loss.sum()) > float(main_only.sum())
