"""BERT model family tests (reference models/bert + masked dataset)."""

import numpy as np
import torch

from megatron_amd.config import TransformerConfig
from megatron_amd.models.bert import BertModel
from megatron_amd.parallel.random import model_parallel_seed
from tests.utils import init_single


def _cfg(**kw):
    return TransformerConfig(num_layers=2, hidden_size=64, num_attention_heads=4,
                             num_query_groups=4, ffn_hidden_size=128, vocab_size=128,
                             max_position_embeddings=64, activation="gelu", **kw)


def test_bert_forward_backward():
    init_single()
    model_parallel_seed(3)
    m = BertModel(_cfg(), add_binary_head=True)
    tokens = torch.randint(0, 128, (2, 32))
    labels = torch.randint(0, 128, (2, 32))
    loss_mask = (torch.rand(2, 32) < 0.15).float()
    loss = m(tokens, labels=labels, loss_mask=loss_mask)
    assert loss.shape == (32, 2)
    # unmasked positions contribute zero loss
    assert float(loss.transpose(0, 1)[loss_mask == 0].abs().sum()) == 0.0
    loss.sum().backward()
    assert m.encoder.layers[0].self_attention.linear_qkv.weight.grad is not None


def test_bert_attention_is_bidirectional():
    init_single()
    model_parallel_seed(3)
    m = BertModel(_cfg())
    t1 = torch.randint(0, 128, (1, 16))
    t2 = t1.clone()
    t2[0, -1] = (t2[0, -1] + 1) % 128
    with torch.no_grad():
        l1 = m(t1)
        l2 = m(t2)
    # changing the LAST token changes the FIRST position's logits (no causality)
    assert not torch.allclose(l1[0], l2[0])


def test_masked_dataset():
    from megatron_amd.datasets.bert_dataset import BertMaskedDataset

    class Base(torch.utils.data.Dataset):
        def __len__(self):
            return 4

        def __getitem__(self, i):
            return {"tokens": torch.arange(1, 33)}

    ds = BertMaskedDataset(Base(), vocab_size=100, mask_id=99, seed=7)
    s = ds[0]
    n = int(s["loss_mask"].sum())
    assert n == 4  # 15% of 32
    masked_pos = s["loss_mask"].bool()
    assert (s["labels"][masked_pos] == torch.arange(1, 33)[masked_pos]).all()
    assert (s["labels"][~masked_pos] == 0).all()
    assert (s["tokens"][~masked_pos] == torch.arange(1, 33)[~masked_pos]).all()
    # deterministic per (seed, idx)
    s2 = ds[0]
    assert (s2["tokens"] == s["tokens"]).all()


def test_pretrain_bert_end_to_end():
    import pretrain_bert
    from megatron_amd.training.pretrain import pretrain

    it = pretrain(pretrain_bert.model_provider, [
        "--num-layers", "1", "--hidden-size", "32", "--num-attention-heads", "2",
        "--num-query-groups", "2", "--ffn-hidden-size", "64", "--seq-length", "32",
        "--micro-batch-size", "2", "--global-batch-size", "2", "--vocab-size", "64",
        "--mock-data", "--train-iters", "2", "--log-interval", "0",
        "--activation", "gelu", "--position-embedding-type", "learned",
    ], forward_step_builder=pretrain_bert.forward_step_builder)
    assert it == 2


def test_bert_padding_mask_isolates_pad_tokens():
    """A padded short sample must produce the same valid-position logits as
    the unpadded sample (the key-padding mask path, reference masked-
    softmax behavior)."""
    init_single()
    cfg = _cfg()
    torch.manual_seed(0)
    m = BertModel(cfg)
    m.eval()
    toks = torch.randint(0, cfg.vocab_size, (1, 12))
    with torch.no_grad():
        full = m(toks, attention_mask=torch.ones(1, 12, dtype=torch.bool))
        padded_toks = torch.cat([toks, torch.zeros(1, 4, dtype=torch.long)], dim=1)
        mask = torch.cat([torch.ones(1, 12, dtype=torch.bool),
                          torch.zeros(1, 4, dtype=torch.bool)], dim=1)
        padded = m(padded_toks, attention_mask=mask)
    torch.testing.assert_close(padded[:12], full, rtol=1e-4, atol=1e-5)


def test_bert_binary_head_and_pooler():
    init_single()
    cfg = _cfg()
    torch.manual_seed(0)
    m = BertModel(cfg, add_binary_head=True)
    toks = torch.randint(0, cfg.vocab_size, (2, 8))
    labels = torch.randint(0, cfg.vocab_size, (2, 8))
    loss = m(toks, labels=labels)
    binary = m.binary_logits
    assert binary.shape == (2, 2)
    # NSP + MLM joint training step
    nsp_tgt = torch.tensor([0, 1])
    total = loss.sum() + torch.nn.functional.cross_entropy(binary, nsp_tgt)
    total.backward()
    assert m.pooler.weight.grad is not None
    assert m.binary_head.weight.grad is not None


def test_bert_dataset_sentence_pairs_nsp():
    """sentence_pairs mode: tokentype marks segment B, is_next=0 iff the
    second half was swapped in from another sample."""
    import numpy as np

    from megatron_amd.datasets.bert_dataset import BertMaskedDataset

    class Base:
        def __len__(self):
            return 8

        def __getitem__(self, i):
            return {"tokens": torch.full((16,), i + 1, dtype=torch.long)}

    ds = BertMaskedDataset(Base(), vocab_size=64, mask_id=63, seed=3,
                           sentence_pairs=True)
    seen = {0: 0, 1: 0}
    for i in range(8):
        s = ds[i]
        assert torch.all(s["tokentype_ids"][:8] == 0)
        assert torch.all(s["tokentype_ids"][8:] == 1)
        label = int(s["is_next"])
        seen[label] += 1
        # unmasked positions of segment B reveal the source sample
        tail = s["tokens"][8:]
        clean = tail[(tail != 63) & (tail < 9)]
        if label == 1 and clean.numel():
            assert torch.all(clean == i + 1)
        if label == 0 and clean.numel():
            assert torch.all(clean != i + 1)
    assert seen[0] > 0 and seen[1] > 0
