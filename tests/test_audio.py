"""Audio-language model (reference models/audio): frame stacking, splice at
the audio token, trains end-to-end."""

import torch

from megatron_amd.config import TransformerConfig
from megatron_amd.models.audio import AudioFeatureProjector, AudioLanguageModel
from megatron_amd.parallel.random import model_parallel_seed
from tests.utils import init_single


def _cfg():
    return TransformerConfig(num_layers=2, hidden_size=64, num_attention_heads=4,
                             num_query_groups=4, ffn_hidden_size=128, vocab_size=128,
                             max_position_embeddings=128)


def test_audio_projector_stacks_frames():
    p = AudioFeatureProjector(feat_dim=8, language_hidden=32, stack_factor=4)
    out = p(torch.randn(2, 10, 8))  # 10 frames -> ceil(10/4) = 3 steps
    assert out.shape == (2, 3, 32)
    out = p(torch.randn(2, 12, 8))
    assert out.shape == (2, 3, 32)


def test_audio_language_model_trains():
    init_single()
    model_parallel_seed(15)
    m = AudioLanguageModel(_cfg(), feat_dim=8, stack_factor=4)
    b, s, t = 2, 12, 16
    ids = torch.randint(1, 128, (b, s))
    ids[:, 3] = m.audio_token_index
    labels = torch.randint(1, 128, (b, s))
    feats = torch.randn(b, t, 8)
    opt = torch.optim.AdamW([p for p in m.parameters() if p.requires_grad], lr=1e-3)
    losses = []
    for _ in range(6):
        loss = m(feats, ids, labels=labels)
        assert loss.shape == (s - 1 + 4, b)
        loss.mean().backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss.mean()))
    assert losses[-1] < losses[0]
    assert m.audio_projector.fc1.weight.grad is None  # zeroed by opt


def test_audio_model_text_only_path():
    init_single()
    model_parallel_seed(15)
    m = AudioLanguageModel(_cfg(), feat_dim=8)
    ids = torch.randint(1, 128, (2, 10))
    loss = m(None, ids, labels=ids)
    assert loss.shape == (10, 2)


def test_pretrain_audio_runs():
    import pretrain_audio as A
    from megatron_amd.training.pretrain import pretrain

    it = pretrain(A.model_provider, [
        "--num-layers", "2", "--hidden-size", "64", "--num-attention-heads", "4",
        "--num-query-groups", "2", "--ffn-hidden-size", "128", "--seq-length", "32",
        "--micro-batch-size", "2", "--global-batch-size", "4", "--vocab-size", "128",
        "--mock-data", "--train-iters", "2", "--log-interval", "0", "--seed", "11",
    ], forward_step_builder=A.forward_step_builder)
    assert it == 2
