"""LLaVA multimodal tests: image splice, loss masking, gradient flow."""

import torch

from megatron_amd.config import TransformerConfig
from megatron_amd.models.llava import DEFAULT_IMAGE_TOKEN_INDEX, LLaVAModel

from tests.utils import init_single


def _make_model():
    lang = TransformerConfig(num_layers=2, hidden_size=64, num_attention_heads=4,
                             vocab_size=128, max_position_embeddings=256)
    vis = TransformerConfig(num_layers=2, hidden_size=48, num_attention_heads=4,
                            vocab_size=1, max_position_embeddings=256)
    return LLaVAModel(lang, vis, img_h=28, img_w=28, patch_dim=14)


def _batch(b=2, s=10):
    ids = torch.randint(0, 128, (b, s))
    ids[:, 2] = DEFAULT_IMAGE_TOKEN_INDEX
    labels = torch.randint(0, 128, (b, s))
    images = torch.randn(b, 3, 28, 28)
    return images, ids, labels


def test_llava_forward_backward():
    init_single()
    torch.manual_seed(0)
    model = _make_model()
    images, ids, labels = _batch()
    loss = model(images=images, input_ids=ids, labels=labels)
    assert loss.shape[0] == 10 - 1 + model.img_seq_len  # expanded sequence
    total = loss.mean()
    total.backward()
    assert torch.isfinite(total)
    # gradients reach both towers and the projector
    assert model.vision_model.conv1.weight.grad is not None
    assert model.vision_projection.fc1.weight.grad is not None
    emb = model.language_model.embedding.weight
    assert emb.grad is not None


def test_llava_image_positions_masked():
    init_single()
    torch.manual_seed(1)
    model = _make_model()
    images, ids, labels = _batch(b=1)
    loss = model(images=images, input_ids=ids, labels=labels)  # [s', 1]
    p = 2
    img_span = loss[p : p + model.img_seq_len, 0]
    assert float(img_span.abs().sum()) == 0.0
    text_span = torch.cat([loss[:p, 0], loss[p + model.img_seq_len :, 0]])
    assert float(text_span.abs().sum()) > 0


def test_llava_image_affects_logits():
    init_single()
    torch.manual_seed(2)
    model = _make_model().eval()
    images, ids, _ = _batch(b=1)
    with torch.no_grad():
        l1 = model(images=images, input_ids=ids)
        l2 = model(images=images + 1.0, input_ids=ids)
    assert not torch.allclose(l1, l2)


def test_llava_text_only():
    init_single()
    torch.manual_seed(3)
    model = _make_model()
    ids = torch.randint(0, 128, (2, 12))
    labels = torch.randint(0, 128, (2, 12))
    loss = model(images=None, input_ids=ids, labels=labels)
    assert loss.shape == (12, 2)
    assert torch.isfinite(loss.mean())
