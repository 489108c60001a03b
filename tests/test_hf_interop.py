"""HF interop wrappers (reference models/huggingface): offline build from an
explicit transformers config, module contract, frozen-tower projection."""

import pytest
import torch

from megatron_amd.config import TransformerConfig
from tests.utils import init_single

transformers = pytest.importorskip("transformers")


def _cfg():
    return TransformerConfig(num_layers=2, hidden_size=64, num_attention_heads=4,
                             num_query_groups=4, ffn_hidden_size=128, vocab_size=128,
                             max_position_embeddings=64)


def _tiny_hf_cfg():
    return transformers.LlamaConfig(
        vocab_size=64, hidden_size=32, intermediate_size=48, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=4, max_position_embeddings=64)


def test_auto_hf_model_wraps_and_tags_params():
    from megatron_amd.models.hf_interop import AutoHuggingFaceModel

    init_single()
    m = AutoHuggingFaceModel(_cfg(), hf_config=_tiny_hf_cfg())
    assert all(getattr(p, "average_gradients_across_tp_domain", False)
               for p in m.parameters())
    ids = torch.randint(0, 64, (2, 8))
    out = m(input_ids=ids)
    assert out.last_hidden_state.shape == (2, 8, 32)
    m.set_input_tensor(torch.zeros(1))  # module contract


def test_hf_encoder_tower_projects_and_freezes():
    from megatron_amd.models.hf_interop import HuggingFaceEncoderTower

    init_single()
    cfg = _cfg()
    tower = HuggingFaceEncoderTower(cfg, hf_config=_tiny_hf_cfg(), freeze=True)
    ids = torch.randint(0, 64, (2, 8))
    h = tower(input_ids=ids)
    assert h.shape == (8, 2, cfg.hidden_size)  # [s, b, h] encoder contract
    h.sum().backward()
    assert tower.projector.weight.grad is not None
    assert all(p.grad is None for p in tower.backbone.parameters())
