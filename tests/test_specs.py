"""ModuleSpec system tests: declarative layer composition, slot overrides,
import-path specs, per-layer dense/MoE block specs."""

import torch
import torch.nn as nn

from megatron_amd.config import TransformerConfig
from megatron_amd.moe.moe_layer import MoELayer
from megatron_amd.transformer.block import TransformerBlock, TransformerLayer
from megatron_amd.transformer.layer_specs import (
    TransformerLayerSubmodules,
    get_gpt_decoder_block_spec,
    get_gpt_layer_spec,
)
from megatron_amd.transformer.mlp import MLP
from megatron_amd.transformer.spec_utils import ModuleSpec, build_module

from tests.utils import assert_close, init_single


def _cfg(**kw):
    d = dict(num_layers=2, hidden_size=32, num_attention_heads=4, vocab_size=64,
             max_position_embeddings=64)
    d.update(kw)
    return TransformerConfig(**d)


def test_spec_builds_default_layer():
    init_single()
    spec = get_gpt_layer_spec()
    layer = build_module(spec, _cfg(), layer_number=0)
    assert isinstance(layer, TransformerLayer)
    assert isinstance(layer.mlp, MLP)


def test_spec_import_path():
    spec = ModuleSpec(module=("megatron_amd.transformer.mlp", "MLP"))
    init_single()
    mlp = build_module(spec, _cfg())
    assert isinstance(mlp, MLP)


class _IdentityMLP(nn.Module):
    def __init__(self, config):
        super().__init__()

    def forward(self, x):
        return torch.zeros_like(x)


def test_spec_slot_override():
    init_single()
    sub = TransformerLayerSubmodules(mlp=_IdentityMLP)
    layer = TransformerLayer(_cfg(), layer_number=0, submodules=sub)
    assert isinstance(layer.mlp, _IdentityMLP)


def test_decoder_block_spec_moe_mix():
    init_single()
    cfg = _cfg(num_layers=4, num_experts=4, moe_layer_freq=2, expert_parallel_size=1)
    specs = get_gpt_decoder_block_spec(cfg)
    mlps = [s.submodules.mlp for s in specs]
    assert mlps[0] is MLP and mlps[2] is MLP
    assert mlps[1] is MoELayer and mlps[3] is MoELayer


def test_block_from_specs_matches_default():
    init_single(seed=77)
    cfg = _cfg()
    torch.manual_seed(3)
    b1 = TransformerBlock(cfg)
    init_single(seed=77)
    torch.manual_seed(3)
    b2 = TransformerBlock(cfg, layer_specs=[get_gpt_layer_spec() for _ in range(cfg.num_layers)])
    x = torch.randn(8, 2, cfg.hidden_size)
    with torch.no_grad():
        assert_close(b1(x), b2(x), rtol=1e-6, atol=1e-7)
