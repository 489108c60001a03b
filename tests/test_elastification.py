"""Elastic-width sub-networks (Flextron-style)."""

import torch
import torch.nn as nn

from megatron_amd.elastification import ElasticLinear, elastic_memory_profile, set_active_width


class _Net(nn.Module):
    def __init__(self):
        super().__init__()
        torch.manual_seed(0)
        self.up = ElasticLinear(16, 64)
        self.down = ElasticLinear(64, 16)

    def forward(self, x):
        return self.down(torch.relu(self.up(x)))


def test_elastic_slice_matches_manual():
    net = _Net()
    x = torch.randn(4, 16)
    full = net(x)
    set_active_width(net, 0.5, per_layer={"up": 0.5, "down": 0.5})
    # interface dims: up.in stays 16, down.out stays 16
    net.up.in_active = 16
    net.down.out_active = 16
    half = net(x)
    # manual computation with sliced weights
    h = torch.relu(torch.nn.functional.linear(x, net.up.weight[:32], net.up.bias[:32]))
    expect = torch.nn.functional.linear(h, net.down.weight[:16, :32], net.down.bias)
    assert torch.allclose(half, expect, atol=1e-6)
    assert half.shape == full.shape
    assert not torch.allclose(half, full, atol=1e-4)


def test_elastic_grads_only_in_active_slice():
    net = _Net()
    set_active_width(net, 0.25)
    net.up.in_active = 16
    net.down.out_active = 16
    out = net(torch.randn(4, 16))
    out.sum().backward()
    g = net.up.weight.grad
    assert g[:16].abs().sum() > 0        # active rows trained
    assert g[16:].abs().sum() == 0       # inactive rows untouched


def test_materialize_subnet():
    net = _Net()
    set_active_width(net, 0.5)
    net.up.in_active = 16
    sub = net.up.materialize()
    assert sub.weight.shape == (32, 16)
    x = torch.randn(2, 16)
    assert torch.allclose(sub(x), net.up(x), atol=1e-6)


def test_memory_profile_monotone():
    net = _Net()
    prof = elastic_memory_profile(net, [0.25, 0.5, 1.0])
    assert prof[0.25] < prof[0.5] < prof[1.0]
    # widths restored
    assert net.up.out_active == 64 and net.down.in_active == 64


def _tiny_gpt():
    from megatron_amd.config import TransformerConfig
    from megatron_amd.models.gpt import GPTModel
    from tests.utils import init_single

    init_single()
    torch.manual_seed(3)
    cfg = TransformerConfig(num_layers=2, hidden_size=32, num_attention_heads=4,
                            vocab_size=64, ffn_hidden_size=48,
                            gradient_accumulation_fusion=False)
    return GPTModel(cfg), cfg


def test_elastify_gpt_widths_and_equivalence():
    from megatron_amd.elastification.elastic import elastify_gpt, set_gpt_width

    model, cfg = _tiny_gpt()
    toks = torch.randint(0, 64, (2, 8))
    ref = model(toks, labels=toks).detach()
    n = elastify_gpt(model)
    assert n == 2
    set_gpt_width(model, 1.0)
    full = model(toks, labels=toks).detach()
    torch.testing.assert_close(full, ref, rtol=1e-5, atol=1e-6)  # full width == original
    set_gpt_width(model, 0.5)
    half = model(toks, labels=toks).detach()
    assert not torch.allclose(half, ref)  # genuinely smaller net


def test_sandwich_step_accumulates_all_widths():
    from megatron_amd.elastification.elastic import elastify_gpt, sandwich_step

    model, cfg = _tiny_gpt()
    elastify_gpt(model)
    toks = torch.randint(0, 64, (2, 8))

    losses = sandwich_step(model, lambda m: m(toks, labels=toks).mean(),
                           widths=(0.25, 0.5, 1.0))
    assert set(losses) >= {1.0, 0.25}
    # grads exist on full weights AND only the active slices of fc2 got the
    # small-width contributions — the shared tail must still have grads from
    # the full-width pass
    w2 = model.decoder.layers[0].mlp.mlp.linear_fc2.weight
    assert w2.grad is not None and w2.grad.abs().sum() > 0


def test_width_router_selects_and_backprops():
    from megatron_amd.elastification.elastic import elastify_gpt

    model, cfg = _tiny_gpt()
    elastify_gpt(model, with_router=True, latency_penalty=0.01)
    toks = torch.randint(0, 64, (2, 8))
    loss = model(toks, labels=toks).mean()
    for layer in model.decoder.layers:
        loss = loss + layer.mlp.router_aux
    loss.backward()
    r = model.decoder.layers[0].mlp.router
    assert r.proj.weight.grad is not None and r.proj.weight.grad.abs().sum() > 0


def test_materialize_gpt_matches_elastic_forward():
    from megatron_amd.elastification.elastic import elastify_gpt, materialize_gpt, set_gpt_width

    model, cfg = _tiny_gpt()
    elastify_gpt(model)
    set_gpt_width(model, 0.5)
    toks = torch.randint(0, 64, (2, 8))
    with torch.no_grad():
        ref = model(toks, labels=toks)
        dense = materialize_gpt(model, 0.5)
        got = dense(toks, labels=toks)
    torch.testing.assert_close(got, ref, rtol=1e-5, atol=1e-6)
