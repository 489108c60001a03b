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
