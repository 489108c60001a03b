"""Generalized tensor parallelism (weight-remat sharding): GTP=2 over gloo
must match a plain nn.Linear forward and backward exactly (the wgrad
reduce-scatter sums partial per-rank wgrads; with identical inputs on both
ranks the shard equals 2x one rank's rows of the reference wgrad / DDP-style
averaging is the caller's business, so we feed rank-identical data)."""

import torch
import torch.distributed as dist
import torch.nn as nn

from megatron_amd.parallel import grid as G
from megatron_amd.parallel.gtp import GTPLinear
from megatron_amd.parallel.random import model_parallel_seed

from tests.utils import assert_close, init_single, spawn_dist


def test_gtp_single_rank_matches_linear():
    init_single()
    torch.manual_seed(0)
    ref = nn.Linear(8, 12)

    def init_from_ref(full):
        full.copy_(ref.weight)

    gtp = GTPLinear(8, 12, init_method=init_from_ref)
    with torch.no_grad():
        gtp.bias.copy_(ref.bias)
    x = torch.randn(4, 8, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    out = gtp(x)
    expect = ref(x2)
    assert_close(out, expect, rtol=1e-6, atol=1e-6)
    out.sum().backward()
    expect.sum().backward()
    assert_close(x.grad, x2.grad, rtol=1e-6, atol=1e-6)
    assert_close(gtp.weight.grad, ref.weight.grad, rtol=1e-6, atol=1e-6)
    assert_close(gtp.bias.grad, ref.bias.grad, rtol=1e-6, atol=1e-6)


def _gtp2_case(rank, world):
    G.initialize_model_parallel()  # dp group == world
    model_parallel_seed(5)
    torch.manual_seed(11)  # same on both ranks
    ref = nn.Linear(16, 8, bias=False)

    def init_from_ref(full):
        full.copy_(ref.weight)

    gtp = GTPLinear(16, 8, bias=False, init_method=init_from_ref)
    assert gtp.weight.shape == (4, 16)
    # shard holds this rank's rows of the full weight
    assert torch.allclose(gtp.weight, ref.weight[rank * 4:(rank + 1) * 4])

    x = torch.randn(6, 16, requires_grad=True)
    dist.broadcast(x.data, src=0)  # identical batch on both ranks
    x_ref = x.detach().clone().requires_grad_(True)

    out = gtp(x)
    expect = ref(x_ref)
    assert torch.allclose(out, expect, atol=1e-6)

    out.pow(2).sum().backward()
    expect.pow(2).sum().backward()
    assert torch.allclose(x.grad, x_ref.grad, atol=1e-5)
    # both ranks computed the same full wgrad; the reduce-scatter sums them,
    # so each shard is world * ref rows
    expect_shard = world * ref.weight.grad[rank * 4:(rank + 1) * 4]
    assert torch.allclose(gtp.weight.grad, expect_shard, atol=1e-4), \
        (gtp.weight.grad - expect_shard).abs().max()


def test_gtp2_matches_linear():
    spawn_dist(_gtp2_case, 2)


def _gtp_chain_case(rank, world):
    """Chained GTP layers prefetch each other's gathers and match the
    unchained computation exactly."""
    import torch.distributed as dist
    import torch.nn as nn

    from megatron_amd.parallel import grid as G
    from megatron_amd.parallel.gtp import GTPChain, GTPLinear

    G.initialize_model_parallel()
    torch.manual_seed(0)

    def build():
        torch.manual_seed(0)
        ls = [GTPLinear(16, 16, group=dist.group.WORLD,
                        init_method=lambda t: nn.init.normal_(t, 0, 0.1))
              for _ in range(3)]
        return ls

    plain = build()
    chained = build()
    GTPChain(chained)
    x = torch.randn(4, 16, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)

    def run(layers, inp):
        h = inp
        for l in layers:
            h = torch.relu(l(h))
        return h

    out_p = run(plain, x)
    out_c = run(chained, x2)
    torch.testing.assert_close(out_c, out_p, rtol=1e-6, atol=1e-7)
    out_p.sum().backward()
    out_c.sum().backward()
    torch.testing.assert_close(x2.grad, x.grad, rtol=1e-6, atol=1e-7)
    for a, b in zip(plain, chained):
        torch.testing.assert_close(b.weight.grad, a.weight.grad, rtol=1e-6, atol=1e-7)


def test_gtp_chain_prefetch_matches():
    from tests.utils import spawn_dist

    spawn_dist(_gtp_chain_case, 2)
