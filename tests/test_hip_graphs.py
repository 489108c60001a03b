"""hipGraph layer capture: CPU-side guards + GPU capture equivalence."""

import pytest
import torch

from megatron_amd.config import TransformerConfig
from megatron_amd.models.gpt import GPTModel
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.random import model_parallel_seed
from megatron_amd.transformer.hip_graphs import capture_block_hip_graphs

from tests.utils import init_single


def _cfg(**kw):
    base = dict(num_layers=2, hidden_size=64, num_attention_heads=4, num_query_groups=2,
                vocab_size=96, ffn_hidden_size=128, gradient_accumulation_fusion=False)
    base.update(kw)
    return TransformerConfig(**base)


def test_capture_requires_gpu():
    init_single()
    model = GPTModel(_cfg())
    if not torch.cuda.is_available():
        with pytest.raises(RuntimeError, match="requires a GPU"):
            capture_block_hip_graphs(model.decoder, torch.randn(8, 2, 64))


def test_uncaptured_block_runs_eager():
    init_single()
    model = GPTModel(_cfg())
    toks = torch.randint(0, 96, (2, 8))
    out = model(toks, position_ids=None, attention_mask=None)
    assert out.shape == (8, 2, 96)
    assert getattr(model.decoder, "_graphed_layers", None) is None


@pytest.mark.gpu
def test_hip_graph_capture_matches_eager():
    init_single()
    dev = torch.device("cuda:0")
    cfg = _cfg(params_dtype=torch.bfloat16, bf16=True)
    model_parallel_seed(11)
    model = GPTModel(cfg).to(dev).bfloat16()
    toks = torch.randint(0, 96, (2, 32), device=dev)

    # eager reference forward+backward
    out_e = model(toks, position_ids=None, attention_mask=None)
    loss_e = out_e.float().square().mean()
    loss_e.backward()
    grads_e = {n: p.grad.detach().clone() for n, p in model.named_parameters() if p.grad is not None}
    model.zero_grad(set_to_none=True)
    out_ref = out_e.detach().clone()
    del out_e, loss_e  # capture requires no live autograd graphs (step boundary)

    # capture on the activation shape [s, b, h]
    sample = torch.randn(32, 2, cfg.hidden_size, device=dev, dtype=torch.bfloat16)
    freqs = model._rotary_freqs(32, dev)
    n = capture_block_hip_graphs(model.decoder, sample, rotary_freqs=freqs)
    assert n == cfg.num_layers
    model.zero_grad(set_to_none=True)

    out_g = model(toks, position_ids=None, attention_mask=None)
    loss_g = out_g.float().square().mean()
    loss_g.backward()
    assert torch.allclose(out_g, out_ref, atol=3e-2, rtol=3e-2)
    for nm, p in model.named_parameters():
        if p.grad is not None and nm in grads_e:
            assert torch.allclose(p.grad, grads_e[nm], atol=5e-2, rtol=5e-2), nm

    # a different shape transparently falls back to eager
    toks2 = torch.randint(0, 96, (2, 16), device=dev)
    out2 = model(toks2, position_ids=None, attention_mask=None)
    assert out2.shape == (16, 2, 96)


@pytest.mark.gpu
def test_hip_graphs_with_overlap_grad_reduce():
    """Graphs + overlap_grad_reduce coexist: graphed backward fires the DDP
    grad-ready callbacks for fused-wgrad params, and main_grad matches the
    eager DDP path."""
    from megatron_amd.config import DDPConfig
    from megatron_amd.distributed.ddp import DistributedDataParallel

    init_single()
    dev = torch.device("cuda:0")
    cfg = _cfg(params_dtype=torch.bfloat16, bf16=True, gradient_accumulation_fusion=True)
    model_parallel_seed(13)
    model = GPTModel(cfg).to(dev).bfloat16()
    ddp = DistributedDataParallel(cfg, DDPConfig(overlap_grad_reduce=True, grad_reduce_in_fp32=True), model)
    toks = torch.randint(0, 96, (2, 32), device=dev)

    def run():
        ddp.zero_grad_buffer() if hasattr(ddp, "zero_grad_buffer") else None
        for b in ddp.buffers:
            b.grad_data.zero_()
            for bk in b.buckets:
                bk.comm_handle = None
                bk.params_with_grad = set()
        out = ddp(toks, position_ids=None, attention_mask=None)
        out.float().square().mean().backward()
        ddp.finish_grad_sync()
        return {n: p.main_grad.detach().clone() for n, p in model.named_parameters()}

    grads_eager = run()

    with ddp.no_sync():
        sample = torch.randn(32, 2, cfg.hidden_size, device=dev, dtype=torch.bfloat16)
        freqs = model._rotary_freqs(32, dev)
        capture_block_hip_graphs(model.decoder, sample, rotary_freqs=freqs)
    grads_graphed = run()
    # every bucket must have been marked fully ready (callbacks fired)
    for nm in grads_eager:
        assert torch.allclose(grads_graphed[nm], grads_eager[nm], atol=5e-2, rtol=5e-2), nm
