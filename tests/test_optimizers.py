"""Muon and CPU-offload optimizers (CPU).

Muon: Newton-Schulz orthogonalization quality, update semantics, and a
training-progress check.  CPU offload: exact step-for-step equivalence with
the in-memory FP32 optimizer (on CPU the copies are identity, so the math
must match bit-for-bit module the clip path).
"""

import copy

import torch
import torch.nn as nn

from megatron_amd.config import OptimizerConfig
from megatron_amd.optimizer import get_optimizer
from megatron_amd.optimizer.cpu_offload import CPUOffloadOptimizer
from megatron_amd.optimizer.muon import MuonOptimizer, muon_param, newton_schulz_orthogonalize
from megatron_amd.optimizer.optimizer import FP32Optimizer

from tests.utils import assert_close, init_single


class _Tiny(nn.Module):
    def __init__(self):
        super().__init__()
        torch.manual_seed(7)
        self.fc1 = nn.Linear(16, 32)
        self.fc2 = nn.Linear(32, 16)
        self.norm = nn.LayerNorm(16)

    def forward(self, x):
        return self.norm(self.fc2(torch.relu(self.fc1(x))))


def _loss_and_backward(model, x, y):
    loss = ((model(x) - y) ** 2).mean()
    model.zero_grad(set_to_none=True)
    loss.backward()
    return loss


def test_newton_schulz_orthogonalizes():
    torch.manual_seed(0)
    # ill-conditioned input: singular values spread over 3 decades
    U, _ = torch.linalg.qr(torch.randn(24, 24))
    V, _ = torch.linalg.qr(torch.randn(48, 24))
    s = torch.logspace(0, -3, 24)
    g = U @ torch.diag(s) @ V.T
    O = newton_schulz_orthogonalize(g, steps=5)
    # Muon's quintic drives all singular values into a band around 1
    # (by design ~[0.7, 1.2], not exactly 1) — vs the input's 1000x spread.
    sv = torch.linalg.svdvals(O)
    assert float(sv.min()) > 0.3 and float(sv.max()) < 1.5, sv
    # sign/direction preserved: <O, g> > 0
    assert float((O * g).sum()) > 0


def test_muon_param_selection():
    m = _Tiny()
    kinds = {name: muon_param(p) for name, p in m.named_parameters()}
    assert kinds["fc1.weight"] and kinds["fc2.weight"]
    assert not kinds["fc1.bias"] and not kinds["norm.weight"]
    m.fc1.weight.muon_exclude = True
    assert not muon_param(m.fc1.weight)


def test_muon_trains():
    init_single()
    model = _Tiny()
    cfg = OptimizerConfig(optimizer="muon", lr=3e-3, weight_decay=0.0, clip_grad=1.0)
    opt = get_optimizer(cfg, [model])
    assert isinstance(opt.chained_optimizers[0], MuonOptimizer)
    torch.manual_seed(1)
    x = torch.randn(64, 16)
    y = torch.tanh(x @ torch.randn(16, 16) * 0.5)  # learnable target
    first = float(_loss_and_backward(model, x, y).detach())
    for _ in range(60):
        _loss_and_backward(model, x, y)
        ok, norm, _ = opt.step()
        assert ok and torch.isfinite(norm)
    last = float(_loss_and_backward(model, x, y).detach())
    assert last < 0.5 * first, (first, last)


def test_muon_state_roundtrip():
    init_single()
    model = _Tiny()
    cfg = OptimizerConfig(optimizer="muon", lr=1e-3)
    opt = MuonOptimizer(cfg, [model])
    x, y = torch.randn(8, 16), torch.randn(8, 16)
    _loss_and_backward(model, x, y)
    opt.step()
    sd = copy.deepcopy(opt.state_dict())
    model2 = _Tiny()
    opt2 = MuonOptimizer(cfg, [model2])
    opt2.load_state_dict(sd)
    for a, b in zip(opt.main_params, opt2.main_params):
        assert_close(a, b, rtol=0, atol=0)
    for p, p2 in zip(model.parameters(), model2.parameters()):
        assert_close(p, p2, rtol=0, atol=0)


def test_cpu_offload_matches_fp32():
    init_single()
    model_a = _Tiny()
    model_b = copy.deepcopy(model_a)
    cfg = OptimizerConfig(lr=1e-3, weight_decay=0.01, clip_grad=1.0)
    opt_a = FP32Optimizer(cfg, [model_a])
    opt_b = CPUOffloadOptimizer(cfg, [model_b])
    torch.manual_seed(2)
    for step in range(5):
        x, y = torch.randn(16, 16), torch.randn(16, 16)
        _loss_and_backward(model_a, x, y)
        _loss_and_backward(model_b, x, y)
        opt_a.step()
        opt_b.step()
    for (na, pa), (nb, pb) in zip(model_a.named_parameters(), model_b.named_parameters()):
        assert_close(pa, pb, rtol=1e-6, atol=1e-6, msg=na)


def test_cpu_offload_selected_by_config():
    init_single()
    model = _Tiny()
    cfg = OptimizerConfig(lr=1e-3, optimizer_cpu_offload=True)
    opt = get_optimizer(cfg, [model])
    assert isinstance(opt.chained_optimizers[0], CPUOffloadOptimizer)


def test_decoupled_lr_groups():
    from megatron_amd.config import TransformerConfig
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.optimizer.scheduler import OptimizerParamScheduler

    init_single()
    torch.manual_seed(0)
    model = GPTModel(TransformerConfig(
        num_layers=1, hidden_size=32, num_attention_heads=4, num_query_groups=2,
        vocab_size=64, ffn_hidden_size=48, gradient_accumulation_fusion=False))
    cfg = OptimizerConfig(lr=1e-3, decoupled_lr=1e-4, lr_decay_style="constant")
    opt = get_optimizer(cfg, [model])
    assert len(opt.chained_optimizers) == 2
    main, emb = opt.chained_optimizers
    emb_names = {id(p) for p in emb.params}
    assert id(model.embedding.weight) in emb_names
    assert id(model.output_layer.weight) in emb_names
    for p in main.params:
        assert not getattr(p, "is_embedding_or_output_parameter", False)
    # every trainable param is in exactly one group
    assert len(main.params) + len(emb.params) == sum(1 for p in model.parameters() if p.requires_grad)
    # scheduler drives both lrs at their ratios
    opt.set_lr(5e-4)
    assert abs(main.get_lr() - 5e-4) < 1e-12
    assert abs(emb.get_lr() - 5e-5) < 1e-12
    # a step trains both groups
    toks = torch.randint(0, 64, (2, 8))
    out = model(toks, position_ids=None, attention_mask=None)
    out.square().mean().backward()
    w_emb = model.embedding.weight.detach().clone()
    w_main = model.decoder.layers[0].self_attention.linear_qkv.weight.detach().clone()
    ok, _, _ = opt.step()
    assert ok
    assert not torch.equal(w_emb, model.embedding.weight)
    assert not torch.equal(w_main, model.decoder.layers[0].self_attention.linear_qkv.weight)


def test_layer_wise_matches_fp32_without_clip():
    from megatron_amd.optimizer.layer_wise import LayerWiseOptimizer

    init_single()
    model_a = _Tiny()
    model_b = copy.deepcopy(model_a)
    cfg = OptimizerConfig(lr=1e-3, weight_decay=0.01, clip_grad=0.0)
    opt_a = FP32Optimizer(cfg, [model_a])
    opt_b = LayerWiseOptimizer(cfg, [model_b])
    torch.manual_seed(4)
    for _ in range(4):
        x, y = torch.randn(16, 16), torch.randn(16, 16)
        # a: standard order
        opt_a.zero_grad()
        ((model_a(x) - y) ** 2).mean().backward()
        opt_a.step()
        # b: updates fire inside backward
        opt_b.zero_grad()
        ((model_b(x) - y) ** 2).mean().backward()
        ok, norm, _ = opt_b.step()
        assert ok and float(norm) > 0
        # grads were freed by the in-backward updates
        assert all(p.grad is None for p in model_b.parameters())
    for (na, pa), (nb, pb) in zip(model_a.named_parameters(), model_b.named_parameters()):
        assert_close(pa, pb, rtol=1e-5, atol=1e-6, msg=na)
    opt_b.close()


def test_sgd_optimizer_matches_torch():
    """--optimizer sgd: momentum-SGD through the mixed-precision wrapper
    equals torch.optim.SGD on an unwrapped twin."""
    import copy

    from megatron_amd.config import DDPConfig, OptimizerConfig, TransformerConfig
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.parallel.random import model_parallel_seed
    from megatron_amd.training.training import setup_model_and_optimizer
    from tests.utils import init_single

    init_single()
    model_parallel_seed(17)

    def provider(config, pre_process=True, post_process=True, vp_stage=None):
        torch.manual_seed(21)
        return GPTModel(config)

    cfg = TransformerConfig(num_layers=2, hidden_size=32, num_attention_heads=4,
                            num_query_groups=4, vocab_size=64, ffn_hidden_size=48,
                            gradient_accumulation_fusion=False)
    opt_cfg = OptimizerConfig(lr=1e-2, weight_decay=0.0, clip_grad=0.0,
                              optimizer="sgd", sgd_momentum=0.9)
    chunks, opt = setup_model_and_optimizer(provider, cfg, opt_cfg,
                                            DDPConfig(grad_reduce_in_fp32=True))
    torch.manual_seed(21)
    twin = GPTModel(cfg)
    twin.load_state_dict({k: v for k, v in chunks[0].module.state_dict().items()})
    topt = torch.optim.SGD(twin.parameters(), lr=1e-2, momentum=0.9)

    tokens = torch.randint(0, 64, (2, 10), generator=torch.Generator().manual_seed(4))
    for _ in range(3):
        chunks[0].zero_grad_buffer()
        chunks[0](tokens, labels=tokens).sum().backward()
        chunks[0].start_grad_sync()
        ok, _, _ = opt.step()
        assert ok

        topt.zero_grad()
        twin(tokens, labels=tokens).sum().backward()
        topt.step()

    for (n, p), (n2, q) in zip(chunks[0].module.named_parameters(),
                               twin.named_parameters()):
        assert n == n2
        torch.testing.assert_close(p.detach(), q.detach(), rtol=1e-5, atol=1e-6, msg=n)
