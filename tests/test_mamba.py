"""Mamba2 / SSD tests: chunked scan vs naive recurrence, decode step vs
prefill, hybrid allocation, MambaModel train step, TP=2 equivalence."""

import pytest
import torch
import torch.nn.functional as F

from megatron_amd.config import TransformerConfig
from megatron_amd.models.mamba import MambaModel
from megatron_amd.ssm.hybrid_allocation import allocate_layers, pattern_from_ratios
from megatron_amd.ssm.mamba_mixer import MambaMixer
from megatron_amd.ssm.ssd import ssd_chunked_scan, ssd_step

from tests.utils import assert_close, init_single, spawn_dist


def naive_recurrence(x, dt, A, B, C, D=None, initial_state=None):
    """Per-timestep fp64 reference of the selective SSM."""
    b, l, h, p = x.shape
    g, n = B.shape[2], B.shape[3]
    hpg = h // g
    x, dt, B, C = x.double(), dt.double(), B.double(), C.double()
    A = A.double()
    state = (
        initial_state.double()
        if initial_state is not None
        else torch.zeros(b, h, p, n, dtype=torch.float64)
    )
    ys = []
    for t in range(l):
        dA = torch.exp(dt[:, t] * A)  # [b, h]
        Bh = B[:, t].unsqueeze(2).expand(b, g, hpg, n).reshape(b, h, n)
        Ch = C[:, t].unsqueeze(2).expand(b, g, hpg, n).reshape(b, h, n)
        state = state * dA[..., None, None] + (
            (dt[:, t].unsqueeze(-1) * x[:, t]).unsqueeze(-1) * Bh.unsqueeze(-2)
        )
        y = torch.einsum("bhpn,bhn->bhp", state, Ch)
        if D is not None:
            y = y + D.double().view(1, h, 1) * x[:, t]
        ys.append(y)
    return torch.stack(ys, dim=1).float(), state.float()


@pytest.mark.parametrize("l,chunk", [(64, 16), (50, 16), (13, 32)])
def test_ssd_chunked_scan_matches_recurrence(l, chunk):
    torch.manual_seed(0)
    b, h, p, g, n = 2, 4, 8, 2, 16
    x = torch.randn(b, l, h, p)
    dt = F.softplus(torch.randn(b, l, h))
    A = -torch.rand(h) - 0.1
    B = torch.randn(b, l, g, n)
    C = torch.randn(b, l, g, n)
    D = torch.rand(h)
    y, final = ssd_chunked_scan(x, dt, A, B, C, D=D, chunk_size=chunk, return_final_state=True)
    y_ref, final_ref = naive_recurrence(x, dt, A, B, C, D=D)
    assert_close(y, y_ref, rtol=1e-3, atol=1e-3)
    assert_close(final, final_ref, rtol=1e-3, atol=1e-3)


def test_ssd_initial_state():
    torch.manual_seed(1)
    b, l, h, p, g, n = 1, 24, 2, 4, 1, 8
    x = torch.randn(b, l, h, p)
    dt = F.softplus(torch.randn(b, l, h))
    A = -torch.rand(h) - 0.1
    B, C = torch.randn(b, l, g, n), torch.randn(b, l, g, n)
    s0 = torch.randn(b, h, p, n)
    y, _ = ssd_chunked_scan(x, dt, A, B, C, chunk_size=8, initial_state=s0, return_final_state=True)
    y_ref, _ = naive_recurrence(x, dt, A, B, C, initial_state=s0)
    assert_close(y, y_ref, rtol=1e-3, atol=1e-3)


def test_ssd_step_matches_scan():
    torch.manual_seed(2)
    b, l, h, p, g, n = 2, 12, 4, 4, 2, 8
    x = torch.randn(b, l, h, p)
    dt = F.softplus(torch.randn(b, l, h))
    A = -torch.rand(h) - 0.1
    B, C = torch.randn(b, l, g, n), torch.randn(b, l, g, n)
    y_scan = ssd_chunked_scan(x, dt, A, B, C, chunk_size=4)
    state = torch.zeros(b, h, p, n)
    ys = []
    for t in range(l):
        y, state = ssd_step(x[:, t], dt[:, t], A, B[:, t], C[:, t], state)
        ys.append(y)
    assert_close(torch.stack(ys, dim=1), y_scan, rtol=1e-3, atol=1e-3)


def test_ssd_backward():
    torch.manual_seed(3)
    b, l, h, p, g, n = 1, 20, 2, 4, 1, 8
    x = torch.randn(b, l, h, p, requires_grad=True)
    dt = F.softplus(torch.randn(b, l, h)).requires_grad_()
    A = (-torch.rand(h) - 0.1).requires_grad_()
    B = torch.randn(b, l, g, n, requires_grad=True)
    C = torch.randn(b, l, g, n, requires_grad=True)
    y = ssd_chunked_scan(x, dt, A, B, C, chunk_size=8)
    y.sum().backward()
    for t in (x, dt, A, B, C):
        assert t.grad is not None and torch.isfinite(t.grad).all()


def _mixer_config(**kw):
    defaults = dict(
        num_layers=2, hidden_size=32, num_attention_heads=4, vocab_size=96,
        mamba_state_dim=16, mamba_head_dim=8, mamba_num_groups=2,
        mamba_chunk_size=16, max_position_embeddings=128,
    )
    defaults.update(kw)
    return TransformerConfig(**defaults)


def test_mamba_mixer_decode_matches_prefill():
    init_single()
    torch.manual_seed(4)
    cfg = _mixer_config()
    mixer = MambaMixer(cfg).eval()
    s, b = 10, 2
    x = torch.randn(s, b, cfg.hidden_size)
    with torch.no_grad():
        full = mixer(x)
        # prefill the first 6 tokens, then decode 4 one at a time
        state = mixer.allocate_inference_state(b, x.device, x.dtype)
        out_pre = mixer(x[:6], inference_state=state)
        outs = [out_pre]
        for t in range(6, s):
            outs.append(mixer(x[t : t + 1], inference_state=state))
        stepped = torch.cat(outs, dim=0)
    assert_close(stepped, full, rtol=1e-4, atol=1e-4)


def test_hybrid_allocation():
    assert allocate_layers(4) == ["M", "M", "M", "M"]
    assert allocate_layers(4, override_pattern="M*M-") == ["M", "*", "M", "-"]
    pat = pattern_from_ratios(8, attention_ratio=0.25, mlp_ratio=0.25)
    assert len(pat) == 8 and pat.count("*") == 2 and pat.count("-") == 2
    with pytest.raises(ValueError):
        allocate_layers(3, override_pattern="MX*")
    with pytest.raises(ValueError):
        allocate_layers(3, override_pattern="MM")


def test_mamba_model_train_step():
    init_single()
    torch.manual_seed(5)
    cfg = _mixer_config(num_layers=3, hybrid_override_pattern="M*-")
    model = MambaModel(cfg)
    ids = torch.randint(0, cfg.vocab_size, (2, 24))
    labels = torch.randint(0, cfg.vocab_size, (2, 24))
    loss = model(input_ids=ids, labels=labels).mean()
    loss.backward()
    assert torch.isfinite(loss)
    grads = [p.grad for p in model.parameters() if p.grad is not None]
    assert len(grads) > 0 and all(torch.isfinite(g).all() for g in grads)


def test_mamba_model_decode_matches_forward():
    init_single()
    torch.manual_seed(6)
    cfg = _mixer_config(num_layers=2, hybrid_override_pattern="MM")
    model = MambaModel(cfg).eval()
    ids = torch.randint(0, cfg.vocab_size, (1, 12))
    with torch.no_grad():
        logits_full = model(input_ids=ids)  # [s, b, V]
        states = model.decoder.allocate_inference_states(1, ids.device, torch.float32)
        pre = model(input_ids=ids[:, :8], inference_states=states)
        outs = [pre]
        for t in range(8, 12):
            outs.append(model(input_ids=ids[:, t : t + 1], inference_states=states))
        logits_step = torch.cat(outs, dim=0)
    assert_close(logits_step, logits_full, rtol=1e-4, atol=1e-4)


def _tp_worker(rank, world):
    from megatron_amd.parallel import grid as G
    from megatron_amd.parallel.random import model_parallel_seed

    G.destroy_model_parallel()
    G.initialize_model_parallel(tensor_parallel_size=world)
    model_parallel_seed(1234)
    cfg = _mixer_config(tensor_parallel_size=world)
    torch.manual_seed(7)
    mixer = MambaMixer(cfg)

    # dense single-rank twin built from the TP shards
    import torch.distributed as dist

    x = torch.randn(8, 2, cfg.hidden_size)
    dist.broadcast(x, src=0)
    out = mixer(x)
    loss = out.float().pow(2).mean()
    loss.backward()
    # TP output is all-reduced -> identical across ranks; loss must match
    losses = [torch.zeros_like(loss) for _ in range(world)]
    dist.all_gather(losses, loss)
    assert torch.allclose(losses[0], losses[1], rtol=1e-5, atol=1e-6)


def test_mamba_mixer_tp2_consistent():
    spawn_dist(_tp_worker, world_size=2)


def _mamba_cp2_case(rank, world):
    import json
    import os

    import torch.distributed as dist

    from megatron_amd.parallel import grid as G
    from megatron_amd.parallel.context_parallel import get_batch_on_this_cp_rank

    from megatron_amd.parallel.random import model_parallel_seed

    G.initialize_model_parallel(context_parallel_size=2)
    model_parallel_seed(1234)
    cfg = _mixer_config(context_parallel_size=2)
    torch.manual_seed(42)
    m = MambaModel(cfg)
    torch.manual_seed(9)
    tokens = torch.randint(0, cfg.vocab_size, (2, 16))
    labels = torch.randint(0, cfg.vocab_size, (2, 16))
    dist.broadcast(tokens, src=0)
    dist.broadcast(labels, src=0)
    local = get_batch_on_this_cp_rank({"tokens": tokens, "labels": labels})
    loss_sb = m(local["tokens"], labels=local["labels"])  # [s/cp, b]
    # total loss over the full sequence = sum over CP ranks
    total = loss_sb.sum()
    dist.all_reduce(total)
    if rank == 0:
        with open(os.environ["MAMBA_CP_OUT"], "w") as f:
            json.dump(float(total), f)


def test_mamba_cp2_matches_single(tmp_path, monkeypatch):
    """Mamba under CP=2 (gathered-scan path): summed per-rank losses equal
    the single-process full-sequence loss."""
    import json

    from megatron_amd.parallel.random import model_parallel_seed as _mps

    out = tmp_path / "mamba_cp.json"
    monkeypatch.setenv("MAMBA_CP_OUT", str(out))
    init_single()
    _mps(1234)
    cfg = _mixer_config()
    torch.manual_seed(42)
    m = MambaModel(cfg)
    torch.manual_seed(9)
    tokens = torch.randint(0, cfg.vocab_size, (2, 16))
    labels = torch.randint(0, cfg.vocab_size, (2, 16))
    ref = float(m(tokens, labels=labels).sum())

    spawn_dist(_mamba_cp2_case, 2)
    got = json.load(open(out))
    assert abs(got - ref) < 2e-3 * max(abs(ref), 1.0), (got, ref)


def _mamba_cp2_headsplit_case(rank, world):
    """Asserts the head-split path (no redundant gathered scan) is the one
    running under CP=2, with grads flowing."""
    import torch.distributed as dist

    from megatron_amd.parallel import grid as G
    from megatron_amd.parallel.context_parallel import get_batch_on_this_cp_rank
    from megatron_amd.parallel.random import model_parallel_seed
    from megatron_amd.ssm.mamba_mixer import MambaMixer

    G.initialize_model_parallel(context_parallel_size=2)
    model_parallel_seed(1234)
    cfg = _mixer_config(context_parallel_size=2)
    torch.manual_seed(42)
    m = MambaModel(cfg)
    calls = []
    orig = MambaMixer._forward_headsplit_cp

    def spy(self, *a, **k):
        calls.append(1)
        return orig(self, *a, **k)

    MambaMixer._forward_headsplit_cp = spy
    try:
        torch.manual_seed(9)
        tokens = torch.randint(0, cfg.vocab_size, (2, 16))
        labels = torch.randint(0, cfg.vocab_size, (2, 16))
        dist.broadcast(tokens, src=0)
        dist.broadcast(labels, src=0)
        local = get_batch_on_this_cp_rank({"tokens": tokens, "labels": labels})
        loss = m(local["tokens"], labels=local["labels"]).sum()
        loss.backward()
    finally:
        MambaMixer._forward_headsplit_cp = orig
    assert len(calls) >= 1, "head-split CP path did not run"
    for n, p in m.named_parameters():
        if "conv_weight" in n or "A_log" in n:
            assert p.grad is not None and torch.isfinite(p.grad).all(), n


def test_mamba_cp2_headsplit_active_and_trains():
    spawn_dist(_mamba_cp2_headsplit_case, 2)
