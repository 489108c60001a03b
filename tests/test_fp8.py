"""FP8 path tests (K15). CPU: recipe/quantization semantics. GPU: fp8 linear
forward/backward vs bf16 within fp8 tolerance, and a training-step smoke."""

import pytest
import torch

from megatron_amd.ops.fp8 import (
    E4M3_MAX,
    DelayedScaling,
    fp8_eligible,
    make_recipes,
    quantize_fp8,
)
from tests.utils import init_single


def test_delayed_scaling_uses_history():
    r = DelayedScaling(history_len=4)
    t1 = torch.full((8,), 2.0)
    s1 = r.scale_for(t1)
    assert torch.isclose(s1, torch.tensor(E4M3_MAX / 2.0))  # first call: current amax
    t2 = torch.full((8,), 8.0)
    s2 = r.scale_for(t2)  # delayed: still based on history max (2.0)
    assert torch.isclose(s2, torch.tensor(E4M3_MAX / 2.0))
    s3 = r.scale_for(t1)  # now 8.0 is in the history
    assert torch.isclose(s3, torch.tensor(E4M3_MAX / 8.0))


def test_quantize_roundtrip():
    t = torch.randn(64, 64)
    r = DelayedScaling()
    q, inv = quantize_fp8(t, r.scale_for(t))
    back = q.float() * inv
    assert (back - t).abs().max() < 0.1 * t.abs().max()
    # saturation: huge outlier clamps instead of inf
    t2 = t.clone()
    t2[0, 0] = 1e6
    q2, inv2 = quantize_fp8(t2, torch.tensor(1.0))
    assert torch.isfinite(q2.float()).all()


def test_eligibility():
    x = torch.randn(32, 64)
    w = torch.randn(128, 64)
    assert not fp8_eligible(x, w)  # cpu
    assert len(make_recipes("hybrid")) == 3


@pytest.mark.gpu
def test_fp8_linear_matches_bf16():
    from megatron_amd.ops.fp8 import fp8_linear

    torch.manual_seed(0)
    x = torch.randn(256, 512, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    w = (torch.randn(1024, 512, device="cuda", dtype=torch.bfloat16) * 0.02).requires_grad_()
    y = fp8_linear(x, w, make_recipes("hybrid"))
    y_ref = x @ w.t()
    rel = (y - y_ref).float().norm() / y_ref.float().norm()
    assert rel < 0.06, f"fwd rel err {rel}"
    dy = torch.randn_like(y)
    y.backward(dy)
    gx, gw = x.grad.clone(), w.grad.clone()
    x.grad = w.grad = None
    y_ref.backward(dy)
    assert (gx - x.grad).float().norm() / x.grad.float().norm() < 0.12
    assert (gw - w.grad).float().norm() / w.grad.float().norm() < 0.06  # bf16 wgrad


@pytest.mark.gpu
def test_fp8_gpt_training_step():
    from megatron_amd.config import TransformerConfig
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.parallel import grid as G
    from megatron_amd.parallel.random import model_parallel_seed

    G.destroy_model_parallel()
    G.initialize_model_parallel(world_size=1, rank=0)
    model_parallel_seed(3)
    cfg = TransformerConfig(num_layers=2, hidden_size=256, num_attention_heads=4,
                            num_query_groups=2, ffn_hidden_size=512, vocab_size=512,
                            max_position_embeddings=256, bf16=True, fp8="hybrid")
    model = GPTModel(cfg).cuda()
    tokens = torch.randint(0, 512, (4, 128), device="cuda")
    losses = []
    opt = torch.optim.AdamW(model.parameters(), lr=3e-4)
    for _ in range(8):
        loss = model(tokens, labels=tokens).float().mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0], f"fp8 training did not reduce loss: {losses}"
    G.destroy_model_parallel()


def test_mxfp4_round_trip_exact_values():
    from megatron_amd.ops.fp4 import FP4_VALUES, dequantize_mxfp4, quantize_mxfp4

    # a block of exactly representable values at scale 1 must round-trip exactly
    vals = torch.tensor([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0] * 4)
    q, s = quantize_mxfp4(vals.unsqueeze(0))
    assert torch.allclose(s, torch.ones_like(s))  # amax=6 -> scale 1
    back = dequantize_mxfp4(q, s)
    assert torch.equal(back.squeeze(0), vals)


def test_mxfp4_error_bounded():
    from megatron_amd.ops.fp4 import mxfp4_quantization_error, quantize_mxfp4

    torch.manual_seed(0)
    t = torch.randn(64, 128) * 3.7
    err = mxfp4_quantization_error(t)
    assert err < 0.12, err  # fp4 w/ block scales: ~6-10% RMS on gaussians
    q, s = quantize_mxfp4(t)
    # scales are powers of two
    assert torch.allclose(torch.log2(s), torch.log2(s).round(), atol=0)
    # every quantized magnitude is on the E2M1 grid
    from megatron_amd.ops.fp4 import FP4_VALUES
    flat = q.abs().reshape(-1, 1)
    on_grid = (flat - FP4_VALUES).abs().min(dim=-1).values
    assert float(on_grid.max()) == 0.0


def test_mxfp4_pack_unpack():
    from megatron_amd.ops.fp4 import pack_fp4_codes, quantize_mxfp4, unpack_fp4_codes

    torch.manual_seed(1)
    t = torch.randn(4, 64)
    q, s = quantize_mxfp4(t)
    packed = pack_fp4_codes(q)
    assert packed.dtype == torch.uint8 and packed.numel() == q.numel() // 2
    back = unpack_fp4_codes(packed, q.numel()).reshape(q.shape)
    assert torch.equal(back, q)


@pytest.mark.gpu
def test_fp8_training_tracks_bf16():
    """Short training run: fp8-GEMM loss trace stays within tolerance of the
    bf16 trace (reference core/fp8_utils.py recipes)."""
    import torch

    from megatron_amd.config import DDPConfig, OptimizerConfig, TransformerConfig
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.parallel.random import model_parallel_seed
    from megatron_amd.training.training import setup_model_and_optimizer, train_step
    from tests.utils import init_single

    def run(fp8):
        init_single()
        model_parallel_seed(77)
        cfg = TransformerConfig(
            num_layers=4, hidden_size=256, num_attention_heads=8, num_query_groups=4,
            ffn_hidden_size=512, vocab_size=512, bf16=True,
            fp8="hybrid" if fp8 else None,
            gradient_accumulation_fusion=True, max_position_embeddings=256,
        )
        opt_cfg = OptimizerConfig(lr=1e-3, bf16=True, use_distributed_optimizer=False)

        def provider(config, pre_process=True, post_process=True, vp_stage=None):
            return GPTModel(config, pre_process=pre_process, post_process=post_process)

        chunks, opt = setup_model_and_optimizer(provider, cfg, opt_cfg, device=torch.device("cuda"))
        g = torch.Generator(device="cpu").manual_seed(5)
        # one fixed batch repeated: the loss must drop (memorization) and the
        # fp8 trajectory must track bf16
        toks = torch.randint(0, 512, (4, 128), generator=g).cuda()
        labels = torch.randint(0, 512, (4, 128), generator=g).cuda()
        losses = []
        for _ in range(20):

            def fwd(it, model):
                out = model(toks, labels=labels)

                def loss_func(loss_sb):
                    s = loss_sb.sum()
                    return s, torch.tensor(loss_sb.numel(), device="cuda"), {"loss_sum": s.detach()}

                return out, loss_func

            r = train_step(fwd, None, chunks, opt, cfg, 1, 128, 4)
            losses.append(r["lm_loss"])
        return losses

    bf16_losses = run(False)
    fp8_losses = run(True)
    # same trajectory within fp8 quantization noise
    for a, b in zip(bf16_losses[5:], fp8_losses[5:]):
        assert abs(a - b) / max(abs(b), 1e-6) < 0.08, (bf16_losses, fp8_losses)
    assert fp8_losses[-1] < fp8_losses[0], "fp8 run is not learning"
