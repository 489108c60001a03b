"""Multi-token prediction tests (reference transformer/multi_token_prediction.py)."""

import torch

from megatron_amd.config import TransformerConfig
from megatron_amd.models.gpt import GPTModel
from megatron_amd.parallel.random import model_parallel_seed
from tests.utils import init_single

KW = dict(num_layers=2, hidden_size=64, num_attention_heads=4, num_query_groups=2,
          ffn_hidden_size=128, vocab_size=128, max_position_embeddings=64)


def test_mtp_zero_matches_baseline():
    init_single()
    model_parallel_seed(5)
    base = GPTModel(TransformerConfig(**KW))
    tokens = torch.randint(0, 128, (2, 32))
    labels = torch.randint(0, 128, (2, 32))
    loss0 = base(tokens, labels=labels)
    init_single()
    model_parallel_seed(5)
    same = GPTModel(TransformerConfig(**KW, mtp_num_layers=0))
    torch.testing.assert_close(loss0, same(tokens, labels=labels))


def test_mtp_adds_scaled_loss_and_trains():
    init_single()
    model_parallel_seed(5)
    cfg = TransformerConfig(**KW, mtp_num_layers=2)
    model = GPTModel(cfg)
    tokens = torch.randint(0, 128, (2, 32))
    labels = torch.randint(0, 128, (2, 32))
    loss = model(tokens, labels=labels)
    assert loss.shape == (32, 2)
    loss.sum().backward()
    # MTP head params get gradients
    g = model.mtp.heads[0].proj.weight.grad
    assert g is not None and torch.isfinite(g).all()
    # loss exceeds the main-only loss (extra positive terms)
    with torch.no_grad():
        model2 = model
        model2.mtp = None
        main_only = model2(tokens, labels=labels)
    assert float(loss.sum()) > float(main_only.sum())


# --- MTP under pipeline parallelism (reference multi_token_prediction.py +
# finalize_model_grads.py:164 embd-group exchange) --------------------------


import json
import os
import zlib

from megatron_amd.config import DDPConfig, OptimizerConfig
from megatron_amd.parallel import grid as G
from megatron_amd.training.training import setup_model_and_optimizer, train_step
from tests.utils import spawn_dist


def _fill_mtp_deterministic(model):
    core = model.module if hasattr(model, "module") else model

    def fill(t, key):
        g = torch.Generator().manual_seed(zlib.crc32(key.encode()) % (2**31))
        with torch.no_grad():
            t.copy_(torch.randn(t.shape, generator=g) * 0.02)

    if core.pre_process:
        # tied models: the single-process reference fills "embedding" then
        # overwrites the SAME tensor via the "output" key -> use "output"
        # so PP replicas (filled per stage) get identical values
        key = "output" if core.share_embeddings_and_output_weights else "embedding"
        fill(core.embedding.weight, key)
    for layer in core.decoder.layers:
        ln = layer.layer_number
        for name, p in layer.named_parameters():
            fill(p, f"layer{ln}.{name}")
    if core.post_process:
        fill(core.decoder.final_layernorm.weight, "final_ln")
        core.decoder.final_layernorm.weight.data.add_(1.0)
        fill(core.output_layer.weight, "output")
        if core.mtp_embedding is not None and not core.share_embeddings_and_output_weights:
            fill(core.mtp_embedding.weight, "embedding")  # replica of stage-0 weight
            # (tied: mtp_embedding IS the output weight, already filled)
        for k, head in enumerate(core.mtp.heads):
            for name, p in head.named_parameters():
                fill(p, f"mtp{k}.{name}")


def _mtp_run(cfg, n_steps=2):
    opt_cfg = OptimizerConfig(lr=1e-3, clip_grad=1.0)
    ddp_cfg = DDPConfig(grad_reduce_in_fp32=True, bucket_size=10_000)
    chunks, opt = setup_model_and_optimizer(_mtp_provider, cfg, opt_cfg, ddp_cfg)
    for c in chunks:
        _fill_mtp_deterministic(c)
    for o in opt.chained_optimizers:
        if hasattr(o, "reload_model_params"):
            o.reload_model_params()
    g = torch.Generator().manual_seed(3)
    batches = []
    for _ in range(n_steps * 2):
        t = torch.randint(0, 128, (2, 33), generator=g)
        batches.append({"tokens": t[:, :-1], "labels": t[:, 1:]})
    it = iter(batches)

    def fwd(data_iterator, model):
        batch = next(it if data_iterator is None else data_iterator)

        def loss_func(loss_sb):
            s = loss_sb.sum()
            return s, torch.tensor(loss_sb.numel()), {"loss_sum": s.detach()}

        return out_model(model, batch), loss_func

    def out_model(model, batch):
        return model(batch["tokens"], labels=batch["labels"])

    losses = []
    for s in range(n_steps):
        its = [iter(batches[s * 2 : (s + 1) * 2]) for _ in chunks]
        r = train_step(fwd, its, chunks, opt, cfg, 2, 32, 2)
        losses.append(r["lm_loss"])
    return losses


def _mtp_provider(config, pre_process=True, post_process=True, vp_stage=None):
    torch.manual_seed(42)
    return GPTModel(config, pre_process=pre_process, post_process=post_process, vp_stage=vp_stage)


def _mtp_pp2_case(rank, world, untie):
    G.initialize_model_parallel(pipeline_parallel_size=world)
    model_parallel_seed(1234)
    cfg = TransformerConfig(**KW, mtp_num_layers=1, pipeline_parallel_size=world,
                            untie_embeddings_and_output_weights=untie,
                            gradient_accumulation_fusion=True)
    losses = _mtp_run(cfg)
    if G.get_grid().is_pipeline_last_stage(ignore_virtual=True):
        with open(os.environ["MTP_TEST_OUT"], "w") as f:
            json.dump(losses, f)


def _mtp_single_ref(untie):
    init_single()
    model_parallel_seed(1234)
    cfg = TransformerConfig(**KW, mtp_num_layers=1,
                            untie_embeddings_and_output_weights=untie,
                            gradient_accumulation_fusion=True)
    return _mtp_run(cfg)


def test_mtp_pp2_matches_single_untied(tmp_path, monkeypatch):
    out = tmp_path / "mtp.json"
    monkeypatch.setenv("MTP_TEST_OUT", str(out))
    ref = _mtp_single_ref(True)
    spawn_dist(_mtp_pp2_case, 2, True)
    pp = json.load(open(out))
    for a, b in zip(ref, pp):
        assert abs(a - b) < 3e-4, (ref, pp)


def test_mtp_pp2_matches_single_tied(tmp_path, monkeypatch):
    out = tmp_path / "mtpt.json"
    monkeypatch.setenv("MTP_TEST_OUT", str(out))
    ref = _mtp_single_ref(False)
    spawn_dist(_mtp_pp2_case, 2, False)
    pp = json.load(open(out))
    for a, b in zip(ref, pp):
        assert abs(a - b) < 3e-4, (ref, pp)
