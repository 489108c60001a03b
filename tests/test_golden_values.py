"""Golden-value functional test (reference analog: tests/functional_tests
golden_values_*.json + check_golden_values.py): a deterministic tiny config's
loss trajectory must match the checked-in goldens."""

import json
import os

import megatron_amd.training.pretrain as P

GOLDEN = os.path.join(os.path.dirname(__file__), "golden", "tiny_gpt.json")


def test_tiny_gpt_loss_curve_matches_golden():
    golden = json.load(open(GOLDEN))
    losses = []

    def provider(config, pre_process=True, post_process=True, vp_stage=None):
        from megatron_amd.models.gpt import GPTModel

        return GPTModel(config, pre_process=pre_process, post_process=post_process)

    orig = P.train_step

    def wrapped(*a, **k):
        r = orig(*a, **k)
        losses.append(r["lm_loss"])
        return r

    P.train_step = wrapped
    try:
        P.pretrain(provider, [
            "--num-layers", "2", "--hidden-size", "64", "--num-attention-heads", "4",
            "--num-query-groups", "2", "--ffn-hidden-size", "128", "--seq-length", "64",
            "--micro-batch-size", "2", "--global-batch-size", "4", "--vocab-size", "256",
            "--mock-data", "--train-iters", "10", "--log-interval", "0", "--seed", "42",
            "--deterministic-mode",
        ])
    finally:
        P.train_step = orig
    assert len(losses) == len(golden["lm_loss"])
    for i, (got, want) in enumerate(zip(losses, golden["lm_loss"])):
        assert abs(got - want) < 2e-3, f"step {i}: {got} vs golden {want}"


def _run_config(extra_args, provider=None):
    losses = []

    if provider is None:
        def provider(config, pre_process=True, post_process=True, vp_stage=None):
            from megatron_amd.models.gpt import GPTModel

            return GPTModel(config, pre_process=pre_process, post_process=post_process)

    orig = P.train_step

    def wrapped(*a, **k):
        r = orig(*a, **k)
        losses.append(r["lm_loss"])
        return r

    P.train_step = wrapped
    try:
        P.pretrain(provider, [
            "--num-layers", "2", "--hidden-size", "64", "--num-attention-heads", "4",
            "--num-query-groups", "2", "--ffn-hidden-size", "128", "--seq-length", "64",
            "--micro-batch-size", "2", "--global-batch-size", "4", "--vocab-size", "256",
            "--mock-data", "--train-iters", "8", "--log-interval", "0", "--seed", "42",
            "--deterministic-mode",
        ] + extra_args)
    finally:
        P.train_step = orig
    return losses


def _check_or_record(name, losses):
    path = os.path.join(os.path.dirname(__file__), "golden", name)
    if os.environ.get("UPDATE_GOLDENS") == "1" or not os.path.exists(path):
        assert os.environ.get("UPDATE_GOLDENS") == "1", f"missing golden {name}"
        json.dump({"lm_loss": losses}, open(path, "w"), indent=1)
        return
    golden = json.load(open(path))
    assert len(losses) == len(golden["lm_loss"])
    for i, (got, want) in enumerate(zip(losses, golden["lm_loss"])):
        assert abs(got - want) < 2e-3, f"step {i}: {got} vs golden {want}"


def test_tiny_moe_loss_curve_matches_golden():
    losses = _run_config(["--num-experts", "4", "--moe-router-topk", "2",
                          "--moe-aux-loss-coeff", "0.01"])
    _check_or_record("tiny_moe.json", losses)


def test_tiny_mamba_loss_curve_matches_golden():
    def provider(config, pre_process=True, post_process=True, vp_stage=None):
        from megatron_amd.models.mamba import MambaModel

        return MambaModel(config, pre_process=pre_process, post_process=post_process)

    losses = _run_config(["--hybrid-override-pattern", "M*", "--mamba-num-groups", "2",
                          "--mamba-head-dim", "32"], provider=provider)
    _check_or_record("tiny_mamba.json", losses)


# --- round-2: widened functional matrix (reference
# tests/functional_tests config-per-case pattern) ----------------------------


def test_tiny_gpt_distopt_golden():
    losses = _run_config(["--use-distributed-optimizer", "--overlap-param-gather"])
    _check_or_record("tiny_distopt.json", losses)


def test_tiny_moe_capacity_drop_golden():
    losses = _run_config(["--num-experts", "4", "--moe-router-topk", "2",
                          "--moe-aux-loss-coeff", "0.01",
                          "--moe-expert-capacity-factor", "1.25",
                          "--moe-token-drop-policy", "probs"])
    _check_or_record("tiny_moe_capacity.json", losses)


def test_tiny_moe_allgather_dispatcher_golden():
    losses = _run_config(["--num-experts", "4", "--moe-router-topk", "2",
                          "--moe-token-dispatcher-type", "allgather"])
    _check_or_record("tiny_moe_allgather.json", losses)


def test_tiny_gpt_label_smoothing_golden():
    losses = _run_config(["--label-smoothing", "0.1"])
    _check_or_record("tiny_label_smoothing.json", losses)


def test_tiny_gpt_recompute_golden():
    # full recompute must reproduce the plain golden exactly (same math)
    golden = json.load(open(GOLDEN))
    losses = _run_config(["--recompute-granularity", "full"])
    for got, want in zip(losses, golden["lm_loss"]):
        assert abs(got - want) < 2e-3


def test_tiny_gpt_mtp_golden():
    losses = _run_config(["--mtp-num-layers", "1"])
    _check_or_record("tiny_mtp.json", losses)


def _dist_golden_case(rank, world, extra, name):
    losses = _run_config(extra)
    from megatron_amd.parallel import grid as G

    if G.get_grid().is_pipeline_last_stage(ignore_virtual=True) and (
            not torch.distributed.is_initialized() or rank == world - 1):
        json.dump({"lm_loss": losses}, open(os.environ["GOLDEN_TMP"], "w"))


import torch

from tests.utils import spawn_dist


def test_tiny_gpt_tp2_golden(tmp_path, monkeypatch):
    """Recorded tp2 curve (sharded init draws differ from single-process
    init, so this is a self-consistency golden, not an equality check —
    TP==single equality is covered weight-for-weight by test_tp_layers)."""
    out = tmp_path / "tp2.json"
    monkeypatch.setenv("GOLDEN_TMP", str(out))
    spawn_dist(_dist_golden_case, 2,
               ["--tensor-model-parallel-size", "2", "--train-iters", "8"], "tp2")
    _check_or_record("tiny_tp2.json", json.load(open(out))["lm_loss"])


def test_tiny_gpt_pp2_golden(tmp_path, monkeypatch):
    out = tmp_path / "pp2.json"
    monkeypatch.setenv("GOLDEN_TMP", str(out))
    spawn_dist(_dist_golden_case, 2,
               ["--pipeline-model-parallel-size", "2", "--train-iters", "8"], "pp2")
    _check_or_record("tiny_pp2.json", json.load(open(out))["lm_loss"])


def test_tiny_bert_loss_curve_matches_golden():
    def provider(config, pre_process=True, post_process=True, vp_stage=None):
        from megatron_amd.models.bert import BertModel

        return BertModel(config, pre_process=pre_process, post_process=post_process)

    losses = _run_config(["--position-embedding-type", "learned",
                          "--normalization", "layernorm", "--activation", "gelu"],
                         provider=provider)
    _check_or_record("tiny_bert.json", losses)


def test_tiny_t5_loss_curve_matches_golden():
    import pretrain_t5 as T5E

    losses = []
    orig = P.train_step

    def wrapped(*a, **k):
        r = orig(*a, **k)
        losses.append(r["lm_loss"])
        return r

    P.train_step = wrapped
    try:
        P.pretrain(T5E.model_provider, [
            "--num-layers", "2", "--hidden-size", "64", "--num-attention-heads", "4",
            "--num-query-groups", "2", "--ffn-hidden-size", "128", "--seq-length", "64",
            "--micro-batch-size", "2", "--global-batch-size", "4", "--vocab-size", "256",
            "--mock-data", "--train-iters", "8", "--log-interval", "0", "--seed", "42",
            "--deterministic-mode",
        ], forward_step_builder=T5E.forward_step_builder)
    finally:
        P.train_step = orig
    _check_or_record("tiny_t5.json", losses)


def test_tiny_gpt_fsdp_tracks_ddp_golden():
    """--use-fsdp (ZeRO-3 path) must follow the DDP golden trajectory
    (same math; torch-AdamW vs fused-AdamW drift only)."""
    golden = json.load(open(GOLDEN))
    losses = _run_config(["--use-fsdp", "--train-iters", "8"])
    for i, (got, want) in enumerate(zip(losses, golden["lm_loss"])):
        assert abs(got - want) < 5e-2, (i, losses, golden["lm_loss"])


def test_tiny_gpt_packed_sequences_matches_golden():
    """--packed-sequences with one document per row is mathematically the
    unpacked run (block-diagonal attention == per-row attention, RoPE
    restarts per row) — must land on the plain golden curve."""
    golden = json.load(open(GOLDEN))
    losses = _run_config(["--packed-sequences", "--train-iters", "10"])
    for i, (got, want) in enumerate(zip(losses, golden["lm_loss"])):
        assert abs(got - want) < 2e-3, (i, losses, golden["lm_loss"])


def test_tiny_gpt_sgd_golden():
    losses = _run_config(["--optimizer", "sgd", "--lr", "0.05"])
    _check_or_record("tiny_sgd.json", losses)


def test_tiny_moe_pattern_golden():
    losses = _run_config(["--num-experts", "4", "--moe-router-topk", "2",
                          "--moe-aux-loss-coeff", "0.01",
                          "--moe-layer-freq", "[0,1]"])
    _check_or_record("tiny_moe_pattern.json", losses)


def test_tiny_moe_recompute_modules_golden():
    """MoE module recompute must track the plain MoE golden exactly."""
    losses = _run_config(["--num-experts", "4", "--moe-router-topk", "2",
                          "--moe-aux-loss-coeff", "0.01",
                          "--recompute-granularity", "selective",
                          "--recompute-modules", "core_attn", "moe"])
    _check_or_record("tiny_moe.json", losses)


def test_tiny_t5_relative_bias_golden():
    import pretrain_t5 as T5E

    losses = []
    orig = P.train_step

    def wrapped(*a, **k):
        r = orig(*a, **k)
        losses.append(r["lm_loss"])
        return r

    P.train_step = wrapped
    try:
        P.pretrain(T5E.model_provider, [
            "--num-layers", "2", "--hidden-size", "64", "--num-attention-heads", "4",
            "--num-query-groups", "2", "--ffn-hidden-size", "128", "--seq-length", "64",
            "--micro-batch-size", "2", "--global-batch-size", "4", "--vocab-size", "256",
            "--mock-data", "--train-iters", "8", "--log-interval", "0", "--seed", "42",
            "--deterministic-mode", "--position-embedding-type", "relative",
        ], forward_step_builder=T5E.forward_step_builder)
    finally:
        P.train_step = orig
    _check_or_record("tiny_t5_relative.json", losses)


def test_tiny_bert_nsp_golden():
    """pretrain_bert's joint MLM+NSP recipe (binary head active) pinned."""
    import pretrain_bert as B

    losses = []
    orig = P.train_step

    def wrapped(*a, **k):
        r = orig(*a, **k)
        losses.append(r["lm_loss"])
        return r

    P.train_step = wrapped
    try:
        P.pretrain(B.model_provider, [
            "--num-layers", "2", "--hidden-size", "64", "--num-attention-heads", "4",
            "--num-query-groups", "2", "--ffn-hidden-size", "128", "--seq-length", "64",
            "--micro-batch-size", "2", "--global-batch-size", "4", "--vocab-size", "256",
            "--mock-data", "--train-iters", "8", "--log-interval", "0", "--seed", "42",
            "--deterministic-mode", "--position-embedding-type", "learned",
            "--normalization", "layernorm", "--activation", "gelu",
        ], forward_step_builder=B.forward_step_builder)
    finally:
        P.train_step = orig
    _check_or_record("tiny_bert_nsp.json", losses)
