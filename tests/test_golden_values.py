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
