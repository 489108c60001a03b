"""Training application tests (reference: megatron/training/training.py pretrain loop,
arguments.py parse_and_validate_args) — CPU, single process, mock data."""

import os

import pytest
import torch

from megatron_amd.training.arguments import parse_and_validate_args
from megatron_amd.training.pretrain import pretrain

TINY = [
    "--num-layers", "2", "--hidden-size", "64", "--num-attention-heads", "4",
    "--num-query-groups", "2", "--ffn-hidden-size", "128", "--seq-length", "32",
    "--micro-batch-size", "2", "--global-batch-size", "4", "--vocab-size", "128",
    "--mock-data", "--log-interval", "0",
]


def model_provider(config, pre_process=True, post_process=True, vp_stage=None):
    from megatron_amd.models.gpt import GPTModel

    return GPTModel(config, pre_process=pre_process, post_process=post_process, vp_stage=vp_stage)


def test_parse_and_validate():
    args = parse_and_validate_args(TINY + ["--train-iters", "5"])
    assert args.num_microbatches == 2
    assert args.data_parallel_size == 1
    assert args.global_batch_size == 4


def test_parse_rejects_bad_gbs():
    with pytest.raises(AssertionError):
        parse_and_validate_args(TINY + ["--global-batch-size", "3"])


def test_pretrain_runs_and_resumes(tmp_path):
    ckpt = str(tmp_path / "ckpt")
    it = pretrain(model_provider, TINY + ["--train-iters", "3", "--save", ckpt, "--seed", "7"])
    assert it == 3
    assert os.path.exists(os.path.join(ckpt, "latest_checkpointed_iteration.txt"))
    # resume continues from saved iteration and finishes the remaining steps
    it2 = pretrain(model_provider, TINY + ["--train-iters", "5", "--save", ckpt,
                                           "--load", ckpt, "--seed", "7"])
    assert it2 == 5


def test_yaml_config_overlay(tmp_path):
    from megatron_amd.training.arguments import parse_and_validate_args

    cfg = tmp_path / "run.yaml"
    cfg.write_text("num-layers: 6\nhidden_size: 256\nlr: 0.0005\n")
    args = parse_and_validate_args(["--yaml-cfg", str(cfg), "--hidden-size", "128",
                                    "--num-attention-heads", "4"])
    assert args.num_layers == 6          # from yaml
    assert args.hidden_size == 128       # CLI beats yaml
    assert args.lr == 0.0005             # underscore/dash both accepted

    bad = tmp_path / "bad.yaml"
    bad.write_text("not-a-flag: 1\n")
    import pytest as _pytest

    with _pytest.raises(ValueError, match="unknown yaml"):
        parse_and_validate_args(["--yaml-cfg", str(bad)])


def test_pretrain_t5_entry(tmp_path):
    import pretrain_t5

    it = pretrain(pretrain_t5.model_provider,
                  TINY + ["--train-iters", "2", "--seed", "3"],
                  forward_step_builder=pretrain_t5.forward_step_builder)
    assert it == 2


def test_pretrain_muon_and_fault_injection(tmp_path):
    from megatron_amd.utils.fault_injection import InjectedFault

    # muon trains through the CLI path
    it = pretrain(model_provider, TINY + ["--train-iters", "2", "--optimizer", "muon"])
    assert it == 2
    # an injected crash surfaces at the configured iteration
    with pytest.raises(InjectedFault):
        pretrain(model_provider, TINY + ["--train-iters", "5",
                                         "--fault-injection-type", "crash",
                                         "--fault-injection-iteration", "1"])


def test_microbatch_rampup_calculator():
    from megatron_amd.training.microbatches import MicrobatchCalculator

    calc = MicrobatchCalculator(32, 2, 1, rampup=(8, 8, 96))
    # 3 increments (8->16->24->32) over 96 samples: one every 32 samples
    assert calc.get(0) == (8, 4)
    assert calc.get(31) == (8, 4)
    assert calc.get(32) == (16, 8)
    assert calc.get(64) == (24, 12)
    assert calc.get(96) == (32, 16)
    assert calc.get(10_000) == (32, 16)
    # constant mode
    assert MicrobatchCalculator(32, 2, 4).get(999) == (32, 4)
    # invalid: increment not divisible by mbs*dp
    import pytest as _pytest

    with _pytest.raises(AssertionError):
        MicrobatchCalculator(32, 2, 2, rampup=(8, 6, 96))


def test_pretrain_with_rampup():
    it = pretrain(model_provider, TINY + ["--train-iters", "4",
                                          "--rampup-batch-size", "2", "2", "8"])
    assert it == 4


def test_run_config_dumped(tmp_path):
    ckpt = str(tmp_path / "ckpt")
    pretrain(model_provider, TINY + ["--train-iters", "1", "--save", ckpt])
    import json

    cfgd = json.load(open(os.path.join(ckpt, "run_config.json")))
    assert cfgd["args"]["num_layers"] == "2"
    assert "hidden_size" in cfgd["transformer_config"]


def test_pretrain_vlm_entry():
    import pretrain_vlm

    it = pretrain(pretrain_vlm.model_provider,
                  TINY + ["--train-iters", "2", "--seed", "5",
                          "--position-embedding-type", "rope"],
                  forward_step_builder=pretrain_vlm.forward_step_builder)
    assert it == 2


def test_pretrain_extra_log_flags_and_finetune(tmp_path, capsys):
    """--log-params-norm / --log-num-zeros-in-grad lines render; --finetune
    restarts iteration 0 from saved weights."""
    ckpt = str(tmp_path / "ck")
    it = pretrain(model_provider, TINY + ["--train-iters", "2", "--save", ckpt,
                                          "--seed", "3", "--log-interval", "1",
                                          "--log-params-norm", "--log-num-zeros-in-grad"])
    assert it == 2
    out = capsys.readouterr().out
    assert "params norm" in out and "zeros in grad" in out
    it2 = pretrain(model_provider, TINY + ["--train-iters", "2", "--load", ckpt,
                                           "--seed", "3", "--finetune"])
    assert it2 == 2  # ran 2 fresh iterations, not resumed-at-2-no-op
    out = capsys.readouterr().out
    assert "loaded checkpoint at iteration 0" in out  # weights-only, reset
