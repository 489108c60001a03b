"""Optimizer stack.

Capability analog of reference megatron/core/optimizer/ (9,751 LoC):
get_megatron_optimizer (__init__.py:991), MixedPrecisionOptimizer /
Float16OptimizerWithFloat16Params (optimizer.py:654/:964), FP32Optimizer
(:1232), ChainedOptimizer (:1419), DistributedOptimizer
(distrib_optimizer.py:113), OptimizerParamScheduler.
"""

from __future__ import annotations

from typing import List, Optional

from megatron_amd.config import OptimizerConfig
from megatron_amd.optimizer.optimizer import (  # noqa: F401
    ChainedOptimizer,
    FP32Optimizer,
    MixedPrecisionOptimizer,
)
from megatron_amd.optimizer.dist_optimizer import DistributedOptimizer  # noqa: F401
from megatron_amd.optimizer.scheduler import OptimizerParamScheduler  # noqa: F401


def _is_embedding_or_output(p) -> bool:
    return getattr(p, "is_embedding_or_output_parameter", False)


def _subclass_with(opt_cls, flt, lr_ratio: float):
    """Specialize an optimizer class with a param filter + lr multiplier
    (decoupled-lr groups, reference optimizer/__init__.py lr_mult)."""

    class _Grouped(opt_cls):
        pass

    _Grouped.__name__ = opt_cls.__name__
    _Grouped.param_filter = staticmethod(flt) if flt is not None else None
    _Grouped.lr_ratio = lr_ratio
    return _Grouped


def get_optimizer(opt_config: OptimizerConfig, model_chunks: List) -> "ChainedOptimizer":
    """Build the optimizer for a list of DDP-wrapped model chunks
    (reference optimizer/__init__.py:991 get_megatron_optimizer).  With
    `decoupled_lr` set, embedding/output params go to a second chained
    sub-optimizer whose lr tracks the scheduler at the decoupled ratio."""
    if opt_config.decoupled_lr is not None:
        assert not opt_config.use_distributed_optimizer, \
            "decoupled-lr groups not supported with the distributed optimizer yet"
        base_cls = (FP32Optimizer if not (opt_config.bf16 or opt_config.fp16)
                    else MixedPrecisionOptimizer)
        if opt_config.optimizer == "muon":
            from megatron_amd.optimizer.muon import MuonOptimizer

            base_cls = MuonOptimizer
        ratio = opt_config.decoupled_lr / opt_config.lr
        main = _subclass_with(base_cls, lambda p: not _is_embedding_or_output(p), 1.0)
        emb = _subclass_with(base_cls, _is_embedding_or_output, ratio)
        return ChainedOptimizer([main(opt_config, model_chunks), emb(opt_config, model_chunks)])
    if opt_config.optimizer == "muon":
        from megatron_amd.optimizer.muon import MuonOptimizer

        opt = MuonOptimizer(opt_config, model_chunks)
    elif opt_config.optimizer_cpu_offload:
        from megatron_amd.optimizer.cpu_offload import CPUOffloadOptimizer

        opt = CPUOffloadOptimizer(opt_config, model_chunks)
    elif opt_config.use_distributed_optimizer:
        opt = DistributedOptimizer(opt_config, model_chunks)
    elif opt_config.bf16 or opt_config.fp16:
        opt = MixedPrecisionOptimizer(opt_config, model_chunks)
    else:
        opt = FP32Optimizer(opt_config, model_chunks)
    return ChainedOptimizer([opt])
