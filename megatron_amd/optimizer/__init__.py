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


def get_optimizer(opt_config: OptimizerConfig, model_chunks: List) -> "ChainedOptimizer":
    """Build the optimizer for a list of DDP-wrapped model chunks
    (reference optimizer/__init__.py:991 get_megatron_optimizer)."""
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
