"""State-space-model (Mamba2/SSD) family.

Capability analog of reference megatron/core/ssm/ (mamba_mixer.py:144,
mamba_block.py, hybrid layer allocation in
megatron/core/models/hybrid/hybrid_layer_allocation.py).  The reference
delegates the hot path to external CUDA/Triton packages (causal_conv1d,
mamba_ssm chunk-scan); here the SSD chunked scan is expressed as batched
matmuls + exponent/cumsum elementwise work so it runs on MFMA via rocBLAS
and stays fusable, with fp32 state arithmetic throughout.
"""

from megatron_amd.ssm.hybrid_allocation import Symbols, allocate_layers, pattern_from_ratios
from megatron_amd.ssm.mamba_mixer import MambaMixer
from megatron_amd.ssm.ssd import ssd_chunked_scan, ssd_step

__all__ = [
    "Symbols",
    "allocate_layers",
    "pattern_from_ratios",
    "MambaMixer",
    "ssd_chunked_scan",
    "ssd_step",
]
