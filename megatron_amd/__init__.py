"""megatron_amd — an MI355X-native Megatron-Core-class LLM training framework.

Built from scratch for one 8-GPU AMD MI355X node (CDNA4 / gfx950):
PyTorch-ROCm on top, hand-written HIP kernels (MFMA + LDS) for the fused hot
ops, hipBLASLt/rocBLAS for plain GEMMs, RCCL collectives over the
fully-connected 7-link xGMI fabric.

Capability contract mirrors NVIDIA/Megatron-LM (PARITY.md maps
every inventory row to code + tests);
the implementation is MI355X-first, not a port.
"""

__version__ = "0.2.0"

from megatron_amd.config import TransformerConfig, OptimizerConfig, DDPConfig  # noqa: F401
from megatron_amd.parallel import grid as parallel_grid  # noqa: F401
