from megatron_amd.training.training import train_step  # noqa: F401
from megatron_amd.training.flops import num_floating_point_operations  # noqa: F401
