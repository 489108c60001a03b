from megatron_amd.moe.moe_layer import MoELayer  # noqa: F401
