from megatron_amd.parallel import grid  # noqa: F401
