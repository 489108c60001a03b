from megatron_amd.pipeline.schedules import get_forward_backward_func  # noqa: F401
