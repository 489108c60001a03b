from megatron_amd.utils.timers import Timers  # noqa: F401
