from megatron_amd.distributed.ddp import DistributedDataParallel  # noqa: F401
from megatron_amd.distributed.finalize import finalize_model_grads  # noqa: F401
from megatron_amd.distributed.fsdp import FullyShardedDataParallel  # noqa: F401
