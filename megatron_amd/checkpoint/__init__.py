from megatron_amd.checkpoint.sharded import ShardedTensor, load, save  # noqa: F401
from megatron_amd.checkpoint.checkpointing import load_checkpoint, save_checkpoint  # noqa: F401
