from megatron_amd.transformer.block import TransformerBlock, TransformerLayer  # noqa: F401
