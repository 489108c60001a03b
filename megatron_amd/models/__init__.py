from megatron_amd.models.gpt import GPTModel  # noqa: F401
