from megatron_amd.models.gpt import GPTModel  # noqa: F401
from megatron_amd.models.mamba import MambaModel  # noqa: F401
