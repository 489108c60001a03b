from megatron_amd.datasets.mock import MockGPTDataIterator  # noqa: F401
