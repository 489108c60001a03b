from megatron_amd.models.gpt import GPTModel  # noqa: F401
from megatron_amd.models.mamba import MambaModel  # noqa: F401


def __getattr__(name):  # lazy: bert/t5/llava/audio pull heavier deps
    if name == "BertModel":
        from megatron_amd.models.bert import BertModel

        return BertModel
    if name == "T5Model":
        from megatron_amd.models.t5 import T5Model

        return T5Model
    if name == "LLaVAModel":
        from megatron_amd.models.llava import LLaVAModel

        return LLaVAModel
    if name == "AudioLanguageModel":
        from megatron_amd.models.audio import AudioLanguageModel

        return AudioLanguageModel
    raise AttributeError(name)
