from megatron_amd.inference.contexts import (
    DynamicInferenceContext,
    KVBlockAllocator,
    StaticInferenceContext,
)
from megatron_amd.inference.engine import (
    DynamicInferenceEngine,
    GenerationResult,
    StaticInferenceEngine,
)
from megatron_amd.inference.sampling import SamplingParams
