from megatron_amd.post_training.distillation import DistillationLoss  # noqa: F401
from megatron_amd.post_training.quant_config import (  # noqa: F401
    QuantRecipe,
    QuantRecipeConfig,
    resolve_layer_recipes,
)
