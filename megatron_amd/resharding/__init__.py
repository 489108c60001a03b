from megatron_amd.resharding.refit import (  # noqa: F401
    assemble_global_tensors,
    refit_model,
)
