from megatron_amd.resharding.refit import (  # noqa: F401
    assemble_global_tensors,
    execute_refit_plan,
    plan_refit,
    refit_model,
    refit_model_planned,
)
