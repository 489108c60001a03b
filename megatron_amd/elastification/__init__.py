from megatron_amd.elastification.elastic import (  # noqa: F401
    ElasticLinear,
    elastic_memory_profile,
    set_active_width,
)
