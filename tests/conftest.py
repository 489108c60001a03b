import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU (run via gpurun)")


@pytest.fixture(autouse=True)
def _reset_global_torch_flags():
    yield
    import torch

    # tests that run the pretrain app with --deterministic-mode must not leak
    # the global determinism flag (it NaN-fills torch.empty) into later tests
    torch.use_deterministic_algorithms(False)


@pytest.fixture(autouse=True)
def _reset_parallel_state():
    yield
    # tests that initialize the single-process grid must not leak it
    from megatron_amd.parallel import grid as G

    import torch.distributed as dist

    if not dist.is_initialized():
        G.destroy_model_parallel()
