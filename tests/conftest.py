import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a ROCm GPU (run on MI355X box)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(autouse=True)
def reset_singletons():
    """Fresh Context + embedding registry per test."""
    yield
    import openembedding_amd.context as ctx_mod
    import openembedding_amd.torch as api
    if ctx_mod._context is not None:
        try:
            ctx_mod._context.finalize()
        except Exception:
            pass
        ctx_mod._context = None
    api._tracked.clear()
