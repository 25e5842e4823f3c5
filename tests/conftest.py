import os
import sys

import pytest

# make the in-tree package importable regardless of cwd
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# Load torch + our HIP extension in the right order BEFORE any test module
# imports: torch-rocm bundles its own libamdhip64 and whichever HIP runtime
# initializes second in a process sees no devices (demodel_amd/gpu's import
# order comment).  demodel_amd.gpu imports torch first, then _hip.
try:
    import demodel_amd.gpu as _early_gpu  # noqa: F401
except Exception:
    pass


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs an AMD GPU (MI355X); skipped on CPU-only CI")


def pytest_collection_modifyitems(config, items):
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session", autouse=True)
def _build_native():
    from demodel_amd.build import build_native

    build_native()
    yield
