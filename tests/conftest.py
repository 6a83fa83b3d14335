import sys
from pathlib import Path

import pytest

try:  # deterministic property tests: the driver's round checks must not be
    # able to hit a fresh random counterexample that local runs never saw
    from hypothesis import settings

    settings.register_profile("ci", derandomize=True, deadline=None)
    settings.load_profile("ci")
except ImportError:  # pragma: no cover
    pass

REPO_ROOT = Path(__file__).resolve().parent.parent
if str(REPO_ROOT) not in sys.path:
    sys.path.insert(0, str(REPO_ROOT))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require an MI355X GPU (run via gpurun)"
    )
    # hang protection for the driver's round checks (pytest-timeout)
    if config.getoption("--timeout", None) in (None, 0):
        config.option.timeout = 600


def pytest_collection_modifyitems(config, items):
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)
