import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pytest

try:
    # deterministic property-test runs: the same examples every time, so a
    # CI/driver run can never trip over a fresh random falsifying example
    from hypothesis import settings as _hyp_settings

    _hyp_settings.register_profile("ci", derandomize=True, deadline=None)
    _hyp_settings.load_profile("ci")
except ImportError:
    pass


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run on gpurun box)")


def pytest_collection_modifyitems(config, items):
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture()
def fresh_container():
    """An isolated CPU ServiceContainer with mock engines."""
    from sentio_amd.config import Settings
    from sentio_amd.serving.container import ServiceContainer

    s = Settings()
    s.mock_compute = True
    s.device = "cpu"
    s.use_reranker = True
    s.use_verifier = False
    return ServiceContainer(s)
