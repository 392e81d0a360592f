import pytest
import torch

try:
    # deterministic property-based runs: the generative tests were
    # additionally exercised under multiple explicit seeds during
    # development; unseeded exploration in CI would only add flake risk
    from hypothesis import settings as _hyp_settings
    _hyp_settings.register_profile("deterministic", derandomize=True)
    _hyp_settings.load_profile("deterministic")
except ImportError:  # pragma: no cover
    pass


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires a ROCm GPU (run on the MI355X box)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
