import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu)"
    )


def pytest_collection_modifyitems(config, items):
    """Skip gpu-marked tests automatically when no GPU is present."""
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


@pytest.fixture(scope="session")
def xlow_voice_path(tmp_path_factory):
    from sonata_amd.models import create_random_voice

    d = tmp_path_factory.mktemp("voice")
    return create_random_voice(str(d), "xlow_test", quality="x_low")


@pytest.fixture(scope="session")
def xlow_voice(xlow_voice_path):
    from sonata_amd.models import load_voice

    return load_voice(xlow_voice_path)
