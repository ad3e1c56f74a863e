import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU (run with -m gpu)")


def pytest_collection_modifyitems(config, items):
    # Skip gpu-marked tests automatically when no GPU is present, regardless of -m selection.
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="module")
def temp_dir(tmp_path_factory):
    return tmp_path_factory.mktemp("perceiver_amd")
