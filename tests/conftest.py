import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def device():
    return torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")


try:  # deeper fuzzing on demand: pytest --hypothesis-profile=thorough
    from hypothesis import settings as _hyp_settings

    _hyp_settings.register_profile("thorough", max_examples=300,
                                   deadline=None)
except ImportError:  # pragma: no cover
    pass
