import random

import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU")


@pytest.fixture(autouse=True)
def _seed():
    torch.manual_seed(0)
    random.seed(0)


@pytest.fixture
def device(request):
    if request.node.get_closest_marker("gpu"):
        assert torch.cuda.is_available(), "gpu-marked test without a GPU"
        return torch.device("cuda:0")
    return torch.device("cpu")
