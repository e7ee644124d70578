import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X (run with -m gpu on a GPU box)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def local_ctx():
    """Single-process DistContext-alike for pipeline/unit tests."""
    class _Ctx:
        rank = 0
        world_size = 1
        local_rank = 0
        distributed = False
        is_main = True
        device = torch.device("cpu")

        def barrier(self):
            pass

        def broadcast_module(self, m):
            pass

        def all_reduce_(self, t):
            pass

        def all_reduce_async(self, t):
            return None

    return _Ctx()
