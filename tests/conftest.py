import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run on the GPU box)")


def pytest_collection_modifyitems(config, items):
    import torch
    if not torch.cuda.is_available():
        skip_gpu = pytest.mark.skip(reason="no GPU in this container")
        for item in items:
            if "gpu" in item.keywords:
                item.add_marker(skip_gpu)
