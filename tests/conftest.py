import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X (run via gpurun / round-end driver)"
    )
    config.addinivalue_line(
        "markers", "treekernel: dark-shipped tree-mask kernel validation "
        "(run explicitly with -m treekernel on a GPU box)"
    )


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)
