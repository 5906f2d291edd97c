import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        'markers', 'gpu: needs a ROCm GPU (run on an MI355X box)')
    config.addinivalue_line(
        'markers', 'slow: long-running CPU test (minutes)')


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason='no GPU in this environment')
    for item in items:
        if 'gpu' in item.keywords:
            item.add_marker(skip)
