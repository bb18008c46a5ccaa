import os
import sys

import pytest

# repo-root imports (torchft_amd is not pip-installed)
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config: pytest.Config) -> None:
    config.addinivalue_line("markers", "gpu: requires an MI355X (HIP) device")


def pytest_collection_modifyitems(
    config: pytest.Config, items: list[pytest.Item]
) -> None:
    import torch

    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no HIP device")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
