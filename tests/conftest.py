import pytest


def pytest_configure(config: pytest.Config) -> None:
    config.addinivalue_line("markers", "gpu: requires an AMD GPU (MI355X) to run")
    config.addinivalue_line("markers", "slow: long-running test")


def pytest_collection_modifyitems(config: pytest.Config, items: list[pytest.Item]) -> None:
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip_gpu = pytest.mark.skip(reason="no AMD GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)
