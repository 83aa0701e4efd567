import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (MI355X); skipped on CPU-only hosts"
    )


def pytest_collection_modifyitems(config, items):
    try:
        import trtlab_amd

        has_gpu = trtlab_amd.has_gpu()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU present")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
