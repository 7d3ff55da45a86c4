import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an AMD GPU (MI355X); run via gpurun")


def pytest_collection_modifyitems(config, items):
    try:
        from stencil_amd import _C

        has_gpu = _C.device_count() > 0
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no HIP device")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
