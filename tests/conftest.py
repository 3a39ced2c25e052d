import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run on MI355X via gpurun)"
    )


def pytest_collection_modifyitems(config, items):
    try:
        import pumiumtally_amd as pt

        has_gpu = pt.have_gpu()
    except Exception:
        has_gpu = False
    skip = pytest.mark.skip(reason="no HIP device available")
    for item in items:
        if "gpu" in item.keywords and not has_gpu:
            item.add_marker(skip)
