import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X GPU (run via gpurun)")


@pytest.fixture(autouse=True)
def _fresh_runner_pool():
    """Isolate tests from the process-global warm runner pool."""
    from gpuschedule_amd.exec.trainer import RUNNER_POOL
    RUNNER_POOL.flush()
    yield
    RUNNER_POOL.flush()


def pytest_collection_modifyitems(config, items):
    try:
        import torch
        have_gpu = torch.cuda.is_available()
    except Exception:
        have_gpu = False
    if have_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
