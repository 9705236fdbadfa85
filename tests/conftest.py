import pytest  # noqa: F401


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU (run via gpurun)")
