import pytest  # noqa: F401


def pytest_configure(config):
    config.addinivalue_line(
        'markers', 'gpu: needs a ROCm GPU (run on an MI355X box)')
