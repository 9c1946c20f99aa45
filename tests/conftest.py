import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an AMD GPU (run via gpurun)")
