import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires a real MI355X GPU (run via gpurun)")
