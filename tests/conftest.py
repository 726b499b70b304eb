import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD MI355X (gfx950) GPU"
    )
