import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests that require an MI355X GPU")
