import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests requiring an MI355X GPU")
