import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test needs an MI355X GPU")
    config.addinivalue_line("markers", "world2: spawns 2 processes")
    config.addinivalue_line("markers", "world4: spawns 4 processes")
    config.addinivalue_line("markers", "world8: spawns 8 processes")
    config.addinivalue_line("markers", "long: long-running test")
