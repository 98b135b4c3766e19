import pytest


def pytest_configure(config):
    config.addinivalue_line(
        'markers', 'gpu: requires an MI355X GPU (run on a gpurun box)')
