import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import random as _random

_PORT_COUNTER = [_random.randrange(20000, 60000, 64)]


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)")


@pytest.fixture
def port_block():
    """A fresh block of 64 loopback ports per test."""
    base = _PORT_COUNTER[0]
    _PORT_COUNTER[0] += 64
    return base
