import os
import sys

import pytest

_TESTS = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, _TESTS)
sys.path.insert(0, os.path.dirname(_TESTS))  # repo root, for risingwave_amd


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X GPU (run via gpurun)")
