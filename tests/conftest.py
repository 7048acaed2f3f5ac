import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an AMD MI355X GPU (run via gpurun)")
    config.addinivalue_line("markers", "slow: long-running test")
