import sys
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that must run on the MI355X box (full-cluster "
        "integration tiers that need a dedicated machine)")
