import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        'markers', 'gpu: requires an AMD GPU (run with -m gpu on an MI355X box)')
