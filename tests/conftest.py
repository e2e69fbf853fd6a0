import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line('markers', 'gpu: tests that need an MI355X GPU')


@pytest.fixture(autouse=True)
def _deterministic_seed():
    torch.manual_seed(0)
