import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

REFERENCE = "/root/reference"


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests that require an MI355X GPU")
    config.addinivalue_line("markers", "slow: long-running CPU tests")


def have_reference() -> bool:
    return os.path.isdir(os.path.join(REFERENCE, "verification"))


requires_reference = pytest.mark.skipif(
    not have_reference(), reason="reference verification decks not mounted")
