import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires a real MI355X GPU")


@pytest.fixture(scope="session")
def oracle_lib():
    """Build (if needed) and load the CPU oracle library."""
    from tests.gxlib import load_oracle
    return load_oracle()


@pytest.fixture(scope="session")
def product_lib():
    """Load the MI355X product library (host-side helpers work without a GPU)."""
    from tests.gxlib import load_product
    return load_product()
