import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires a HIP device (MI355X); run on the GPU box")


@pytest.fixture(scope="session")
def oracle():
    from oracle import Oracle
    return Oracle()


@pytest.fixture(scope="session")
def reference():
    """The reference's own compiled cores; only present where oracle/ref was
    built (dev container or a snapshot carrying oracle/_ref)."""
    from oracle import Reference
    path = os.path.join(REPO_ROOT, "oracle", "_ref", "libref.so")
    if not os.path.exists(path):
        pytest.skip("oracle/_ref/libref.so not built")
    return Reference()
