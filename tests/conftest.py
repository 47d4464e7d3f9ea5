import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: marks tests that require a real MI355X GPU"
    )


@pytest.fixture()
def mem_db():
    """A shared in-memory RW/RO sqlite pair."""
    from gpud_amd.pkg.sqlite_util import open_memory_pair

    rw, ro = open_memory_pair()
    yield rw, ro
    rw.close()
    ro.close()
