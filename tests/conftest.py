import os
import sys
from pathlib import Path

import pytest

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires an MI355X GPU (run with -m gpu on a GPU box)")


@pytest.fixture
def db():
    """In-memory Quoroom-format database (mirrors the reference's initTestDb)."""
    from room_amd.db import init_test_db
    conn = init_test_db()
    yield conn
    conn.close()


@pytest.fixture
def ldb(db):
    from room_amd.db import LockedDb
    return LockedDb(db)
