"""Database layer: SQLite (WAL) with the preserved Quoroom schema.

The reference keeps all state in a single SQLite file opened WAL-mode with a
5s busy timeout from two processes (src/mcp/db.ts:26-28, src/server/db.ts:41-44).
We do the same via the stdlib sqlite3 module (the same C library underneath).
"""
from __future__ import annotations

import sqlite3
import threading
from pathlib import Path

from .schema import SCHEMA
from .migrations import run_migrations


def _dict_factory(cursor: sqlite3.Cursor, row: tuple) -> dict:
    return {d[0]: row[i] for i, d in enumerate(cursor.description)}


def connect(path: str = ":memory:") -> sqlite3.Connection:
    """Open (and initialize) a Quoroom-format database."""
    if path != ":memory:":
        Path(path).parent.mkdir(parents=True, exist_ok=True)
    db = sqlite3.connect(path, timeout=5.0, check_same_thread=False)
    db.row_factory = _dict_factory
    db.executescript(SCHEMA)
    run_migrations(db)
    db.commit()
    return db


def init_test_db() -> sqlite3.Connection:
    """In-memory database with full schema — the universal test fixture
    (mirrors the reference's initTestDb, src/shared/__tests__/helpers/test-db.ts:4-8)."""
    return connect(":memory:")


class LockedDb:
    """Serialize access to one sqlite3 connection across asyncio/threads.

    sqlite3 connections are not safe for concurrent statement execution from
    multiple threads; the agent loops, server routes, and scheduler all share
    one connection, so writes go through this lock (the reference relied on
    Node's single thread for the same guarantee).
    """

    def __init__(self, conn: sqlite3.Connection):
        self.conn = conn
        self.lock = threading.RLock()

    def __enter__(self) -> sqlite3.Connection:
        self.lock.acquire()
        return self.conn

    def __exit__(self, *exc) -> None:
        self.conn.commit()
        self.lock.release()
