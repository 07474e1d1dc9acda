"""Additive schema migrations (reference: src/shared/db-migrations.ts:13-143).

Migrations are idempotent column/index additions keyed by schema_version so an
existing Quoroom data.db can be opened by this runtime.
"""
from __future__ import annotations

import sqlite3

# (version, [statements]) — version 1 is the base schema.
MIGRATIONS: list[tuple[int, list[str]]] = [
    # Future additive migrations land here, e.g.:
    # (2, ["ALTER TABLE workers ADD COLUMN gpu_rank INTEGER"]),
]


def _has_column(db: sqlite3.Connection, table: str, column: str) -> bool:
    rows = db.execute(f"PRAGMA table_info({table})").fetchall()
    return any(r["name"] == column for r in rows)


def run_migrations(db: sqlite3.Connection) -> int:
    applied = 0
    cur = db.execute("SELECT MAX(version) AS v FROM schema_version").fetchone()
    current = cur["v"] or 1
    for version, stmts in MIGRATIONS:
        if version <= current:
            continue
        for stmt in stmts:
            try:
                db.execute(stmt)
            except sqlite3.OperationalError as e:
                if "duplicate column" not in str(e):
                    raise
        db.execute("INSERT OR IGNORE INTO schema_version (version) VALUES (?)", (version,))
        applied += 1
    return applied
