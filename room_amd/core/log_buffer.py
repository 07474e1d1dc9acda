"""Seq-numbered streaming cycle-log buffer with batched SQLite flush
(reference: src/shared/console-log-buffer.ts — 1s batch cadence, synthetic
system entries, callback for WS fan-out)."""
from __future__ import annotations

import time

from ..db import LockedDb
from ..db import queries as q
from .events import EventBus

FLUSH_INTERVAL_S = 1.0


class CycleLogBuffer:
    def __init__(self, ldb: LockedDb, cycle_id: int, bus: EventBus | None = None,
                 room_id: int | None = None, table: str = "cycle"):
        self.ldb = ldb
        self.cycle_id = cycle_id
        self.bus = bus
        self.room_id = room_id
        self.table = table
        self._seq = 0
        self._pending: list[tuple[int, str, str]] = []
        self._last_flush = time.time()

    def append(self, entry_type: str, content: str) -> int:
        seq = self._seq
        self._seq += 1
        self._pending.append((seq, entry_type, content))
        if self.bus and self.room_id is not None:
            channel = f"run:{self.cycle_id}" if self.table == "console" \
                else f"room:{self.room_id}"
            self.bus.emit(channel, "log",
                          {"cycle_id": self.cycle_id, "seq": seq,
                           "entry_type": entry_type, "content": content})
        if time.time() - self._last_flush >= FLUSH_INTERVAL_S:
            self.flush()
        return seq

    def system(self, content: str) -> int:
        return self.append("system", content)

    def flush(self) -> None:
        if not self._pending:
            return
        batch, self._pending = self._pending, []
        with self.ldb as db:
            if self.table == "console":
                q.add_console_logs(db, self.cycle_id, batch)
            else:
                q.add_cycle_logs(db, self.cycle_id, batch)
        self._last_flush = time.time()
