"""CycleLogBuffer: seq numbering, batched flush, WS fan-out, both tables.

Reference: src/shared/console-log-buffer.ts (seq-numbered entries, 1s batch
SQLite flush, synthetic system entries, callback for WS fan-out).
"""
import time

import pytest

from room_amd.core.events import EventBus
from room_amd.core.log_buffer import CycleLogBuffer
from room_amd.core.room import create_room
from room_amd.db import LockedDb, connect
from room_amd.db import queries as q


@pytest.fixture
def env(tmp_path):
    ldb = LockedDb(connect(str(tmp_path / "t.db")))
    with ldb as db:
        r = create_room(db, "logroom", goal="g", worker_model="stub")
        cyc = q.create_worker_cycle(db, r["queen_worker_id"], r["id"])
    return ldb, r, cyc  # cyc is the cycle id (int)


def test_seq_numbers_monotonic_and_persisted(env):
    ldb, r, cyc = env
    buf = CycleLogBuffer(ldb, cyc, room_id=r["id"])
    seqs = [buf.append("text", f"entry {i}") for i in range(5)]
    assert seqs == [0, 1, 2, 3, 4]
    buf.flush()
    with ldb as db:
        logs = q.get_cycle_logs(db, cyc)
    assert [(l["seq"], l["content"]) for l in logs] == \
        [(i, f"entry {i}") for i in range(5)]


def test_batching_no_flush_before_interval(env):
    ldb, r, cyc = env
    buf = CycleLogBuffer(ldb, cyc, room_id=r["id"])
    buf.append("text", "a")
    with ldb as db:
        assert q.get_cycle_logs(db, cyc) == []  # still pending
    buf.flush()
    with ldb as db:
        assert len(q.get_cycle_logs(db, cyc)) == 1


def test_time_based_auto_flush(env):
    ldb, r, cyc = env
    buf = CycleLogBuffer(ldb, cyc, room_id=r["id"])
    buf.append("text", "a")
    buf._last_flush = time.time() - 2.0   # age past FLUSH_INTERVAL_S
    buf.append("text", "b")               # triggers flush of both
    with ldb as db:
        assert len(q.get_cycle_logs(db, cyc)) == 2


def test_system_entries_and_bus_fanout(env):
    ldb, r, cyc = env
    bus = EventBus()
    got = []
    bus.on(f"room:{r['id']}", lambda ch, ev: got.append(ev))
    buf = CycleLogBuffer(ldb, cyc, bus=bus, room_id=r["id"])
    buf.system("cycle started")
    buf.append("tool_call", "recall(...)")
    assert [g["data"]["entry_type"] for g in got] == ["system", "tool_call"]
    assert got[0]["data"]["content"] == "cycle started"
    assert got[0]["data"]["seq"] == 0


def test_console_table_routes_to_run_channel(env):
    ldb, r, cyc = env
    bus = EventBus()
    got = []
    with ldb as db:
        task = q.create_task(db, "t", "do", trigger_type="manual",
                             room_id=r["id"])
        run_id = q.create_task_run(db, task["id"])
    bus.on(f"run:{run_id}", lambda ch, ev: got.append(ev))
    buf = CycleLogBuffer(ldb, run_id, bus=bus, room_id=r["id"],
                         table="console")
    buf.append("text", "task output")
    buf.flush()
    assert len(got) == 1
    with ldb as db:
        logs = q.get_console_logs(db, run_id)
    assert logs and logs[0]["content"] == "task output"


def test_flush_empty_is_noop(env):
    ldb, r, cyc = env
    buf = CycleLogBuffer(ldb, cyc, room_id=r["id"])
    buf.flush()  # nothing pending — must not error
    with ldb as db:
        assert q.get_cycle_logs(db, cyc) == []
