"""Task runner + cron scheduler tests (reference: task-runner tests pattern)."""
import asyncio
from datetime import datetime

import pytest

from room_amd.core import room
from room_amd.core.cron import CronExpression, validate_cron
from room_amd.core.tasks import TaskRunner
from room_amd.db import LockedDb
from room_amd.db import queries as q


def test_cron_parse_and_match():
    c = CronExpression("*/15 9-17 * * 1-5")
    assert c.matches(datetime(2026, 9, 11, 9, 0))    # Friday
    assert c.matches(datetime(2026, 9, 11, 17, 45))
    assert not c.matches(datetime(2026, 9, 11, 8, 45))
    assert not c.matches(datetime(2026, 9, 13, 9, 0))  # Sunday
    assert not c.matches(datetime(2026, 9, 11, 9, 7))


def test_cron_dow_sunday_zero():
    c = CronExpression("0 0 * * 0")
    assert c.matches(datetime(2026, 9, 13, 0, 0))   # Sunday
    assert not c.matches(datetime(2026, 9, 14, 0, 0))


def test_cron_next_after():
    c = CronExpression("30 6 * * *")
    nxt = c.next_after(datetime(2026, 9, 11, 7, 0))
    assert nxt == datetime(2026, 9, 12, 6, 30)


def test_cron_dom_dow_or_semantics():
    """Standard cron: when both day-of-month and day-of-week are restricted,
    either may match ('0 0 13 * 5' fires every Friday AND every 13th)."""
    c = CronExpression("0 0 13 * 5")
    assert c.matches(datetime(2026, 9, 13, 0, 0))   # the 13th (a Sunday)
    assert c.matches(datetime(2026, 9, 18, 0, 0))   # a Friday (the 18th)
    assert not c.matches(datetime(2026, 9, 14, 0, 0))  # Monday the 14th
    # only one restricted → AND as usual
    c2 = CronExpression("0 0 13 * *")
    assert c2.matches(datetime(2026, 9, 13, 0, 0))
    assert not c2.matches(datetime(2026, 9, 18, 0, 0))
    c3 = CronExpression("0 0 * * 5")
    assert c3.matches(datetime(2026, 9, 18, 0, 0))
    assert not c3.matches(datetime(2026, 9, 13, 0, 0))


def test_cron_validate():
    assert validate_cron("0 9 * * *")
    assert not validate_cron("not a cron")
    assert not validate_cron("61 * * * *")


def test_task_execute_with_stub(db):
    r = room.create_room(db, "taskroom", worker_model="stub")
    ldb = LockedDb(db)
    t = q.create_task(db, "report", "write a report", trigger_type="manual",
                      room_id=r["id"], executor="stub")
    runner = TaskRunner(ldb, default_model="stub")
    out = asyncio.run(runner.execute_task(t["id"]))
    assert out["status"] == "completed"
    with ldb as conn:
        task = q.get_task(conn, t["id"])
        assert task["run_count"] == 1
        run = q.get_latest_task_run(conn, t["id"])
        assert run["status"] == "completed"
        assert run["duration_ms"] is not None


def test_task_once_completes_and_max_runs(db):
    r = room.create_room(db, "onceroom", worker_model="stub")
    ldb = LockedDb(db)
    t = q.create_task(db, "one-shot", "do it once", trigger_type="once",
                      scheduled_at="2020-01-01 00:00:00", room_id=r["id"])
    runner = TaskRunner(ldb, default_model="stub")
    asyncio.run(runner.execute_task(t["id"]))
    with ldb as conn:
        assert q.get_task(conn, t["id"])["status"] == "completed"


def test_task_concurrency_slot(db):
    r = room.create_room(db, "slots", worker_model="stub")
    db.execute("UPDATE rooms SET max_concurrent_tasks = 1 WHERE id = ?", (r["id"],))
    ldb = LockedDb(db)
    runner = TaskRunner(ldb, default_model="stub")
    runner._room_slots[r["id"]] = 1  # room slot already taken
    t = q.create_task(db, "blocked", "p", trigger_type="manual", room_id=r["id"])
    out = asyncio.run(runner.execute_task(t["id"]))
    assert out is None  # no slot available


def test_learned_context_distills(db):
    from room_amd.core.learned_context import distill_learned_context, should_distill
    assert not should_distill(2)
    assert should_distill(3)
    assert should_distill(6)
    assert not should_distill(4, has_context=True)
    # first-eligible distillation happens even off the 3-run cadence when
    # no memo exists yet (learned-context.ts:29)
    assert should_distill(4, has_context=False)
    r = room.create_room(db, "lc", worker_model="stub")
    t = q.create_task(db, "lc-task", "p", room_id=r["id"])
    for i in range(3):
        rid = q.create_task_run(db, t["id"])
        q.finish_task_run(db, rid, "completed", result=f"result {i}")
    memo = distill_learned_context(db, t["id"], model="stub")
    assert memo


def test_builtin_keeper_executors(db, tmp_path, monkeypatch):
    monkeypatch.setenv("ROOMAMD_DATA_DIR", str(tmp_path))
    r = room.create_room(db, "builtin", worker_model="stub")
    q.create_escalation(db, r["id"], "need approval for X")
    ldb = LockedDb(db)
    runner = TaskRunner(ldb, default_model="stub")
    t1 = q.create_task(db, "contact check", "-", trigger_type="manual",
                       room_id=r["id"], executor="keeper_contact_check")
    out = asyncio.run(runner.execute_task(t1["id"]))
    assert out["status"] == "completed"
    assert "1 pending" in out["result"]
    t2 = q.create_task(db, "standup", "time for standup", trigger_type="manual",
                       room_id=r["id"], executor="keeper_reminder")
    out2 = asyncio.run(runner.execute_task(t2["id"]))
    assert out2["result"] == "reminder delivered"
    from room_amd.core.notifications import read_outbox
    box = read_outbox()
    assert len(box) == 2
    assert box[1]["subject"] == "Reminder: standup"
