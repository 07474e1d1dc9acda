"""Clerk commentary pacing semantics (reference clerk-commentary.ts:21-33):
active 8-30 s / light 2-3 h intervals, presence-driven mode switch,
keeper-message pause with 60 s silence threshold, off switch, usage rows —
plus the escalation → keeper-notification relay."""
import json

import pytest

from room_amd.core.clerk import CommentaryEngine, clerk_chat
from room_amd.core.events import EventBus
from room_amd.db import LockedDb, init_test_db
from room_amd.db import queries as q


class Clock:
    def __init__(self):
        self.t = 1000.0

    def __call__(self):
        return self.t


@pytest.fixture()
def eng():
    ldb = LockedDb(init_test_db())
    bus = EventBus()
    clock = Clock()
    e = CommentaryEngine(ldb, bus, model="stub", time_source=clock)
    return e, bus, clock, ldb


def _feed(bus, n=3):
    for i in range(n):
        bus.emit("room:1", "cycle_finished", {"type": "cycle_finished", "i": i})


def test_active_pace_with_presence(eng):
    e, bus, clock, ldb = eng
    bus.emit("clerk", "presence", {"type": "presence"})
    _feed(bus)
    assert e.current_pace() == "active"
    clock.t += 31                       # past the max active interval
    line = e.tick()
    assert line is not None
    # next due is within the active window
    assert e._next_due - clock.t <= CommentaryEngine.ACTIVE_MAX_S
    # a second tick immediately after is not due yet
    _feed(bus)
    assert e.tick() is None


def test_light_pace_when_keeper_away(eng):
    e, bus, clock, ldb = eng
    # no presence heartbeat → light pace
    assert e.current_pace() == "light"
    _feed(bus)
    clock.t += 31
    e.tick()  # may or may not fire depending on initial due; force schedule
    e._schedule_next()
    assert e._next_due - clock.t >= CommentaryEngine.LIGHT_MIN_S


def test_light_mode_setting_overrides_presence(eng):
    e, bus, clock, ldb = eng
    with ldb as db:
        q.set_setting(db, "clerk_commentary_mode", "light")
    bus.emit("clerk", "presence", {"type": "presence"})
    assert e.current_pace() == "light"


def test_keeper_message_pauses_commentary(eng):
    e, bus, clock, ldb = eng
    bus.emit("clerk", "presence", {"type": "presence"})
    _feed(bus)
    clock.t += 31
    bus.emit("clerk", "keeper_message", {"type": "keeper_message"})
    assert e.tick() is None             # paused right after keeper message
    clock.t += 30
    assert e.tick() is None             # still inside the 60 s silence window
    clock.t += 31                       # 61 s of silence → resumes
    assert e.tick() is not None


def test_commentary_off_switch(eng):
    e, bus, clock, ldb = eng
    with ldb as db:
        q.set_setting(db, "clerk_commentary", "off")
    _feed(bus)
    clock.t += 10_000
    assert e.tick() is None


def test_commentary_usage_rows_accumulate(eng):
    e, bus, clock, ldb = eng
    bus.emit("clerk", "presence", {"type": "presence"})
    _feed(bus)
    clock.t += 31
    e.tick()
    with ldb as db:
        rows = db.execute("SELECT * FROM clerk_usage WHERE source ="
                          " 'commentary'").fetchall()
    assert rows and rows[0]["model"] == "stub"


def test_clerk_chat_accumulates_usage():
    ldb = LockedDb(init_test_db())
    clerk_chat(ldb, "list the rooms", model="stub")
    clerk_chat(ldb, "what is running?", model="stub")
    with ldb as db:
        rows = db.execute(
            "SELECT * FROM clerk_usage WHERE source = 'chat'").fetchall()
    assert len(rows) == 2


def test_escalation_alert_relay(tmp_path, monkeypatch):
    """Pending keeper escalations land in the notification outbox."""
    import asyncio

    from room_amd.core import room as room_mod
    from room_amd.server.runtime import ServerRuntime
    from room_amd.core.tasks import TaskRunner

    monkeypatch.setenv("ROOMAMD_DATA_DIR", str(tmp_path))
    ldb = LockedDb(init_test_db())
    with ldb as db:
        r = room_mod.create_room(db, "alert-room", worker_model="stub")
        q.create_escalation(db, r["id"], "Keeper, which vendor do we pick?")

    async def run():
        rt = ServerRuntime(ldb, TaskRunner(ldb), bus=EventBus())
        task = asyncio.create_task(rt._alert_relay_loop())
        for _ in range(60):                 # poll: loop start can lag under load
            await asyncio.sleep(0.05)
            if (tmp_path / "outbox.jsonl").exists():
                break
        rt._stop.set()
        task.cancel()

    asyncio.run(run())
    outbox = (tmp_path / "outbox.jsonl").read_text().strip().splitlines()
    entries = [json.loads(l) for l in outbox]
    assert any("escalation" in e["subject"] for e in entries), entries
