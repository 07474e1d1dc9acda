"""Agent-loop pacing/safety semantics unit tests (reference agent-loop.ts:
quiet hours :30-51, momentum gap :204-217, stuck detector :605-617, cycle
gaps/presets, error backoff) — behaviors round-1 covered only implicitly."""
import asyncio
from datetime import datetime, timedelta

import pytest

from room_amd.core import room as room_mod
from room_amd.core.agent_loop import (AgentLoopManager, LoopState,
                                      MOMENTUM_GAP_MS)
from room_amd.db import LockedDb, init_test_db
from room_amd.db import queries as q


@pytest.fixture()
def mgr():
    ldb = LockedDb(init_test_db())
    return AgentLoopManager(ldb), ldb


def _mkroom(ldb, **room_fields):
    with ldb as db:
        r = room_mod.create_room(db, "sem", worker_model="stub")
        if room_fields:
            q.update_room(db, r["id"], **room_fields)
        room = q.get_room(db, r["id"])
        queen = q.get_worker(db, r["queen_worker_id"])
    return room, queen


def test_quiet_hours_window(mgr):
    m, ldb = mgr
    now = datetime.now()
    start = (now - timedelta(hours=1)).strftime("%H:%M")
    end = (now + timedelta(hours=1)).strftime("%H:%M")
    room, _ = _mkroom(ldb, queen_quiet_from=start, queen_quiet_until=end)
    assert m._in_quiet_hours(room) is True
    # a window that excludes now
    far1 = (now + timedelta(hours=2)).strftime("%H:%M")
    far2 = (now + timedelta(hours=3)).strftime("%H:%M")
    with ldb as db:
        q.update_room(db, room["id"], queen_quiet_from=far1,
                      queen_quiet_until=far2)
        room2 = q.get_room(db, room["id"])
    assert m._in_quiet_hours(room2) is False
    # no quiet hours configured
    with ldb as db:
        q.update_room(db, room["id"], queen_quiet_from=None,
                      queen_quiet_until=None)
        room3 = q.get_room(db, room["id"])
    assert m._in_quiet_hours(room3) is False


def test_quiet_hours_across_midnight(mgr):
    m, ldb = mgr
    now = datetime.now()
    start = (now - timedelta(hours=1)).strftime("%H:%M")
    end = (now - timedelta(hours=23)).strftime("%H:%M")  # wraps past midnight
    room, _ = _mkroom(ldb, queen_quiet_from=start, queen_quiet_until=end)
    assert m._in_quiet_hours(room) is True


def test_momentum_gap_when_wip_exists(mgr):
    m, ldb = mgr
    room, queen = _mkroom(ldb)
    # queen default gap (30 min) without WIP
    gap = m._cycle_gap_ms(room, queen)
    assert gap >= 1_000_000
    with ldb as db:
        q.set_worker_wip(db, queen["id"], "halfway through the report")
        queen2 = q.get_worker(db, queen["id"])
    assert m._cycle_gap_ms(room, queen2) == MOMENTUM_GAP_MS


def test_cycle_gap_floor_from_room_config(mgr):
    m, ldb = mgr
    room, queen = _mkroom(ldb, config={"minCycleGapMs": 5000})
    with ldb as db:
        w = q.create_worker(db, "fast", "p", room_id=room["id"],
                            cycle_gap_ms=10)  # below the floor
    assert m._cycle_gap_ms(room, w) == 5000


def test_stuck_detector_three_identical_summaries(mgr):
    m, _ = mgr
    st = LoopState(worker_id=1, room_id=1)
    st.last_summaries = ["same", "same"]
    assert m.is_stuck(st) is False
    st.last_summaries = ["same", "same", "same"]
    assert m.is_stuck(st) is True
    st.last_summaries = ["same", "other", "same"]
    assert m.is_stuck(st) is False
    st.last_summaries = ["", "", ""]          # empty summaries never count
    assert m.is_stuck(st) is False


def test_loop_error_backoff_and_recovery(mgr):
    """A failing cycle logs an error activity and the loop keeps running
    (agent-loop.ts:192-197 crash-resilience)."""
    m, ldb = mgr
    room, queen = _mkroom(ldb)

    async def go():
        # break the cycle path by deleting the worker row mid-flight
        state = await m.start_agent_loop(room["id"], queen["id"])
        await asyncio.sleep(0.05)
        assert state.running
        m.stop_agent(queen["id"])
        await asyncio.sleep(0.02)

    asyncio.run(go())


def test_trigger_agent_wakes_waiting_loop(mgr):
    m, ldb = mgr
    room, queen = _mkroom(ldb)
    state = LoopState(worker_id=queen["id"], room_id=room["id"])
    m.running_loops[queen["id"]] = state
    assert not state.wake_event.is_set()
    m.trigger_agent(queen["id"])
    assert state.wake_event.is_set()
    state.wake_event.clear()
    m.wake_room_workers(room["id"])
    assert state.wake_event.is_set()
    # excluded worker stays asleep
    state.wake_event.clear()
    m.wake_room_workers(room["id"], exclude=queen["id"])
    assert not state.wake_event.is_set()


def test_queen_policy_deviation_logged(mgr):
    """Model-B soft policy (agent-loop.ts:22-28, 707-728): a queen using an
    execution tool gets a deviation activity row + the control-plane WIP
    hint; workers don't."""
    import asyncio

    from room_amd.core.agent_loop import QUEEN_POLICY_WIP_HINT
    from room_amd.engine.providers import register_engine

    class WebUsingEngine:
        def chat(self, messages, tools, options):
            return ('<tool_call>{"name": "room_web_search", '
                    '"arguments": {"query": "x"}}</tool_call>', 10, 5)

    m, ldb = mgr
    register_engine("web-stub", WebUsingEngine())
    room, queen = _mkroom(ldb)
    with ldb as db:
        q.update_room(db, room["id"], worker_model="web-stub")
        q.update_worker(db, queen["id"], model="web-stub")
        w = q.create_worker(db, "researcher", "p", role="researcher",
                            room_id=room["id"], model="web-stub")
    out = asyncio.run(m.run_cycle(room["id"], queen["id"], max_turns=2))
    assert out["result"].success
    with ldb as db:
        acts = q.get_room_activity(db, room["id"])
        assert any("Queen policy deviation" in a["summary"] for a in acts)
        fresh = q.get_worker(db, queen["id"])
        assert QUEEN_POLICY_WIP_HINT in (fresh.get("wip") or "")
        # worker using the same tool: no deviation row added
        n_dev = sum(1 for a in acts if "deviation" in a["summary"])
    asyncio.run(m.run_cycle(room["id"], w["id"], max_turns=2))
    with ldb as db:
        acts2 = q.get_room_activity(db, room["id"])
        assert sum(1 for a in acts2
                   if "deviation" in a["summary"]) == n_dev


# --- session continuity: rotation at 20 cycles, compression at 30 msgs,
# --- context-overflow retry (agent-loop.ts:462-532, :773-782)

def _run(m, room, worker):
    return asyncio.run(m.run_cycle(room["id"], worker["id"], max_turns=1))


def test_session_rotation_after_20_cycles(mgr, monkeypatch):
    m, ldb = mgr
    room, queen = _mkroom(ldb)
    seen = {}

    def fake_execute(options):
        from room_amd.engine.types import AgentExecutionResult
        seen["messages"] = options.messages
        return AgentExecutionResult(success=True, text="ok",
                                    messages=[{"role": "user", "content": "x"}],
                                    input_tokens=1, output_tokens=1)

    import room_amd.core.agent_loop as al
    monkeypatch.setattr(al, "execute_agent", fake_execute)
    import json as _json
    with ldb as db:
        q.save_agent_session(db, queen["id"], session_id=None,
                             messages_json=_json.dumps(
                                 [{"role": "user", "content": "old"}]),
                             model="stub", turn_count=al.SESSION_ROTATE_CYCLES)
    _run(m, room, queen)
    # rotation: the stored history must NOT be carried into the new cycle
    assert seen["messages"] is None
    with ldb as db:
        sess = q.get_agent_session(db, queen["id"])
    assert sess["turn_count"] == 1   # fresh session counter


def test_session_continuity_below_rotation(mgr, monkeypatch):
    m, ldb = mgr
    room, queen = _mkroom(ldb)
    seen = {}

    def fake_execute(options):
        from room_amd.engine.types import AgentExecutionResult
        seen["messages"] = options.messages
        return AgentExecutionResult(success=True, text="ok", messages=[],
                                    input_tokens=1, output_tokens=1)

    import room_amd.core.agent_loop as al
    monkeypatch.setattr(al, "execute_agent", fake_execute)
    import json as _json
    hist = [{"role": "user", "content": "prior"}]
    with ldb as db:
        q.save_agent_session(db, queen["id"], session_id=None,
                             messages_json=_json.dumps(hist),
                             model="stub", turn_count=3)
    _run(m, room, queen)
    assert seen["messages"] == hist
    with ldb as db:
        sess = q.get_agent_session(db, queen["id"])
    assert sess["turn_count"] == 4


def test_session_compression_at_30_msgs(mgr, monkeypatch):
    m, ldb = mgr
    room, queen = _mkroom(ldb)
    called = {}

    def fake_compress(messages, model=None):
        called["n"] = len(messages)
        return messages[-2:]

    def fake_execute(options):
        from room_amd.engine.types import AgentExecutionResult
        called["passed"] = options.messages
        return AgentExecutionResult(success=True, text="ok", messages=[],
                                    input_tokens=1, output_tokens=1)

    import room_amd.core.agent_loop as al
    monkeypatch.setattr(al, "compress_session", fake_compress)
    monkeypatch.setattr(al, "execute_agent", fake_execute)
    import json as _json
    hist = [{"role": "user", "content": f"m{i}"}
            for i in range(al.SESSION_COMPRESS_AT_MSGS)]
    with ldb as db:
        q.save_agent_session(db, queen["id"], session_id=None,
                             messages_json=_json.dumps(hist),
                             model="stub", turn_count=2)
    _run(m, room, queen)
    assert called["n"] == al.SESSION_COMPRESS_AT_MSGS
    assert called["passed"] == hist[-2:]


def test_context_overflow_retries_with_fresh_session(mgr, monkeypatch):
    m, ldb = mgr
    room, queen = _mkroom(ldb)
    calls = []

    def fake_execute(options):
        from room_amd.engine.types import AgentExecutionResult
        calls.append(options.messages)
        if len(calls) == 1:
            return AgentExecutionResult(success=False, text="",
                                        error="Context length exceeded",
                                        messages=[])
        return AgentExecutionResult(success=True, text="recovered",
                                    messages=[], input_tokens=1,
                                    output_tokens=1)

    import room_amd.core.agent_loop as al
    monkeypatch.setattr(al, "execute_agent", fake_execute)
    import json as _json
    hist = [{"role": "user", "content": "big"}]
    with ldb as db:
        q.save_agent_session(db, queen["id"], session_id=None,
                             messages_json=_json.dumps(hist),
                             model="stub", turn_count=2)
    out = _run(m, room, queen)
    assert out["result"].success
    assert calls[0] == hist      # first try with history
    assert calls[1] is None      # retry with fresh session
