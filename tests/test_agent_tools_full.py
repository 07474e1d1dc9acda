"""Per-branch coverage of the agent tool executor (queen-tools.ts parity):
every tool in the registry exercised against real SQL, including the ones
only implicitly covered elsewhere (vote, wip, messaging, escalation, goal
progress/complete, wallet send, web fetch / browser offline degradation).
"""
import json

import pytest

from room_amd.core import room as room_mod
from room_amd.core.agent_tools import (QUEEN_TOOLS, WORKER_TOOLS,
                                       execute_agent_tool, tools_for_role)
from room_amd.db import LockedDb, init_test_db
from room_amd.db import queries as q
from room_amd.engine.types import ToolCall


@pytest.fixture
def env():
    db = init_test_db()
    r = room_mod.create_room(db, "tools", goal="g", worker_model="stub")
    w = q.create_worker(db, "exec", "p", role="executor", room_id=r["id"])
    return db, r, w


def run(db, r, w, name, args):
    return json.loads(execute_agent_tool(db, r["id"], w["id"],
                                         ToolCall(name, args)))


def test_role_split():
    qn = {t.name for t in tools_for_role("queen")}
    wn = {t.name for t in tools_for_role("executor")}
    assert len(qn) == len(QUEEN_TOOLS) and len(wn) == len(WORKER_TOOLS)
    assert "room_delegate_task" in qn and "room_delegate_task" not in wn
    assert "room_save_wip" in wn


def test_save_wip(env):
    db, r, w = env
    out = run(db, r, w, "room_save_wip", {"wip": "half-done analysis"})
    assert out["saved"]
    assert q.get_worker(db, w["id"])["wip"] == "half-done analysis"


def test_vote(env):
    db, r, w = env
    d = q.create_decision(db, r["id"], r["queen_worker_id"], "plan",
                          "low_impact")
    out = run(db, r, w, "room_vote", {"decision_id": d["id"], "vote": "yes"})
    assert "error" not in out
    votes = q.get_votes(db, d["id"])
    assert any(v["worker_id"] == w["id"] and v["vote"] == "yes"
               for v in votes)


def test_send_message_keeper_becomes_escalation(env):
    db, r, w = env
    out = run(db, r, w, "room_send_message",
              {"to": "keeper", "body": "need api key"})
    assert "escalation_id" in out


def test_send_message_to_room(env):
    db, r, w = env
    out = run(db, r, w, "room_send_message",
              {"to": "other-room", "subject": "hi", "body": "msg"})
    assert "message_id" in out


def test_escalate_logs_activity(env):
    db, r, w = env
    out = run(db, r, w, "room_escalate", {"question": "blocked on X?"})
    assert "escalation_id" in out
    acts = q.get_room_activity(db, r["id"], limit=5)
    assert any(a["event_type"] == "escalation" for a in acts)


def test_goal_progress_and_complete(env):
    db, r, w = env
    g = q.create_goal(db, r["id"], "subtask")
    out = run(db, r, w, "room_update_goal_progress",
              {"goal_id": g["id"], "progress": 0.5, "observation": "half"})
    assert out["progress"] == 0.5
    out2 = run(db, r, w, "room_complete_goal", {"goal_id": g["id"]})
    assert out2["status"] == "completed"


def test_send_token_offline_is_structured_error(env):
    db, r, w = env
    out = run(db, r, w, "room_send_token",
              {"to_address": "0x" + "1" * 40, "amount": "1.0"})
    # no chain RPC in this environment: either a structured error or a
    # recorded-but-unbroadcast tx — never a crash
    assert isinstance(out, dict)
    assert "error" in out or "tx" in out or "status" in out


def test_web_fetch_and_browser_degrade_offline(env):
    db, r, w = env
    out = run(db, r, w, "room_web_fetch", {"url": "https://example.com"})
    assert isinstance(out, dict) and "error" in out  # no egress
    out2 = run(db, r, w, "room_browser",
               {"session_id": "s1", "action": {"type": "navigate",
                                               "url": "https://x.com"}})
    assert isinstance(out2, dict)
    assert "error" in out2 or "snapshot" in out2  # no chromium in CI


def test_missing_args_are_tool_errors_not_raises(env):
    db, r, w = env
    for name in [t.name for t in QUEEN_TOOLS]:
        out = run(db, r, w, name, {})
        assert isinstance(out, dict)  # error fed back to the model
