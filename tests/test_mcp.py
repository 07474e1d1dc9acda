"""MCP stdio server tests: JSON-RPC protocol + tool round-trips against the
shared SQLite file (reference: src/mcp tests pattern)."""
import json

import pytest

from room_amd.db import LockedDb
from room_amd.mcp.server import McpServer


@pytest.fixture
def mcp(db):
    nudged = []
    srv = McpServer(LockedDb(db), nudge=lambda wid: nudged.append(wid) or True)
    srv._nudged = nudged
    return srv


def call(srv, name, args, mid=1):
    resp = srv.handle({"jsonrpc": "2.0", "id": mid, "method": "tools/call",
                       "params": {"name": name, "arguments": args}})
    assert "result" in resp, resp
    payload = json.loads(resp["result"]["content"][0]["text"])
    return payload


def test_initialize_and_list(mcp):
    init = mcp.handle({"jsonrpc": "2.0", "id": 0, "method": "initialize",
                       "params": {}})
    assert init["result"]["serverInfo"]["name"] == "room-amd"
    lst = mcp.handle({"jsonrpc": "2.0", "id": 1, "method": "tools/list"})
    tools = lst["result"]["tools"]
    names = {t["name"] for t in tools}
    assert len(tools) >= 30
    assert {"room_create_room", "room_announce", "room_recall",
            "room_delegate_task", "room_modify_skill"} <= names
    # every tool has a schema
    assert all("inputSchema" in t for t in tools)


def test_room_tools_round_trip(mcp):
    out = call(mcp, "room_create_room", {"name": "mcp-room", "goal": "g",
                                         "worker_model": "stub"})
    rid = out["room_id"]
    status = call(mcp, "room_get_status", {"room_id": rid})
    assert status["room"] == "mcp-room"
    w = call(mcp, "room_create_worker", {"room_id": rid, "name": "e",
                                         "role": "executor"})
    d = call(mcp, "room_delegate_task", {"room_id": rid, "description": "do",
                                         "worker_id": w["worker_id"]})
    assert mcp._nudged == [w["worker_id"]]
    tree = call(mcp, "room_goal_tree", {"room_id": rid})
    assert any(g["assigned_worker_id"] == w["worker_id"] for g in tree)


def test_quorum_tools(mcp):
    rid = call(mcp, "room_create_room", {"name": "qr", "worker_model": "stub"})["room_id"]
    w = call(mcp, "room_create_worker", {"room_id": rid, "name": "v",
                                         "role": "guardian"})
    d = call(mcp, "room_announce", {"room_id": rid, "proposal": "big",
                                    "decision_type": "strategy"})
    assert d["status"] == "announced"
    o = call(mcp, "room_object", {"decision_id": d["decision_id"],
                                  "worker_id": w["worker_id"], "reason": "no"})
    assert o["status"] == "objected"


def test_memory_and_skill_tools(mcp):
    rid = call(mcp, "room_create_room", {"name": "mr", "worker_model": "stub"})["room_id"]
    call(mcp, "room_remember", {"room_id": rid, "name": "fact",
                                "content": "the answer is 42"})
    hits = call(mcp, "room_recall", {"room_id": rid, "query": "answer"})
    assert hits and hits[0]["name"] == "fact"
    s = call(mcp, "room_create_skill", {"room_id": rid, "name": "recipe",
                                        "content": "v1"})
    wid = call(mcp, "room_create_worker", {"room_id": rid, "name": "m",
                                           "role": "analyst"})["worker_id"]
    m = call(mcp, "room_modify_skill", {"room_id": rid, "worker_id": wid,
                                        "skill_id": s["skill_id"],
                                        "content": "v2"})
    r = call(mcp, "room_revert_modification", {"audit_id": m["audit_id"]})
    assert r["reverted"]


def test_task_and_wallet_tools(mcp):
    rid = call(mcp, "room_create_room", {"name": "tw", "worker_model": "stub"})["room_id"]
    t = call(mcp, "room_create_task", {"name": "nightly", "prompt": "p",
                                       "cron_expression": "0 2 * * *",
                                       "room_id": rid})
    tasks = call(mcp, "room_list_tasks", {"room_id": rid})
    assert any(x["id"] == t["task_id"] for x in tasks)
    addr = call(mcp, "room_wallet_address", {"room_id": rid})
    assert addr["address"].startswith("0x")
    ident = call(mcp, "room_register_identity", {"room_id": rid})
    assert ident["agent_uri"].startswith("data:application/json;base64,")


def test_watch_path_validation(mcp):
    bad = call(mcp, "room_watch_path", {"path": "/etc/passwd"})
    assert "error" in bad
    bad2 = call(mcp, "room_watch_path", {"path": "~/.ssh/config"})
    assert "error" in bad2
    ok = call(mcp, "room_watch_path", {"path": "~/projects/data"})
    assert "watch_id" in ok


def test_unknown_tool_and_method(mcp):
    resp = mcp.handle({"jsonrpc": "2.0", "id": 5, "method": "tools/call",
                       "params": {"name": "nope", "arguments": {}}})
    assert resp["error"]["code"] == -32602
    resp = mcp.handle({"jsonrpc": "2.0", "id": 6, "method": "bogus"})
    assert resp["error"]["code"] == -32601
