"""Whole-surface authorized route smoke: every registered route, called with
synthesized path params and an empty/qualifying body, must produce a handled
response — 2xx/3xx/4xx or a deliberate 5xx with a JSON error body — never
FastAPI's bare 500 "Internal Server Error" (the unhandled-exception marker).

Complements tests/test_route_auth_matrix.py (which proves 401/403) by
exercising the authorized path of all ~169 handlers.
"""
import pytest
from fastapi.routing import APIRoute
from fastapi.testclient import TestClient

from room_amd.core import room as room_mod
from room_amd.core.agent_loop import AgentLoopManager
from room_amd.core.events import EventBus
from room_amd.core.tasks import TaskRunner
from room_amd.db import LockedDb, init_test_db
from room_amd.db import queries as q
from room_amd.memory.vector_store import GpuVectorStore, MemoryService
from room_amd.server.app import create_app
from room_amd.server.auth import AuthManager


@pytest.fixture(scope="module")
def env():
    ldb = LockedDb(init_test_db())
    bus = EventBus()
    auth = AuthManager(skip_token_file=True)
    memory = MemoryService(ldb, store=GpuVectorStore(capacity=1000,
                                                     device="cpu"))
    mgr = AgentLoopManager(ldb, bus=bus, memory=memory)
    runner = TaskRunner(ldb, bus=bus, memory=memory, default_model="stub")
    app = create_app(ldb, loop_mgr=mgr, runner=runner, memory=memory,
                     auth=auth, bus=bus)
    client = TestClient(app, raise_server_exceptions=False)
    h = {"Authorization": f"Bearer {auth.agent_token}"}

    with ldb as db:
        r = room_mod.create_room(db, "route-smoke", goal="g",
                                 worker_model="stub")
        ids = {
            "room_id": r["id"],
            "worker_id": r["queen_worker_id"],
            "goal_id": q.create_goal(db, r["id"], "g1")["id"],
            "decision_id": q.create_decision(db, r["id"],
                                             r["queen_worker_id"],
                                             "d", "low_impact")["id"],
            "entity_id": q.create_entity(db, "e", room_id=r["id"])["id"],
            "task_id": q.create_task(db, "t", "p", trigger_type="manual",
                                     room_id=r["id"])["id"],
        }
        ids["run_id"] = q.create_task_run(db, ids["task_id"])
    return app, client, h, ids


def _fill(path: str, ids: dict) -> str:
    import re

    def sub(m):
        name = m.group(1)
        if name in ids:
            return str(ids[name])
        if name == "token":
            return "0" * 64
        if "id" in name.lower():
            return "1"
        return "smoke"

    return re.sub(r"\{([^}]+)\}", sub, path)


def _routes(app):
    out = []
    for r in app.routes:
        if isinstance(r, APIRoute) and r.path.startswith("/api"):
            for m in sorted(r.methods - {"HEAD", "OPTIONS"}):
                out.append((m, r.path))
    return sorted(set(out))


def test_surface_is_large(env):
    app, *_ = env
    assert len(_routes(app)) >= 160


def test_every_route_handled(env):
    app, client, h, ids = env
    failures = []
    for method, path in _routes(app):
        url = _fill(path, ids)
        try:
            if method == "GET":
                resp = client.get(url, headers=h)
            elif method == "DELETE":
                resp = client.delete(url, headers=h)
            else:
                resp = client.request(method, url, headers=h, json={})
        except Exception as e:
            failures.append((method, path, f"raised {e!r}"))
            continue
        if resp.status_code == 500 and resp.text.strip() == "Internal Server Error":
            failures.append((method, path, "unhandled 500"))
    assert not failures, failures


def test_missing_field_returns_400(env):
    """App-level KeyError handler: a bare payload["x"] miss is a 400 with
    the field named, not an unhandled 500."""
    app, client, h, ids = env
    r = client.post("/api/clerk/chat", headers=h, json={})
    assert r.status_code == 400
    r2 = client.post("/api/tasks", headers=h, json={})
    assert r2.status_code == 400
    assert "missing field" in r2.text or "required" in r2.text


def test_dangling_reference_is_4xx(env):
    app, client, h, ids = env
    r = client.post("/api/goals/999999/updates", headers=h,
                    json={"observation": "x"})
    assert r.status_code == 404
    r2 = client.post("/api/decisions/999999/vote", headers=h,
                     json={"worker_id": 1, "vote": "yes"})
    assert r2.status_code == 404
