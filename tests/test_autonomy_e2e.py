"""Full-autonomy scenario (BASELINE config 5, CPU/stub analogue): template
room → queen cycle → quorum lifecycle → self-modification with revert →
cron/webhook scheduler → escalation answer → memory accrual, all through
the public surfaces (HTTP API + core modules) against one SQLite file."""
import asyncio
import json

import pytest
from fastapi.testclient import TestClient

from room_amd.core import self_mod
from room_amd.core.agent_loop import AgentLoopManager
from room_amd.core.tasks import TaskRunner
from room_amd.db import LockedDb, init_test_db
from room_amd.db import queries as q
from room_amd.server.app import create_app


@pytest.fixture()
def env(tmp_path, monkeypatch):
    monkeypatch.setenv("ROOMAMD_DATA_DIR", str(tmp_path))
    monkeypatch.setenv("ROOMAMD_RESULTS_DIR", str(tmp_path / "results"))
    ldb = LockedDb(init_test_db())
    mgr = AgentLoopManager(ldb)
    runner = TaskRunner(ldb)
    app = create_app(ldb, loop_mgr=mgr, runner=runner)
    c = TestClient(app)
    tok = c.post("/api/auth/handshake").json()["token"]
    c.headers["Authorization"] = f"Bearer {tok}"
    return c, ldb, mgr, runner


def test_full_autonomy_scenario(env):
    c, ldb, mgr, runner = env

    # 1. room from template: queen + 5 workers, goal tree seeded
    room = c.post("/api/rooms/from-template",
                  json={"template": "product-studio", "name": "studio",
                        "worker_model": "stub"}).json()
    rid = room["id"]
    workers = c.get(f"/api/rooms/{rid}/workers").json()
    assert len(workers) == 6                      # queen + 5 archetypes
    queen_id = room["queen_worker_id"]

    # 2. one real queen cycle through the stub engine
    out = asyncio.run(mgr.run_cycle(rid, queen_id, max_turns=2))
    assert out["result"].success
    cycles = c.get(f"/api/rooms/{rid}/cycles").json()
    assert cycles and cycles[0]["status"] == "completed"

    # 3. quorum: objection kills one announcement; silence passes another
    d1 = c.post(f"/api/rooms/{rid}/decisions",
                json={"proposal": "rewrite everything in assembly",
                      "decision_type": "strategy", "delay_minutes": 10}).json()
    wid = workers[1]["id"]
    obj = c.post(f"/api/decisions/{d1['id']}/object",
                 json={"worker_id": wid, "reason": "scope explosion"})
    assert obj.status_code == 200
    assert c.get(f"/api/decisions/{d1['id']}").json()["status"] == "objected"
    # low_impact is on the default autoApprove list → resolves instantly
    auto = c.post(f"/api/rooms/{rid}/decisions",
                  json={"proposal": "minor copy tweak",
                        "decision_type": "low_impact"}).json()
    assert auto["status"] == "approved"
    d2 = c.post(f"/api/rooms/{rid}/decisions",
                json={"proposal": "adopt weekly demo cadence",
                      "decision_type": "strategy",
                      "delay_minutes": 0}).json()
    from room_amd.core import quorum
    with ldb as db:                                # expiry sweep (cycle-top)
        quorum.check_expired_decisions(db)
    assert c.get(f"/api/decisions/{d2['id']}").json()["status"] == "effective"

    # 4. self-modification: audited skill edit + rate limit + true revert
    sk = c.post(f"/api/rooms/{rid}/skills",
                json={"name": "deploy", "content": "v1 steps"}).json()
    with ldb as db:
        audit = self_mod.perform_skill_modification(
            db, rid, queen_id, sk["id"], "v2 steps improved",
            reason="sharpen the runbook")
        assert audit["audit_id"]
        with pytest.raises(PermissionError):       # 60s per-worker rate limit
            self_mod.perform_skill_modification(db, rid, queen_id, sk["id"],
                                                "v3 too fast")
    assert c.get(f"/api/skills/{sk['id']}").json()["content"] == "v2 steps improved"
    with ldb as db:
        self_mod.revert_modification(db, audit["audit_id"])
    assert c.get(f"/api/skills/{sk['id']}").json()["content"] == "v1 steps"

    # 5. scheduler: cron task fires through the runner; result file lands
    t = c.post("/api/tasks", json={"name": "digest", "prompt": "write digest",
                                   "room_id": rid, "model": "stub",
                                   "cron_expression": "* * * * *"}).json()
    run_out = asyncio.run(runner.execute_task(t["id"]))
    assert run_out["status"] == "completed"
    runs = c.get(f"/api/tasks/{t['id']}/runs").json()
    assert runs and runs[0]["status"] == "completed"

    # 6. webhook → escalation → keeper answer → queen sees it
    with ldb as db:
        hook = q.get_room(db, rid)["webhook_token"]
    wh = c.post(f"/api/hooks/queen/{hook}",
                json={"message": "customer asks for an invoice"})
    assert wh.status_code == 200
    esc = c.get(f"/api/rooms/{rid}/escalations").json()
    pending = [e for e in esc if e["status"] == "pending"]
    assert pending
    ans = c.post(f"/api/escalations/{pending[0]['id']}/resolve",
                 json={"answer": "send invoice #42"}).json()
    assert ans["status"] == "answered"

    # 7. memory accrues and is recallable
    c.post("/api/memory/entities",
           json={"name": "studio-pricing", "room_id": rid,
                 "content": "enterprise tier is 500 per month"})
    hits = c.get("/api/memory/search",
                 params={"q": "enterprise pricing tier", "room_id": rid}).json()
    assert any("pricing" in (h.get("name") or "") for h in hits)

    # 8. status rollup reflects the session's work
    st = c.get(f"/api/rooms/{rid}/status").json()
    assert st["token_usage"]["cycles"] >= 1
    badges = c.get(f"/api/rooms/{rid}/badges").json()
    assert badges["active_votes"] == 0             # both decisions resolved
