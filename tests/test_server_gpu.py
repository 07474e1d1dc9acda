"""Full-stack GPU E2E: real server app + 30B engine + agent loop, driven
through the HTTP API (BASELINE config 5 shape: autonomy surface on GPU)."""
import asyncio
import time

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)


def test_room_cycle_via_http_api():
    from fastapi.testclient import TestClient

    from room_amd.core.agent_loop import AgentLoopManager
    from room_amd.core.events import EventBus
    from room_amd.core.tasks import TaskRunner
    from room_amd.db import LockedDb, init_test_db
    from room_amd.engine.llm import LocalEngine
    from room_amd.engine.providers import register_engine
    from room_amd.memory.vector_store import GpuVectorStore, MemoryService
    from room_amd.models.qwen3_moe import Qwen3MoEConfig
    from room_amd.server.app import create_app
    from room_amd.server.auth import AuthManager

    eng = LocalEngine(cfg=Qwen3MoEConfig.tiny(), kv_gb=2.0, max_seqs=16)
    register_engine("tiny-e2e", eng)
    try:
        ldb = LockedDb(init_test_db())
        bus = EventBus()
        auth = AuthManager(skip_token_file=True)
        memory = MemoryService(ldb, store=GpuVectorStore(capacity=10_000,
                                                         device="cuda"))
        mgr = AgentLoopManager(ldb, bus=bus, memory=memory)
        runner = TaskRunner(ldb, bus=bus, memory=memory,
                            default_model="tiny-e2e")
        app = create_app(ldb, loop_mgr=mgr, runner=runner, memory=memory,
                         auth=auth, bus=bus)
        client = TestClient(app)
        h = {"Authorization": f"Bearer {auth.agent_token}"}

        room = client.post("/api/rooms",
                           json={"name": "gpu-e2e", "goal": "run on the GPU",
                                 "worker_model": "tiny-e2e"}, headers=h).json()
        rid = room["id"]
        # GPU memory store through the API
        client.post("/api/memory/entities",
                    json={"room_id": rid, "name": "gpu fact",
                          "content": "the vector store lives in HBM"}, headers=h)
        hits = client.get("/api/memory/search",
                          params={"query": "vector store HBM", "room_id": rid},
                          headers=h).json()
        assert hits and hits[0]["name"] == "gpu fact"

        # one real cycle through the engine (decode on CDNA4 kernels)
        out = asyncio.run(mgr.run_cycle(rid, room["queen_worker_id"],
                                        max_turns=1))
        assert out["result"].success, out["result"].error
        assert out["result"].output_tokens > 0

        cycles = client.get(f"/api/rooms/{rid}/cycles", headers=h).json()
        assert cycles and cycles[0]["status"] == "completed"
        assert cycles[0]["output_tokens"] > 0
        status = client.get(f"/api/rooms/{rid}/status", headers=h).json()
        assert status["token_usage"]["cycles"] >= 1
    finally:
        eng.shutdown()


def test_autonomy_loop_selfmod_scheduler(  ):
    """BASELINE config 5 surface on GPU: skill self-modification (audited,
    revertable), webhook->queen escalation, and a scheduled task executing
    through the engine."""
    import asyncio

    from fastapi.testclient import TestClient

    from room_amd.core.agent_loop import AgentLoopManager
    from room_amd.core.events import EventBus
    from room_amd.core.tasks import TaskRunner
    from room_amd.db import LockedDb, init_test_db
    from room_amd.engine.llm import LocalEngine
    from room_amd.engine.providers import register_engine
    from room_amd.models.qwen3_moe import Qwen3MoEConfig
    from room_amd.server.app import create_app
    from room_amd.server.auth import AuthManager

    eng = LocalEngine(cfg=Qwen3MoEConfig.tiny(), kv_gb=2.0, max_seqs=16)
    register_engine("tiny-auto", eng)
    try:
        ldb = LockedDb(init_test_db())
        auth = AuthManager(skip_token_file=True)
        bus = EventBus()
        mgr = AgentLoopManager(ldb, bus=bus)
        runner = TaskRunner(ldb, bus=bus, default_model="tiny-auto")
        app = create_app(ldb, loop_mgr=mgr, runner=runner, auth=auth, bus=bus)
        client = TestClient(app)
        h = {"Authorization": f"Bearer {auth.agent_token}"}

        room = client.post("/api/rooms",
                           json={"name": "auto", "goal": "autonomy",
                                 "worker_model": "tiny-auto"},
                           headers=h).json()
        rid = room["id"]

        # audited self-modification of a skill + revert (agent tool path)
        from room_amd.core import self_mod
        sk = client.post(f"/api/rooms/{rid}/skills",
                         json={"name": "deploy", "content": "v1"},
                         headers=h).json()
        with ldb as db:
            self_mod.perform_skill_modification(
                db, rid, room["queen_worker_id"], sk["id"], "v2",
                reason="improve recipe")
        assert client.get(f"/api/skills/{sk['id']}",
                          headers=h).json()["content"] == "v2"
        audit = client.get(f"/api/rooms/{rid}/self-mod", headers=h).json()
        assert audit, "skill edit must be audited"
        client.post(f"/api/self-mod/{audit[0]['id']}/revert", headers=h)
        assert client.get(f"/api/skills/{sk['id']}",
                          headers=h).json()["content"] == "v1"

        # webhook -> escalation -> queen wake
        tok = "wh-" + "0" * 29
        from room_amd.db import queries as q
        with ldb as db:
            q.update_room(db, rid, webhook_token=tok)
        out = client.post(f"/api/hooks/queen/{tok}",
                          json={"message": "external event"})
        assert out.status_code == 200
        escs = client.get(f"/api/rooms/{rid}/escalations", headers=h).json()
        assert any("external event" in e["question"] for e in escs)

        # scheduled task executes through the GPU engine
        task = client.post("/api/tasks",
                           json={"name": "auto-task", "prompt": "say hi",
                                 "trigger_type": "manual", "room_id": rid},
                           headers=h).json()
        asyncio.run(runner.execute_task(task["id"]))
        runs = client.get(f"/api/tasks/{task['id']}/runs", headers=h).json()
        assert runs and runs[0]["status"] in ("completed", "failed")
        assert runs[0]["status"] == "completed", runs[0]
    finally:
        eng.shutdown()
