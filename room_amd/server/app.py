"""HTTP + WebSocket API server (reference surface: src/server/index.ts 1020 LoC
+ 19 route modules + ws.ts).

FastAPI/uvicorn replaces the reference's raw node:http server. Preserved
contract: Bearer dual-token auth with member RBAC, /api/auth/handshake
(localhost-only), the REST resource paths, webhook pass-through before auth
with a 30/min per-token limit, WS channel subscribe protocol
{type, channel, data, timestamp}, and the event-bus fan-out.
"""
from __future__ import annotations

import asyncio
import json
import os
import time
from collections import defaultdict, deque
from typing import Optional

from fastapi import (Body, Depends, FastAPI, HTTPException, Query, Request,
                     WebSocket, WebSocketDisconnect)
from fastapi.responses import HTMLResponse

from ..core import goals as goals_mod
from ..core import quorum as quorum_mod
from ..core import room as room_mod
from ..core import self_mod, wallet as wallet_mod
from ..core.constants import WEBHOOK_RATE_LIMIT_PER_MIN
from ..core.events import EventBus
from ..core.secret_store import encrypt_secret
from ..db import LockedDb
from ..db import queries as q
from .auth import ROLE_MEMBER, AuthManager, member_can_write


class ServerState:
    def __init__(self, ldb: LockedDb, auth: AuthManager, bus: EventBus,
                 loop_mgr=None, runner=None, memory=None, runtime=None):
        self.ldb = ldb
        self.auth = auth
        self.bus = bus
        self.loop_mgr = loop_mgr
        self.runner = runner
        self.memory = memory
        self.runtime = runtime
        self.started_at = time.time()
        self.webhook_hits: dict[str, deque] = defaultdict(deque)
        # strong refs for fire-and-forget task executions (asyncio keeps only
        # weak refs; an untracked create_task can be GC'd mid-run)
        self.bg_tasks: set = set()

    def spawn(self, coro) -> None:
        t = asyncio.create_task(coro)
        self.bg_tasks.add(t)
        t.add_done_callback(self.bg_tasks.discard)


def create_app(ldb: LockedDb, loop_mgr=None, runner=None, memory=None,
               auth: AuthManager | None = None,
               bus: EventBus | None = None) -> FastAPI:
    bus = bus or (loop_mgr.bus if loop_mgr else EventBus())
    auth = auth or AuthManager(skip_token_file=True)
    state = ServerState(ldb, auth, bus, loop_mgr, runner, memory)

    app = FastAPI(title="room_amd", version="0.1.0")
    app.state.ctx = state

    # malformed bodies and dangling references become structured 4xx, not
    # unhandled 500s (reference routes return 400/404 on bad payloads):
    # a KeyError from payload["field"] is a missing required field; an
    # IntegrityError is an insert referencing a deleted row
    import sqlite3 as _sqlite3

    from fastapi.responses import JSONResponse

    @app.exception_handler(KeyError)
    async def _missing_field(request, exc):
        return JSONResponse(status_code=400,
                            content={"detail": f"missing field: {exc}"})

    @app.exception_handler(_sqlite3.IntegrityError)
    async def _dangling_ref(request, exc):
        return JSONResponse(status_code=409,
                            content={"detail": f"constraint failed: {exc}"})

    # opt-in HTTP profiling (reference: QUOROOM_PROFILE_HTTP=1,
    # server/index.ts:289-320 — per-request hrtime, normalized endpoint
    # buckets, slow-request threshold log)
    import os as _os
    if _os.environ.get("ROOMAMD_PROFILE_HTTP") == "1":
        import logging
        _plog = logging.getLogger("room_amd.http_profile")
        state.http_profile = defaultdict(lambda: {"count": 0, "total_ms": 0.0,
                                                  "max_ms": 0.0})

        @app.middleware("http")
        async def _profile_http(request: Request, call_next):
            t0 = time.perf_counter()
            response = await call_next(request)
            ms = (time.perf_counter() - t0) * 1000
            route = request.scope.get("route")
            bucket = f"{request.method} {route.path if route else request.url.path}"
            b = state.http_profile[bucket]
            b["count"] += 1
            b["total_ms"] += ms
            b["max_ms"] = max(b["max_ms"], ms)
            if ms > 500:
                _plog.warning("slow request: %s took %.0f ms", bucket, ms)
            return response

    # ---------------------------------------------------------------- auth

    def get_role(request: Request) -> str:
        hdr = request.headers.get("authorization", "")
        token = hdr[7:] if hdr.lower().startswith("bearer ") else None
        role = auth.role_for(token)
        if role is None:
            raise HTTPException(401, "invalid or missing token")
        return role

    def require_write(request: Request, role: str) -> None:
        if role == ROLE_MEMBER:
            template = request.scope.get("route").path if request.scope.get("route") else ""
            template = (template.replace("{room_id}", "{room_id}")
                        .replace("{decision_id}", "{decision_id}"))
            if not member_can_write(request.method, template):
                raise HTTPException(403, "member tokens are read-only here")

    def rw(request: Request) -> str:
        role = get_role(request)
        if request.method not in ("GET", "HEAD", "OPTIONS"):
            require_write(request, role)
        return role

    # --------------------------------------------------------- auth routes

    @app.post("/api/auth/handshake")
    async def handshake(request: Request):
        # localhost-only (reference index.ts:502-522)
        host = request.client.host if request.client else ""
        if host not in ("127.0.0.1", "::1", "localhost", "testclient"):
            raise HTTPException(403, "handshake is localhost-only")
        token = auth.issue_user_token()
        return {"token": token, "role": "user"}

    @app.get("/api/auth/verify")
    async def verify(role: str = Depends(get_role)):
        return {"ok": True, "role": role}

    # ------------------------------------------------------------ webhooks
    # pass-through BEFORE auth (index.ts:601-608), 30/min per token

    def _webhook_limit(token: str) -> None:
        dq = state.webhook_hits[token]
        now = time.time()
        while dq and dq[0] < now - 60:
            dq.popleft()
        if len(dq) >= WEBHOOK_RATE_LIMIT_PER_MIN:
            raise HTTPException(429, "webhook rate limit")
        dq.append(now)

    @app.post("/api/hooks/task/{token}")
    async def hook_task(token: str, payload: dict = Body(default={})):
        _webhook_limit(token)
        with ldb as db:
            task = q.get_task_by_webhook_token(db, token)
        if task is None:
            raise HTTPException(404, "unknown webhook token")
        if runner is not None:
            state.spawn(runner.execute_task(task["id"]))
        return {"queued": True, "task_id": task["id"]}

    @app.post("/api/hooks/queen/{token}")
    async def hook_queen(token: str, payload: dict = Body(default={})):
        _webhook_limit(token)
        with ldb as db:
            room = q.get_room_by_webhook_token(db, token)
            if room is None:
                raise HTTPException(404, "unknown webhook token")
            if room["status"] != "active":
                raise HTTPException(409, f"room is {room['status']}")
            message = payload.get("message") or json.dumps(payload)[:2000]
            esc = q.create_escalation(db, room["id"], message)
        if loop_mgr is not None and room["queen_worker_id"]:
            loop_mgr.trigger_agent(room["queen_worker_id"])
        bus.emit(f"room:{room['id']}", "escalation", {"id": esc["id"]})
        return {"escalation_id": esc["id"]}

    # --------------------------------------------------------------- rooms

    @app.get("/api/rooms")
    async def list_rooms(role: str = Depends(get_role)):
        with ldb as db:
            return q.list_rooms(db)

    @app.post("/api/rooms")
    async def create_room(payload: dict = Body(...), role: str = Depends(rw)):
        if not payload.get("name"):
            raise HTTPException(400, "name required")
        from ..core.constants import (CHATGPT_DEFAULTS_BY_PLAN,
                                      QUEEN_DEFAULTS_BY_PLAN)
        with ldb as db:
            room = room_mod.create_room(
                db, payload["name"], goal=payload.get("goal"),
                worker_model=payload.get("worker_model", "qwen3-coder-30b"),
                queen_cycle_gap_ms=payload.get("queen_cycle_gap_ms"))
            # plan-aware queen pacing defaults (reference rooms.ts:131-147):
            # explicit gap in the payload wins; otherwise the keeper's
            # queen_model + plan settings pick the pacing map entry
            if payload.get("queen_cycle_gap_ms") is None:
                queen_model = q.get_setting(db, "queen_model")
                if queen_model == "codex":
                    raw = q.get_setting(db, "chatgpt_plan") or ""
                    plan = CHATGPT_DEFAULTS_BY_PLAN.get(
                        raw, CHATGPT_DEFAULTS_BY_PLAN["none"])
                else:
                    raw = q.get_setting(db, "claude_plan") or ""
                    plan = QUEEN_DEFAULTS_BY_PLAN.get(
                        raw, QUEEN_DEFAULTS_BY_PLAN["none"])
                q.update_room(db, room["id"],
                              queen_cycle_gap_ms=plan["queenCycleGapMs"],
                              queen_max_turns=plan["queenMaxTurns"])
                if queen_model:
                    q.update_worker(db, room["queen_worker_id"],
                                    model=queen_model)
                room = q.get_room(db, room["id"])
        bus.emit("rooms", "room_created", {"id": room["id"]})
        return room

    @app.get("/api/rooms/{room_id}")
    async def get_room(room_id: int, role: str = Depends(get_role)):
        with ldb as db:
            room = q.get_room(db, room_id)
        if room is None:
            raise HTTPException(404)
        return room

    @app.patch("/api/rooms/{room_id}")
    async def update_room(room_id: int, payload: dict = Body(...),
                          role: str = Depends(rw)):
        with ldb as db:
            room = q.update_room(db, room_id, **{
                k: v for k, v in payload.items()
                if k in ("name", "goal", "autonomy_mode", "max_concurrent_tasks",
                         "worker_model", "queen_cycle_gap_ms", "queen_max_turns",
                         "queen_quiet_from", "queen_quiet_until", "visibility",
                         "queen_nickname")})
        if room is None:
            raise HTTPException(404)
        return room

    @app.delete("/api/rooms/{room_id}")
    async def delete_room(room_id: int, role: str = Depends(rw)):
        if loop_mgr is not None:
            with ldb as db:
                for w in q.list_room_workers(db, room_id):
                    loop_mgr.stop_agent(w["id"])
        with ldb as db:
            room_mod.delete_room(db, room_id)
        return {"deleted": True}

    @app.get("/api/rooms/{room_id}/status")
    async def room_status(room_id: int, role: str = Depends(get_role)):
        with ldb as db:
            try:
                return room_mod.get_room_status(db, room_id)
            except ValueError:
                raise HTTPException(404)

    @app.post("/api/rooms/{room_id}/start")
    async def start_room(room_id: int, role: str = Depends(rw)):
        with ldb as db:
            room = room_mod.resume_room(db, room_id)
        if loop_mgr is not None and room["queen_worker_id"]:
            await loop_mgr.start_agent_loop(room_id, room["queen_worker_id"])
        bus.emit(f"room:{room_id}", "room_started", {})
        return room

    @app.post("/api/rooms/{room_id}/pause")
    async def pause_room(room_id: int, role: str = Depends(rw)):
        with ldb as db:
            room = room_mod.pause_room(db, room_id)
        return room

    @app.post("/api/rooms/{room_id}/stop")
    async def stop_room(room_id: int, role: str = Depends(rw)):
        if loop_mgr is not None:
            with ldb as db:
                workers = q.list_room_workers(db, room_id)
            for w in workers:
                loop_mgr.stop_agent(w["id"])
        with ldb as db:
            return room_mod.stop_room(db, room_id)

    @app.post("/api/rooms/{room_id}/restart")
    async def restart_room(room_id: int, role: str = Depends(rw)):
        with ldb as db:
            if q.get_room(db, room_id) is None:
                raise HTTPException(404, "room not found")
            return room_mod.restart_room(db, room_id)

    @app.get("/api/rooms/{room_id}/activity")
    async def room_activity(room_id: int, limit: int = 50,
                            role: str = Depends(get_role)):
        with ldb as db:
            return q.get_room_activity(db, room_id, limit=limit)

    @app.get("/api/rooms/{room_id}/cycles")
    async def room_cycles(room_id: int, limit: int = 50,
                          role: str = Depends(get_role)):
        with ldb as db:
            return q.list_room_cycles(db, room_id, limit=limit)

    @app.get("/api/cycles/{cycle_id}/logs")
    async def cycle_logs(cycle_id: int, after_seq: int = -1,
                         role: str = Depends(get_role)):
        with ldb as db:
            return q.get_cycle_logs(db, cycle_id, after_seq=after_seq)

    # ------------------------------------------------------------- workers

    @app.get("/api/rooms/{room_id}/workers")
    async def list_workers(room_id: int, role: str = Depends(get_role)):
        with ldb as db:
            return q.list_room_workers(db, room_id)

    @app.post("/api/rooms/{room_id}/workers")
    async def create_worker(room_id: int, payload: dict = Body(...),
                            role: str = Depends(rw)):
        from ..core.constants import WORKER_ROLE_PRESETS
        preset = WORKER_ROLE_PRESETS.get(payload.get("role", ""), {})
        with ldb as db:
            w = q.create_worker(
                db, payload["name"],
                payload.get("system_prompt") or preset.get("systemPromptPrefix", ""),
                role=payload.get("role"), room_id=room_id,
                model=payload.get("model"),
                cycle_gap_ms=payload.get("cycle_gap_ms") or preset.get("cycleGapMs"),
                max_turns=payload.get("max_turns") or preset.get("maxTurns"))
        bus.emit(f"room:{room_id}", "worker_created", {"id": w["id"]})
        return w

    @app.get("/api/workers/{worker_id}")
    async def get_worker(worker_id: int, role: str = Depends(get_role)):
        with ldb as db:
            w = q.get_worker(db, worker_id)
        if w is None:
            raise HTTPException(404)
        return w

    @app.patch("/api/workers/{worker_id}")
    async def update_worker(worker_id: int, payload: dict = Body(...),
                            role: str = Depends(rw)):
        with ldb as db:
            w = q.update_worker(db, worker_id, **{
                k: v for k, v in payload.items()
                if k in ("name", "role", "system_prompt", "description", "model",
                         "cycle_gap_ms", "max_turns", "wip")})
        if w is None:
            raise HTTPException(404)
        return w

    @app.delete("/api/workers/{worker_id}")
    async def delete_worker(worker_id: int, role: str = Depends(rw)):
        if loop_mgr is not None:
            loop_mgr.stop_agent(worker_id)
        with ldb as db:
            q.delete_worker(db, worker_id)
        return {"deleted": True}

    @app.post("/api/workers/{worker_id}/start")
    async def start_worker(worker_id: int, role: str = Depends(rw)):
        with ldb as db:
            w = q.get_worker(db, worker_id)
        if w is None:
            raise HTTPException(404)
        if loop_mgr is not None and w["room_id"]:
            await loop_mgr.start_agent_loop(w["room_id"], worker_id)
            loop_mgr.trigger_agent(worker_id)  # the MCP nudge semantics
        return {"started": True}

    @app.post("/api/workers/{worker_id}/stop")
    async def stop_worker(worker_id: int, role: str = Depends(rw)):
        if loop_mgr is not None:
            loop_mgr.stop_agent(worker_id)
        return {"stopped": True}

    # --------------------------------------------------------------- goals

    @app.get("/api/rooms/{room_id}/goals")
    async def list_goals(room_id: int, status: Optional[str] = None,
                         tree: bool = False, role: str = Depends(get_role)):
        with ldb as db:
            if tree:
                return goals_mod.get_goal_tree(db, room_id)
            return q.list_room_goals(db, room_id, status=status)

    @app.post("/api/rooms/{room_id}/goals")
    async def create_goal(room_id: int, payload: dict = Body(...),
                          role: str = Depends(rw)):
        with ldb as db:
            return q.create_goal(db, room_id, payload["description"],
                                 parent_goal_id=payload.get("parent_goal_id"),
                                 assigned_worker_id=payload.get("assigned_worker_id"))

    @app.patch("/api/goals/{goal_id}")
    async def update_goal(goal_id: int, payload: dict = Body(...),
                          role: str = Depends(rw)):
        with ldb as db:
            if payload.get("status") == "completed":
                return goals_mod.complete_goal(db, goal_id,
                                               observation=payload.get("observation"))
            if payload.get("status") == "abandoned":
                return goals_mod.abandon_goal(db, goal_id,
                                              reason=payload.get("reason"))
            if "progress" in payload:
                return goals_mod.update_goal_progress(
                    db, goal_id, payload["progress"],
                    observation=payload.get("observation"))
            g = q.update_goal(db, goal_id, **{
                k: v for k, v in payload.items()
                if k in ("description", "assigned_worker_id", "status")})
        if g is None:
            raise HTTPException(404)
        return g

    # ----------------------------------------------------------- decisions

    @app.get("/api/rooms/{room_id}/decisions")
    async def list_decisions(room_id: int, status: Optional[str] = None,
                             role: str = Depends(get_role)):
        with ldb as db:
            return q.list_room_decisions(db, room_id, status=status)

    @app.post("/api/rooms/{room_id}/decisions")
    async def announce(room_id: int, payload: dict = Body(...),
                       role: str = Depends(rw)):
        with ldb as db:
            d = quorum_mod.announce(db, room_id, payload.get("proposer_id"),
                                    payload["proposal"],
                                    payload.get("decision_type", "low_impact"),
                                    delay_minutes=payload.get("delay_minutes"))
        bus.emit(f"room:{room_id}", "decision", {"id": d["id"]})
        return d

    @app.post("/api/decisions/{decision_id}/object")
    async def object_decision(decision_id: int, payload: dict = Body(...),
                              role: str = Depends(rw)):
        with ldb as db:
            try:
                return quorum_mod.object_to(db, decision_id,
                                            payload.get("worker_id", 0),
                                            payload.get("reason", ""))
            except ValueError as e:
                raise HTTPException(409, str(e))

    @app.post("/api/decisions/{decision_id}/vote")
    async def vote(decision_id: int, payload: dict = Body(...),
                   role: str = Depends(rw)):
        if "worker_id" not in payload or "vote" not in payload:
            raise HTTPException(400, "worker_id and vote required")
        with ldb as db:
            if q.get_decision(db, decision_id) is None:
                raise HTTPException(404, "decision not found")
            try:
                return quorum_mod.vote(db, decision_id, payload["worker_id"],
                                       payload["vote"], payload.get("reasoning"))
            except ValueError as e:
                raise HTTPException(409, str(e))

    @app.post("/api/decisions/{decision_id}/keeper-vote")
    async def keeper_vote(decision_id: int, payload: dict = Body(...),
                          role: str = Depends(rw)):
        if "vote" not in payload:
            raise HTTPException(400, "vote required")
        with ldb as db:
            if q.get_decision(db, decision_id) is None:
                raise HTTPException(404, "decision not found")
            try:
                return quorum_mod.keeper_vote(db, decision_id, payload["vote"])
            except ValueError as e:
                raise HTTPException(409, str(e))

    @app.get("/api/decisions/{decision_id}/votes")
    async def decision_votes(decision_id: int, role: str = Depends(get_role)):
        with ldb as db:
            d = q.get_decision(db, decision_id)
            votes = q.get_votes(db, decision_id)
        # sealed ballot: redact values while the vote is open
        # (reference routes/decisions.ts:103-107)
        if d and d.get("sealed") and d["status"] == "voting":
            return [{**v, "vote": "sealed", "reasoning": None} for v in votes]
        return votes

    @app.get("/api/tasks")
    async def list_tasks(room_id: Optional[int] = None,
                         status: Optional[str] = None,
                         role: str = Depends(get_role)):
        with ldb as db:
            return q.list_tasks(db, room_id=room_id, status=status)

    @app.post("/api/tasks")
    async def create_task(payload: dict = Body(...), role: str = Depends(rw)):
        from ..core.cron import validate_cron
        import secrets as _secrets
        if payload.get("trigger_type", "cron") == "cron" \
                and payload.get("cron_expression") \
                and not validate_cron(payload["cron_expression"]):
            raise HTTPException(422, "invalid cron expression")
        webhook_token = (_secrets.token_hex(16)
                         if payload.get("trigger_type") == "webhook" else None)
        with ldb as db:
            return q.create_task(
                db, payload["name"], payload["prompt"],
                trigger_type=payload.get("trigger_type", "cron"),
                cron_expression=payload.get("cron_expression"),
                scheduled_at=payload.get("scheduled_at"),
                room_id=payload.get("room_id"), worker_id=payload.get("worker_id"),
                session_continuity=payload.get("session_continuity", False),
                max_runs=payload.get("max_runs"),
                description=payload.get("description"),
                webhook_token=webhook_token,
                timeout_minutes=payload.get("timeout_minutes"),
                max_turns=payload.get("max_turns"))

    @app.get("/api/tasks/{task_id}")
    async def get_task(task_id: int, role: str = Depends(get_role)):
        with ldb as db:
            t = q.get_task(db, task_id)
        if t is None:
            raise HTTPException(404)
        return t

    @app.patch("/api/tasks/{task_id}")
    async def update_task(task_id: int, payload: dict = Body(...),
                          role: str = Depends(rw)):
        with ldb as db:
            t = q.update_task(db, task_id, **{
                k: v for k, v in payload.items()
                if k in ("name", "description", "prompt", "cron_expression",
                         "status", "max_runs", "worker_id", "session_continuity",
                         "timeout_minutes", "max_turns", "scheduled_at")})
        if t is None:
            raise HTTPException(404)
        return t

    @app.delete("/api/tasks/{task_id}")
    async def delete_task(task_id: int, role: str = Depends(rw)):
        with ldb as db:
            q.delete_task(db, task_id)
        return {"deleted": True}

    @app.post("/api/tasks/{task_id}/run")
    async def run_task(task_id: int, role: str = Depends(rw)):
        if runner is None:
            raise HTTPException(503, "task runner not available")
        state.spawn(runner.execute_task(task_id))
        return {"queued": True}

    @app.get("/api/tasks/{task_id}/runs")
    async def task_runs(task_id: int, limit: int = 20,
                        role: str = Depends(get_role)):
        with ldb as db:
            return q.list_task_runs(db, task_id, limit=limit)

    @app.get("/api/runs/{run_id}")
    async def get_run(run_id: int, role: str = Depends(get_role)):
        with ldb as db:
            r = q.get_task_run(db, run_id)
        if r is None:
            raise HTTPException(404)
        return r

    @app.get("/api/runs/{run_id}/logs")
    async def run_logs(run_id: int, after_seq: int = -1,
                       role: str = Depends(get_role)):
        with ldb as db:
            return q.get_console_logs(db, run_id, after_seq=after_seq)

    # --------------------------------------------------------------- memory

    @app.get("/api/memory/search")
    async def memory_search(query: Optional[str] = None,
                            q_param: Optional[str] = Query(None, alias="q"),
                            room_id: Optional[int] = None,
                            roomId: Optional[int] = None,
                            limit: int = 5, role: str = Depends(get_role)):
        # the SPA client sends ?q= (client.ts:319); agent tools send ?query=
        text = query or q_param
        if not text:
            raise HTTPException(400, "q is required")
        rid = room_id if room_id is not None else roomId
        if memory is not None:
            return memory.recall(rid, text, limit=limit)
        with ldb as db:
            return q.hybrid_search(db, text, None, limit=limit, room_id=rid)

    @app.post("/api/memory/entities")
    async def remember(payload: dict = Body(...), role: str = Depends(rw)):
        """Accepts both the SPA client shape ({name, type, category, roomId},
        client.ts:313 — content optional) and the agent-tool shape
        ({name, content, room_id}). Returns the full entity row plus the
        entity_id alias older callers read."""
        room_id = payload.get("room_id", payload.get("roomId"))
        name = payload.get("name")
        if not name:
            raise HTTPException(400, "name is required")
        content = payload.get("content")
        obs = payload.get("observations") or ([content] if content else [])
        if memory is not None and content:
            eid = memory.remember(room_id, name, content,
                                  category=payload.get("category"))
        else:
            with ldb as db:
                ent = q.create_entity(db, name,
                                      entity_type=payload.get("type", "fact"),
                                      category=payload.get("category"),
                                      room_id=room_id, observations=obs)
                eid = ent["id"]
        with ldb as db:
            row = dict(q.get_entity(db, eid))
        row["entity_id"] = eid
        return row

    @app.get("/api/memory/entities/{entity_id}")
    async def get_entity(entity_id: int, role: str = Depends(get_role)):
        with ldb as db:
            e = q.get_entity(db, entity_id)
            if e is None:
                raise HTTPException(404)
            e = dict(e)
            e["observations"] = q.get_observations(db, entity_id)
        return e

    @app.delete("/api/memory/entities/{entity_id}")
    async def delete_entity(entity_id: int, role: str = Depends(rw)):
        with ldb as db:
            q.delete_entity(db, entity_id)
        if memory is not None:
            memory.store.remove(entity_id)
        return {"deleted": True}

    # --------------------------------------------------------------- skills

    @app.get("/api/rooms/{room_id}/skills")
    async def list_skills(room_id: int, role: str = Depends(get_role)):
        with ldb as db:
            return q.list_room_skills(db, room_id)

    @app.post("/api/rooms/{room_id}/skills")
    async def create_skill(room_id: int, payload: dict = Body(...),
                           role: str = Depends(rw)):
        with ldb as db:
            return q.create_skill(db, room_id, payload["name"], payload["content"],
                                  activation_context=payload.get("activation_context"),
                                  auto_activate=payload.get("auto_activate", False))

    @app.patch("/api/skills/{skill_id}")
    async def update_skill(skill_id: int, payload: dict = Body(...),
                           role: str = Depends(rw)):
        with ldb as db:
            s = q.update_skill(db, skill_id, content=payload.get("content"),
                               activation_context=payload.get("activation_context"),
                               auto_activate=payload.get("auto_activate"))
        if s is None:
            raise HTTPException(404)
        return s

    @app.delete("/api/skills/{skill_id}")
    async def delete_skill(skill_id: int, role: str = Depends(rw)):
        with ldb as db:
            q.delete_skill(db, skill_id)
        return {"deleted": True}

    # ---------------------------------------------------------- escalations

    @app.get("/api/rooms/{room_id}/escalations")
    async def list_escalations(room_id: int, status: Optional[str] = None,
                               role: str = Depends(get_role)):
        with ldb as db:
            return q.list_escalations(db, room_id, status=status)

    @app.post("/api/rooms/{room_id}/escalations")
    async def create_escalation(room_id: int, payload: dict = Body(...),
                                role: str = Depends(rw)):
        if not payload.get("question"):
            raise HTTPException(400, "question required")
        with ldb as db:
            e = q.create_escalation(db, room_id, payload["question"],
                                    from_agent_id=payload.get("from_agent_id"))
        bus.emit(f"room:{room_id}", "escalation", {"id": e["id"]})
        return e

    @app.post("/api/escalations/{escalation_id}/answer")
    async def answer_escalation(escalation_id: int, payload: dict = Body(...),
                                role: str = Depends(rw)):
        if "answer" not in payload:
            raise HTTPException(400, "answer required")
        with ldb as db:
            q.answer_escalation(db, escalation_id, payload["answer"])
            row = db.execute("SELECT * FROM escalations WHERE id = ?",
                             (escalation_id,)).fetchone()
        if row and loop_mgr is not None:
            with ldb as db:
                room = q.get_room(db, row["room_id"])
            if room and room.get("queen_worker_id"):
                loop_mgr.trigger_agent(room["queen_worker_id"])
        return row or {}

    # ------------------------------------------------------------- self-mod

    @app.get("/api/rooms/{room_id}/self-mod")
    async def self_mod_audit(room_id: int, role: str = Depends(get_role)):
        with ldb as db:
            return q.list_self_mod_audit(db, room_id)

    @app.post("/api/self-mod/{audit_id}/revert")
    async def self_mod_revert(audit_id: int, role: str = Depends(rw)):
        with ldb as db:
            try:
                return self_mod.revert_modification(db, audit_id)
            except ValueError as e:
                raise HTTPException(409, str(e))

    # --------------------------------------------------------------- wallet

    @app.get("/api/rooms/{room_id}/wallet")
    async def get_wallet(room_id: int, role: str = Depends(get_role)):
        with ldb as db:
            w = q.get_room_wallet(db, room_id)
        if w is None:
            raise HTTPException(404)
        return {k: v for k, v in w.items() if k != "private_key_encrypted"}

    @app.get("/api/rooms/{room_id}/wallet/transactions")
    async def wallet_txs(room_id: int, role: str = Depends(get_role)):
        with ldb as db:
            w = q.get_room_wallet(db, room_id)
            if w is None:
                raise HTTPException(404)
            return q.list_wallet_txs(db, w["id"])

    @app.post("/api/rooms/{room_id}/wallet/send")
    async def wallet_send(room_id: int, payload: dict = Body(...),
                          role: str = Depends(rw)):
        with ldb as db:
            try:
                return wallet_mod.send_token(
                    db, room_id, payload["to_address"], payload["amount"],
                    chain=payload.get("chain", "base"),
                    token=payload.get("token", "usdc"),
                    description=payload.get("description"))
            except ValueError as e:
                raise HTTPException(422, str(e))

    # ---------------------------------------------------------- credentials

    @app.get("/api/rooms/{room_id}/credentials")
    async def list_credentials(room_id: int, role: str = Depends(get_role)):
        with ldb as db:
            return q.list_credentials(db, room_id)  # values never exposed

    @app.post("/api/rooms/{room_id}/credentials")
    async def set_credential(room_id: int, payload: dict = Body(...),
                             role: str = Depends(rw)):
        if "name" not in payload or "value" not in payload:
            raise HTTPException(400, "name and value required")
        with ldb as db:
            row = q.set_credential(db, room_id, payload["name"],
                                   encrypt_secret(payload["value"]),
                                   cred_type=payload.get("type", "other"))
        return {k: v for k, v in row.items() if k != "value_encrypted"}

    @app.delete("/api/rooms/{room_id}/credentials/{name}")
    async def delete_credential(room_id: int, name: str, role: str = Depends(rw)):
        with ldb as db:
            q.delete_credential(db, room_id, name)
        return {"deleted": True}

    # ------------------------------------------------------- room messages

    @app.get("/api/rooms/{room_id}/messages")
    async def list_messages(room_id: int, role: str = Depends(get_role)):
        with ldb as db:
            return q.list_room_messages(db, room_id)

    @app.post("/api/rooms/{room_id}/messages")
    async def send_message(room_id: int, payload: dict = Body(...),
                           role: str = Depends(rw)):
        """SPA client shape {toRoomId, body, subject} (client.ts:770-771 →
        outbound, room-messages.ts validation) or agent shape
        {direction, body, subject, from_room_id}."""
        body_text = str(payload.get("body") or payload.get("content") or "").strip()
        if not body_text:
            raise HTTPException(400, "body is required")
        direction = payload.get("direction",
                                "outbound" if payload.get("toRoomId") else "inbound")
        with ldb as db:
            m = q.create_room_message(db, room_id, direction,
                                      payload.get("subject") or "(no subject)",
                                      body_text,
                                      from_room_id=payload.get("from_room_id"))
        bus.emit(f"room:{room_id}", "message", {"id": m["id"]})
        return m

    # ----------------------------------------------------------- chat/clerk

    @app.get("/api/rooms/{room_id}/chat")
    async def room_chat_history(room_id: int, limit: int = 50,
                                role: str = Depends(get_role)):
        with ldb as db:
            return q.list_chat_messages(db, room_id, limit=limit)

    @app.post("/api/rooms/{room_id}/chat")
    async def room_chat(room_id: int, payload: dict = Body(...),
                        role: str = Depends(rw)):
        if not payload.get("content"):
            raise HTTPException(400, "content required")
        with ldb as db:
            q.add_chat_message(db, room_id, "user", payload["content"])
            e = q.create_escalation(db, room_id, payload["content"])
        if loop_mgr is not None:
            with ldb as db:
                room = q.get_room(db, room_id)
            if room and room.get("queen_worker_id"):
                loop_mgr.trigger_agent(room["queen_worker_id"])
        return {"escalation_id": e["id"]}

    @app.get("/api/clerk/messages")
    async def clerk_messages(limit: int = 50, role: str = Depends(get_role)):
        with ldb as db:
            return q.list_clerk_messages(db, limit=limit)

    @app.post("/api/clerk/chat")
    async def clerk_chat(payload: dict = Body(...), role: str = Depends(rw)):
        from ..core.clerk import clerk_chat as do_chat
        if not payload.get("content"):
            raise HTTPException(400, "content required")
        # keeper message → commentary engine pauses for its silence window
        bus.emit("clerk", "keeper_message", {})
        reply = await asyncio.to_thread(do_chat, ldb, payload["content"],
                                        memory)
        bus.emit("clerk", "message", {"content": reply})
        return {"reply": reply}

    # ------------------------------------------------------------- settings

    @app.get("/api/settings")
    async def list_settings(role: str = Depends(get_role)):
        with ldb as db:
            return q.list_settings(db)

    @app.put("/api/settings/{key}")
    async def set_setting(key: str, payload: dict = Body(...),
                          role: str = Depends(rw)):
        with ldb as db:
            q.set_setting(db, key, str(payload.get("value", "")))
        return {"key": key, "value": payload.get("value")}

    # --------------------------------------------------------------- status

    @app.get("/api/status/http-profile")
    async def http_profile(role: str = Depends(get_role)):
        prof = getattr(state, "http_profile", None)
        if prof is None:
            return {"enabled": False,
                    "hint": "set ROOMAMD_PROFILE_HTTP=1 before boot"}
        return {"enabled": True, "endpoints": {
            k: {**v, "avg_ms": round(v["total_ms"] / max(v["count"], 1), 2)}
            for k, v in sorted(prof.items())}}

    @app.get("/api/status")
    async def status(role: str = Depends(get_role)):
        with ldb as db:
            rooms = q.list_rooms(db)
        out = {
            "version": "0.1.0",
            "uptime_s": int(time.time() - state.started_at),
            "rooms": len(rooms),
            "active_rooms": sum(1 for r in rooms if r["status"] == "active"),
            "running_loops": len(loop_mgr.running_loops) if loop_mgr else 0,
        }
        try:
            import torch
            if torch.cuda.is_available():
                free_b, total_b = torch.cuda.mem_get_info()
                out["gpu"] = {"name": torch.cuda.get_device_name(0),
                              "hbm_free_gb": round(free_b / 2**30, 1),
                              "hbm_total_gb": round(total_b / 2**30, 1)}
        except Exception:
            pass
        # update availability (updateChecker.ts surfaces this in /api/status)
        uc = getattr(state, "update_checker", None)
        if uc is None:
            from ..core.update_checker import UpdateChecker
            uc = state.update_checker = UpdateChecker(out["version"])
        out["update"] = uc.status()
        return out

    @app.get("/api/templates")
    async def list_templates(role: str = Depends(get_role)):
        from ..core.templates import list_templates as lt
        return lt()

    @app.post("/api/rooms/from-template")
    async def room_from_template(payload: dict = Body(...),
                                 role: str = Depends(rw)):
        from ..core.templates import instantiate_room_template
        if "template" not in payload or "name" not in payload:
            raise HTTPException(400, "template and name required")
        with ldb as db:
            try:
                room = instantiate_room_template(
                    db, payload["template"], payload["name"],
                    worker_model=payload.get("worker_model", "qwen3-coder-30b"))
            except ValueError as e:
                raise HTTPException(422, str(e))
        bus.emit("rooms", "room_created", {"id": room["id"]})
        return room

    @app.post("/api/rooms/{room_id}/prompts/export")
    async def export_prompts(room_id: int, role: str = Depends(rw)):
        from ..core.prompt_sync import export_worker_prompts
        with ldb as db:
            files = export_worker_prompts(db, room_id)
        return {"files": files}

    @app.post("/api/rooms/{room_id}/prompts/import")
    async def import_prompts(room_id: int, payload: dict = Body(default={}),
                             role: str = Depends(rw)):
        from ..core.prompt_sync import import_worker_prompts
        with ldb as db:
            return import_worker_prompts(db, room_id,
                                         force=payload.get("force", False))

    @app.get("/api/rooms/{room_id}/profile")
    async def room_profile(room_id: int, role: str = Depends(get_role)):
        from ..core.public_feed import get_public_room_profile
        with ldb as db:
            prof = get_public_room_profile(db, room_id)
        if prof is None:
            raise HTTPException(404, "room not public")
        return prof

    @app.get("/api/feed")
    async def public_feed(limit: int = 50, role: str = Depends(get_role)):
        from ..core.public_feed import get_public_feed
        with ldb as db:
            return get_public_feed(db, limit=limit)

    # ------------------------------------------- route-shape parity surface
    # The reference exposes several flat/detail routes alongside the nested
    # ones (routes/*.ts); these aliases + detail endpoints keep
    # a reference API client working unchanged.

    @app.get("/api/decisions/{decision_id}")
    async def decision_detail(decision_id: int, role: str = Depends(get_role)):
        with ldb as db:
            d = q.get_decision(db, decision_id)
            if d is None:
                raise HTTPException(404, "decision not found")
            return {**d, "votes": q.get_votes(db, decision_id)}

    @app.post("/api/decisions/{decision_id}/resolve")
    async def decision_resolve(decision_id: int, body: dict = Body(default={}),
                               role: str = Depends(rw)):
        with ldb as db:
            q.resolve_decision(db, decision_id, body.get("result", "approved"),
                               body.get("reason"))
            return q.get_decision(db, decision_id)

    @app.get("/api/goals/{goal_id}")
    async def goal_detail(goal_id: int, role: str = Depends(get_role)):
        with ldb as db:
            g = q.get_goal(db, goal_id)
            if g is None:
                raise HTTPException(404, "goal not found")
            return g

    @app.get("/api/goals/{goal_id}/subgoals")
    async def goal_subgoals(goal_id: int, role: str = Depends(get_role)):
        with ldb as db:
            return db.execute("SELECT * FROM goals WHERE parent_goal_id = ?"
                              " ORDER BY id", (goal_id,)).fetchall()

    @app.get("/api/goals/{goal_id}/updates")
    async def goal_updates(goal_id: int, role: str = Depends(get_role)):
        with ldb as db:
            return db.execute("SELECT * FROM goal_updates WHERE goal_id = ?"
                              " ORDER BY id DESC", (goal_id,)).fetchall()

    @app.post("/api/goals/{goal_id}/updates")
    async def goal_add_update(goal_id: int, body: dict = Body(...),
                              role: str = Depends(rw)):
        with ldb as db:
            if q.get_goal(db, goal_id) is None:
                raise HTTPException(404, "goal not found")
            q.add_goal_update(db, goal_id, body.get("observation", ""),
                              metric_value=body.get("progress"))
            return q.get_goal(db, goal_id)

    @app.delete("/api/goals/{goal_id}")
    async def goal_delete(goal_id: int, role: str = Depends(rw)):
        with ldb as db:
            db.execute("DELETE FROM goals WHERE id = ?", (goal_id,))
        return {"deleted": True}

    @app.get("/api/memory/entities")
    async def memory_entities(room_id: int | None = None, limit: int = 100,
                              role: str = Depends(get_role)):
        with ldb as db:
            if room_id is not None:
                return db.execute(
                    "SELECT * FROM entities WHERE room_id = ?"
                    " ORDER BY id DESC LIMIT ?", (room_id, limit)).fetchall()
            return db.execute("SELECT * FROM entities ORDER BY id DESC"
                              " LIMIT ?", (limit,)).fetchall()

    @app.patch("/api/memory/entities/{entity_id}")
    async def memory_update(entity_id: int, body: dict = Body(...),
                            role: str = Depends(rw)):
        with ldb as db:
            if "name" in body:
                db.execute("UPDATE entities SET name = ? WHERE id = ?",
                           (body["name"], entity_id))
            if "content" in body:
                q.add_observation(db, entity_id, body["content"])
            ent = q.get_entity(db, entity_id)
            if ent is None:
                raise HTTPException(404, "entity not found")
            return ent

    @app.post("/api/memory/relations")
    async def memory_relate(body: dict = Body(...),
                            role: str = Depends(rw)):
        if "from_entity" not in body or "to_entity" not in body:
            raise HTTPException(400, "from_entity and to_entity required")
        with ldb as db:
            if (q.get_entity(db, body["from_entity"]) is None
                    or q.get_entity(db, body["to_entity"]) is None):
                raise HTTPException(404, "entity not found")
            return q.create_relation(db, body["from_entity"],
                                     body["to_entity"],
                                     body.get("relation_type", "related_to"))

    @app.delete("/api/memory/relations/{relation_id}")
    async def memory_unrelate(relation_id: int,
                              role: str = Depends(rw)):
        with ldb as db:
            db.execute("DELETE FROM relations WHERE id = ?", (relation_id,))
        return {"deleted": True}

    @app.delete("/api/memory/observations/{obs_id}")
    async def memory_del_obs(obs_id: int, role: str = Depends(rw)):
        with ldb as db:
            db.execute("DELETE FROM observations WHERE id = ?", (obs_id,))
        return {"deleted": True}

    @app.get("/api/memory/stats")
    async def memory_stats(role: str = Depends(get_role)):
        with ldb as db:
            ents = db.execute("SELECT COUNT(*) AS n FROM entities").fetchone()["n"]
            obs = db.execute("SELECT COUNT(*) AS n FROM observations").fetchone()["n"]
            rels = db.execute("SELECT COUNT(*) AS n FROM relations").fetchone()["n"]
            emb = db.execute("SELECT COUNT(*) AS n FROM embeddings").fetchone()["n"]
        gpu_rows = (memory.store.size if memory is not None
                    and hasattr(memory, "store") else None)
        return {"entities": ents, "observations": obs, "relations": rels,
                "embeddings": emb, "gpu_index_rows": gpu_rows}

    @app.get("/api/messages/{message_id}")
    async def message_detail(message_id: int, role: str = Depends(get_role)):
        with ldb as db:
            m = db.execute("SELECT * FROM room_messages WHERE id = ?",
                           (message_id,)).fetchone()
            if m is None:
                raise HTTPException(404, "message not found")
            db.execute("UPDATE room_messages SET status = 'read' WHERE id = ?",
                       (message_id,))
            return m

    @app.post("/api/messages/{message_id}/reply")
    async def message_reply(message_id: int, body: dict = Body(...),
                            role: str = Depends(rw)):
        with ldb as db:
            m = db.execute("SELECT * FROM room_messages WHERE id = ?",
                           (message_id,)).fetchone()
            if m is None:
                raise HTTPException(404, "message not found")
            out = q.create_room_message(
                db, m["room_id"], "outbound",
                to_room_id=m["from_room_id"],
                subject=f"Re: {m['subject']}", body=body.get("body", ""))
            return out

    @app.delete("/api/messages/{message_id}")
    async def message_delete(message_id: int,
                             role: str = Depends(rw)):
        with ldb as db:
            db.execute("DELETE FROM room_messages WHERE id = ?", (message_id,))
        return {"deleted": True}

    @app.get("/api/rooms/queen-states")
    async def queen_states(role: str = Depends(get_role)):
        with ldb as db:
            rows = db.execute(
                "SELECT r.id AS room_id, r.name, w.id AS queen_id,"
                " w.agent_state FROM rooms r JOIN workers w"
                " ON w.id = r.queen_worker_id").fetchall()
        return rows

    @app.get("/api/rooms/{room_id}/queen")
    async def room_queen(room_id: int, role: str = Depends(get_role)):
        with ldb as db:
            room = q.get_room(db, room_id)
            if room is None:
                raise HTTPException(404, "room not found")
            return q.get_worker(db, room["queen_worker_id"])

    @app.post("/api/rooms/{room_id}/queen/start")
    async def queen_start(room_id: int, role: str = Depends(rw)):
        with ldb as db:
            room = q.get_room(db, room_id)
            if room is None:
                raise HTTPException(404, "room not found")
            qid = room["queen_worker_id"]
        if loop_mgr is not None:
            loop_mgr.resume_agent(qid)
        return {"started": qid}

    @app.post("/api/rooms/{room_id}/queen/stop")
    async def queen_stop(room_id: int, role: str = Depends(rw)):
        with ldb as db:
            room = q.get_room(db, room_id)
            if room is None:
                raise HTTPException(404, "room not found")
            qid = room["queen_worker_id"]
        if loop_mgr is not None:
            loop_mgr.pause_agent(qid)
        return {"stopped": qid}

    @app.get("/api/rooms/{room_id}/voter-health")
    async def voter_health(room_id: int, role: str = Depends(get_role)):
        with ldb as db:
            room = q.get_room(db, room_id)
            if room is None:
                raise HTTPException(404, "room not found")
            thr = (room.get("config") or {}).get("voterHealthThreshold", 0.5)
            return q.get_voter_health(db, room_id, threshold=thr)

    @app.get("/api/rooms/{room_id}/usage")
    async def room_usage(room_id: int, role: str = Depends(get_role)):
        with ldb as db:
            row = db.execute(
                "SELECT COUNT(*) AS cycles,"
                " COALESCE(SUM(input_tokens),0) AS input_tokens,"
                " COALESCE(SUM(output_tokens),0) AS output_tokens"
                " FROM worker_cycles WHERE room_id = ?", (room_id,)).fetchone()
        return row

    @app.get("/api/runs")
    async def runs_all(limit: int = 50, role: str = Depends(get_role)):
        with ldb as db:
            return db.execute("SELECT * FROM task_runs ORDER BY id DESC"
                              " LIMIT ?", (limit,)).fetchall()

    # flat aliases for nested collections (reference routes/workers.ts,
    # skills.ts, self-mod.ts, prompt-sync, tasks.ts shapes)
    @app.get("/api/workers")
    async def workers_all(room_id: int | None = None,
                          role: str = Depends(get_role)):
        with ldb as db:
            if room_id is not None:
                return q.list_room_workers(db, room_id)
            return db.execute("SELECT * FROM workers ORDER BY id").fetchall()

    @app.post("/api/workers")
    async def workers_create(body: dict = Body(...),
                             role: str = Depends(rw)):
        return await create_worker(body["room_id"], body, role)  # type: ignore

    @app.get("/api/skills")
    async def skills_all(room_id: int | None = None,
                         role: str = Depends(get_role)):
        with ldb as db:
            if room_id is not None:
                return q.list_room_skills(db, room_id)
            return db.execute("SELECT * FROM skills ORDER BY id").fetchall()

    @app.get("/api/skills/{skill_id}")
    async def skill_detail(skill_id: int, role: str = Depends(get_role)):
        with ldb as db:
            s = q.get_skill(db, skill_id)
            if s is None:
                raise HTTPException(404, "skill not found")
            return s

    @app.post("/api/skills")
    async def skills_create(body: dict = Body(...),
                            role: str = Depends(rw)):
        from ..core import skills as skills_mod
        with ldb as db:
            return skills_mod.create_agent_skill(
                db, body.get("room_id"), body["name"], body["content"],
                activation_context=body.get("activation_context"),
                auto_activate=body.get("auto_activate", False))

    @app.get("/api/self-mod/audit")
    async def self_mod_audit_all(room_id: int | None = None,
                                 role: str = Depends(get_role)):
        with ldb as db:
            if room_id is not None:
                return q.list_self_mod_audit(db, room_id)
            return db.execute("SELECT * FROM self_mod_audit ORDER BY id DESC"
                              " LIMIT 200").fetchall()

    @app.post("/api/self-mod/audit/{audit_id}/revert")
    async def self_mod_revert_alias(audit_id: int,
                                    role: str = Depends(rw)):
        from ..core import self_mod as self_mod_mod
        with ldb as db:
            try:
                return self_mod_mod.revert_modification(db, audit_id)
            except ValueError as e:
                raise HTTPException(404, str(e))

    @app.post("/api/tasks/{task_id}/pause")
    async def task_pause(task_id: int, role: str = Depends(rw)):
        with ldb as db:
            t_ = q.update_task(db, task_id, status="paused")
            if t_ is None:
                raise HTTPException(404, "task not found")
            return t_

    @app.post("/api/tasks/{task_id}/resume")
    async def task_resume(task_id: int, role: str = Depends(rw)):
        with ldb as db:
            t_ = q.update_task(db, task_id, status="active")
            if t_ is None:
                raise HTTPException(404, "task not found")
            return t_

    @app.post("/api/tasks/{task_id}/reset-session")
    async def task_reset_session(task_id: int,
                                 role: str = Depends(rw)):
        with ldb as db:
            t_ = q.update_task(db, task_id, session_id=None)
            if t_ is None:
                raise HTTPException(404, "task not found")
            return {"reset": t_["name"]}

    @app.post("/api/workers/prompts/export")
    async def prompts_export_all(role: str = Depends(rw)):
        from ..core.prompt_sync import export_worker_prompts
        with ldb as db:
            rooms = q.list_rooms(db)
            return {"exported": sum(
                len(export_worker_prompts(db, r["id"])) for r in rooms)}

    @app.post("/api/workers/prompts/import")
    async def prompts_import_all(body: dict = Body(default={}),
                                 role: str = Depends(rw)):
        from ..core.prompt_sync import import_worker_prompts
        with ldb as db:
            rooms = q.list_rooms(db)
            return {"imported": sum(
                len(import_worker_prompts(db, r["id"],
                                          force=body.get("force", False)))
                for r in rooms)}

    # clerk status/usage + contact/provider/local-model status (the reference
    # manages external providers and an Ollama sidecar here; the engine is
    # in-process so these report the native equivalent)
    @app.get("/api/clerk/status")
    async def clerk_status(role: str = Depends(get_role)):
        with ldb as db:
            n = db.execute("SELECT COUNT(*) AS n FROM chat_messages"
                           " WHERE room_id IS NULL").fetchone()["n"]
        return {"available": True, "model": "qwen3-coder-30b (in-process)",
                "messages": n}

    @app.get("/api/clerk/usage")
    async def clerk_usage(role: str = Depends(get_role)):
        with ldb as db:
            return db.execute(
                "SELECT source, model, SUM(input_tokens) AS input_tokens,"
                " SUM(output_tokens) AS output_tokens, COUNT(*) AS calls"
                " FROM clerk_usage GROUP BY source, model").fetchall()

    @app.post("/api/clerk/reset")
    async def clerk_reset(role: str = Depends(rw)):
        with ldb as db:
            db.execute("DELETE FROM chat_messages WHERE room_id IS NULL")
        return {"reset": True}

    @app.post("/api/clerk/typing")
    async def clerk_typing(role: str = Depends(get_role)):
        bus.emit("clerk", "typing", {})
        return {"ok": True}

    @app.post("/api/clerk/presence")
    async def clerk_presence(role: str = Depends(get_role)):
        bus.emit("clerk", "presence", {})
        return {"ok": True}

    @app.get("/api/contacts/status")
    async def contacts_status(role: str = Depends(get_role)):
        import os as _os
        with ldb as db:
            email = q.get_setting(db, "keeper_email")
            tg = q.get_setting(db, "keeper_telegram_chat_id")
        return {"email": {"configured": bool(email or
                                             _os.environ.get("ROOMAMD_KEEPER_EMAIL")),
                          "address": email},
                "telegram": {"configured": bool(tg)}}

    @app.get("/api/providers/status")
    async def providers_status(role: str = Depends(get_role)):
        import torch as _torch

        from ..engine.cloud_providers import get_model_auth_status
        return {"providers": [{
            "id": "local", "name": "room_amd in-process engine",
            "model": "qwen3-coder-30b", "ready": True,
            "gpu": _torch.cuda.is_available()}],
            "auth": get_model_auth_status(ldb)}

    @app.get("/api/local-model/status")
    async def local_model_status(role: str = Depends(get_role)):
        import torch as _torch
        ok = _torch.cuda.is_available()
        return {"installed": True, "running": ok,
                "model": "qwen3-coder-30b",
                "backend": "in-process CDNA4 engine (no sidecar)",
                "detail": None if ok else "no GPU visible in this process"}

    @app.get("/api/rooms/{room_id}/badges")
    async def room_badges(room_id: int, role: str = Depends(get_role)):
        """Unread counters for the UI (reference rooms.ts:228-247)."""
        with ldb as db:
            if q.get_room(db, room_id) is None:
                raise HTTPException(404, "room not found")
            pend = db.execute(
                "SELECT COUNT(*) AS n FROM escalations WHERE room_id = ?"
                " AND status = 'pending' AND to_agent_id IS NULL"
                " AND from_agent_id IS NOT NULL", (room_id,)).fetchone()["n"]
            unread = db.execute(
                "SELECT COUNT(*) AS n FROM room_messages WHERE room_id = ?"
                " AND status = 'unread'", (room_id,)).fetchone()["n"]
            votes = db.execute(
                "SELECT COUNT(*) AS n FROM quorum_decisions WHERE room_id = ?"
                " AND status IN ('voting', 'announced')",
                (room_id,)).fetchone()["n"]
        return {"room_id": room_id, "pending_escalations": pend,
                "unread_messages": unread, "active_votes": votes}

    @app.post("/api/rooms/{room_id}")
    async def update_room_alias(room_id: int, payload: dict = Body(...),
                                role: str = Depends(rw)):
        return await update_room(room_id, payload, role)  # type: ignore

    @app.get("/api/settings/{key}")
    async def get_setting_route(key: str, role: str = Depends(get_role)):
        with ldb as db:
            return {"key": key, "value": q.get_setting(db, key)}

    # keeper contact verification (reference routes/contacts.ts; email codes
    # go through the notification relay — SMTP if configured, outbox.jsonl
    # otherwise — and telegram binding is cloud-mediated)
    def _hash_email_code(email: str, code: str) -> str:
        import hashlib
        return hashlib.sha256(f"{email}:{code}".encode()).hexdigest()

    @app.post("/api/contacts/email/start")
    async def contacts_email_start(body: dict = Body(...),
                                   role: str = Depends(rw)):
        import secrets as _secrets
        from datetime import datetime, timedelta

        from ..core.notifications import notify_keeper
        email = (body.get("email") or "").strip().lower()
        if "@" not in email or "." not in email.split("@")[-1]:
            raise HTTPException(400, "Valid email is required")
        with ldb as db:
            if (q.get_setting(db, "contact_email") == email
                    and q.get_setting(db, "contact_email_verified_at")):
                return {"ok": True, "already_verified": True, "email": email}
            code = f"{_secrets.randbelow(1_000_000):06d}"
            expires = (datetime.now() + timedelta(minutes=15)).isoformat()
            q.set_setting(db, "contact_email", email)
            q.set_setting(db, "contact_email_verified_at", "")
            q.set_setting(db, "contact_email_code_hash",
                          _hash_email_code(email, code))
            q.set_setting(db, "contact_email_code_expires_at", expires)
        notify_keeper("Verification code",
                      f"Your room_amd verification code is {code}",
                      channel="email")
        return {"ok": True, "sent_to": email, "expires_at": expires}

    @app.post("/api/contacts/email/verify")
    async def contacts_email_verify(body: dict = Body(...),
                                    role: str = Depends(rw)):
        from datetime import datetime
        code = (body.get("code") or "").strip()
        with ldb as db:
            email = q.get_setting(db, "contact_email") or ""
            expect = q.get_setting(db, "contact_email_code_hash")
            expires = q.get_setting(db, "contact_email_code_expires_at") or ""
            if not expect or _hash_email_code(email, code) != expect:
                raise HTTPException(400, "Invalid code")
            if expires and expires < datetime.now().isoformat():
                raise HTTPException(400, "Code expired")
            q.set_setting(db, "contact_email_verified_at",
                          datetime.now().isoformat())
            q.set_setting(db, "contact_email_code_hash", "")
            q.set_setting(db, "keeper_email", email)
        return {"ok": True, "verified": True, "email": email}

    @app.post("/api/contacts/email/resend")
    async def contacts_email_resend(role: str = Depends(rw)):
        with ldb as db:
            email = q.get_setting(db, "contact_email")
        if not email:
            raise HTTPException(400, "No pending email verification")
        return await contacts_email_start({"email": email}, role)  # type: ignore

    @app.post("/api/contacts/telegram/start")
    async def contacts_telegram_start(role: str = Depends(rw)):
        from ..core import cloud_sync
        if cloud_sync.cloud_api() is None:
            raise HTTPException(503, "Telegram binding needs the cloud relay "
                                     "(ROOMAMD_CLOUD_API)")
        return {"ok": False, "error": "cloud telegram verification relay "
                                      "not reachable"}

    @app.post("/api/contacts/telegram/check")
    async def contacts_telegram_check(role: str = Depends(rw)):
        with ldb as db:
            chat = q.get_setting(db, "keeper_telegram_chat_id")
        return {"connected": bool(chat), "chat_id": chat}

    @app.post("/api/contacts/telegram/disconnect")
    async def contacts_telegram_disconnect(role: str = Depends(rw)):
        with ldb as db:
            q.set_setting(db, "keeper_telegram_chat_id", "")
        return {"ok": True}

    @app.get("/api/rooms/{room_id}/cloud-id")
    async def room_cloud_id(room_id: int, role: str = Depends(get_role)):
        from ..core import cloud_sync
        tok = cloud_sync.load_room_tokens().get(str(room_id))
        return {"room_id": room_id, "registered": tok is not None,
                "cloud_id": f"room-{room_id}" if tok else None}

    @app.get("/api/rooms/{room_id}/network")
    async def room_network(room_id: int, role: str = Depends(get_role)):
        from ..core import cloud_sync
        if cloud_sync.cloud_api() is None:
            return {"rooms": [], "note": "cloud not configured"}
        return {"rooms": []}

    @app.get("/api/settings/referral")
    async def settings_referral(role: str = Depends(get_role)):
        with ldb as db:
            return {"code": q.get_setting(db, "keeper_referral_code")}

    # -------------------------------------------------------------- UI stub

    # ------------------------------------------------- server self-restart
    # (reference index.ts:526-576: /api/server/restart re-execs the process;
    # /api/server/update-restart stages the update first; the download
    # endpoint triggers staging without restarting)

    def _schedule_restart(delay_s: float = 0.5) -> None:
        import sys as _sys
        import threading as _threading

        def _go():
            time.sleep(delay_s)
            os.execv(_sys.executable, [_sys.executable, "-m", "room_amd.cli",
                                       "serve"])
        t = _threading.Thread(target=_go, daemon=True)
        t.start()

    @app.post("/api/server/restart")
    async def server_restart(role: str = Depends(rw)):
        if os.environ.get("ROOMAMD_ALLOW_RESTART") != "1":
            # in-place re-exec is opt-in (test/CI processes must not re-exec)
            return {"ok": False,
                    "error": "restart disabled (set ROOMAMD_ALLOW_RESTART=1)"}
        _schedule_restart()
        return {"ok": True, "restarting": True}

    def _update_checker():
        uc = getattr(state, "update_checker", None)
        if uc is None:
            from ..core.update_checker import UpdateChecker
            from .auth import data_dir
            uc = state.update_checker = UpdateChecker(
                "0.1.0", data_dir=data_dir())
        return uc

    @app.get("/api/status/update/download")
    async def update_download(role: str = Depends(rw)):
        uc = _update_checker()
        if uc.latest_version is None:
            uc.check()
        return uc.stage_update()

    @app.post("/api/server/update-restart")
    async def server_update_restart(role: str = Depends(rw)):
        uc = _update_checker()
        staged = uc.stage_update()
        if not staged.get("staged"):
            return {"ok": False, "staged": staged}
        if os.environ.get("ROOMAMD_ALLOW_RESTART") != "1":
            return {"ok": False, "staged": staged,
                    "error": "restart disabled (set ROOMAMD_ALLOW_RESTART=1)"}
        _schedule_restart()
        return {"ok": True, "staged": staged, "restarting": True}

    # -------------------------------------------- UI client-contract routes
    # Completes the endpoint set the reference SPA's typed client calls
    # (docs/ui_client_contract.json, extracted from src/ui/lib/client.ts).

    @app.get("/api/credentials/{cred_id}")
    async def get_credential_by_id(cred_id: int, role: str = Depends(get_role)):
        with ldb as db:
            row = db.execute(
                "SELECT id, room_id, name, type, provided_by, created_at"
                " FROM credentials WHERE id = ?", (cred_id,)).fetchone()
        if row is None:
            raise HTTPException(404, "credential not found")
        return row

    @app.delete("/api/credentials/{cred_id}")
    async def delete_credential_by_id(cred_id: int, role: str = Depends(rw)):
        with ldb as db:
            db.execute("DELETE FROM credentials WHERE id = ?", (cred_id,))
        return {"ok": True}

    @app.post("/api/rooms/{room_id}/credentials/validate")
    async def validate_credential(room_id: int, payload: dict = Body(...),
                                  role: str = Depends(rw)):
        """Shape check only — there is no egress to hit provider APIs
        (reference validates against the provider; we validate format)."""
        name = str(payload.get("name", ""))
        value = str(payload.get("value", ""))
        ok = bool(name) and len(value) >= 8 and not value.isspace()
        return {"ok": ok,
                "error": None if ok else "value too short to be a credential"}

    @app.get("/api/memory/entities/{entity_id}/observations")
    async def entity_observations(entity_id: int, role: str = Depends(get_role)):
        with ldb as db:
            if q.get_entity(db, entity_id) is None:
                raise HTTPException(404, "entity not found")
            return q.get_observations(db, entity_id)

    @app.post("/api/memory/entities/{entity_id}/observations")
    async def add_entity_observation(entity_id: int, payload: dict = Body(...),
                                     role: str = Depends(rw)):
        with ldb as db:
            if q.get_entity(db, entity_id) is None:
                raise HTTPException(404, "entity not found")
            oid = q.add_observation(db, entity_id, payload["content"],
                                    source=payload.get("source", "keeper"))
            return {"id": oid, "entity_id": entity_id,
                    "content": payload["content"]}

    @app.get("/api/memory/entities/{entity_id}/relations")
    async def entity_relations(entity_id: int, role: str = Depends(get_role)):
        with ldb as db:
            return db.execute(
                "SELECT * FROM relations WHERE from_entity = ? OR to_entity = ?"
                " ORDER BY id", (entity_id, entity_id)).fetchall()

    @app.post("/api/rooms/{room_id}/messages/{message_id}/read")
    async def mark_message_read(room_id: int, message_id: int,
                                role: str = Depends(rw)):
        with ldb as db:
            q.mark_room_message_read(db, message_id)
        return {"ok": True}

    @app.post("/api/rooms/{room_id}/messages/read-all")
    async def mark_all_messages_read(room_id: int, role: str = Depends(rw)):
        with ldb as db:
            cur = db.execute(
                "UPDATE room_messages SET status = 'read' WHERE room_id = ?"
                " AND status = 'unread'", (room_id,))
        return {"ok": True, "marked": cur.rowcount}

    @app.post("/api/escalations/{escalation_id}/resolve")
    async def resolve_escalation(escalation_id: int, payload: dict = Body(...),
                                 role: str = Depends(rw)):
        """Reference routes/escalations.ts:40-63: answer + wake sender and
        queen. Same semantics as our /answer alias."""
        answer = payload.get("answer")
        if not answer or not isinstance(answer, str):
            raise HTTPException(400, "answer is required")
        with ldb as db:
            row = db.execute("SELECT * FROM escalations WHERE id = ?",
                             (escalation_id,)).fetchone()
            if row is None:
                raise HTTPException(404, "Escalation not found")
            q.answer_escalation(db, escalation_id, answer)
            updated = db.execute("SELECT * FROM escalations WHERE id = ?",
                                 (escalation_id,)).fetchone()
            room = q.get_room(db, row["room_id"])
        bus.emit(f"room:{row['room_id']}", "escalation:resolved", dict(updated))
        if loop_mgr is not None:
            if row["from_agent_id"]:
                loop_mgr.trigger_agent(row["from_agent_id"])
            if room and room.get("queen_worker_id") and \
                    room["queen_worker_id"] != row["from_agent_id"]:
                loop_mgr.trigger_agent(room["queen_worker_id"])
        return updated

    # ------------------------------------------------- wallet (UI contract)

    @app.get("/api/rooms/{room_id}/wallet/summary")
    async def wallet_summary(room_id: int, role: str = Depends(get_role)):
        """Revenue summary (db-queries.ts:2210-2231 shape)."""
        with ldb as db:
            w = q.get_room_wallet(db, room_id)
            if w is None:
                return {"totalIncome": 0, "totalExpenses": 0, "netProfit": 0,
                        "transactionCount": 0}
            inc = db.execute(
                "SELECT COALESCE(SUM(CAST(amount AS REAL)), 0) AS t FROM"
                " wallet_transactions WHERE wallet_id = ? AND type IN"
                " ('receive', 'fund')", (w["id"],)).fetchone()["t"]
            exp = db.execute(
                "SELECT COALESCE(SUM(CAST(amount AS REAL)), 0) AS t FROM"
                " wallet_transactions WHERE wallet_id = ? AND type IN"
                " ('send', 'purchase')", (w["id"],)).fetchone()["t"]
            cnt = db.execute(
                "SELECT COUNT(*) AS c FROM wallet_transactions WHERE"
                " wallet_id = ?", (w["id"],)).fetchone()["c"]
        return {"totalIncome": inc, "totalExpenses": exp,
                "netProfit": inc - exp, "transactionCount": cnt}

    @app.get("/api/rooms/{room_id}/wallet/balance")
    async def wallet_balance(room_id: int, role: str = Depends(get_role)):
        """On-chain balance. No egress here → the zero-balance shape the
        reference returns when every chain RPC fails (wallet.ts:103-148)."""
        import datetime as _dt
        with ldb as db:
            w = q.get_room_wallet(db, room_id)
        if w is None:
            return None
        try:
            with ldb as db:
                bal = wallet_mod.get_on_chain_balance(db, room_id)
        except Exception:
            bal = None  # RPC unreachable (offline) → zero-balance shape
        if bal and "totalBalance" in bal:
            return bal
        by_chain = ({bal["chain"]: bal["balance"]}
                    if bal and bal.get("balance") is not None else {})
        return {"totalBalance": sum(v for v in by_chain.values() if v) or 0,
                "byChain": by_chain, "address": w["address"],
                "fetchedAt": _dt.datetime.now(_dt.timezone.utc)
                .isoformat().replace("+00:00", "Z")}

    @app.get("/api/rooms/{room_id}/wallet/onramp-url")
    async def wallet_onramp_url(room_id: int, role: str = Depends(get_role)):
        with ldb as db:
            w = q.get_room_wallet(db, room_id)
        if w is None:
            raise HTTPException(400, "Room has no wallet")
        raise HTTPException(503, "On-ramp unavailable")  # cloud-mediated; offline

    @app.post("/api/rooms/{room_id}/wallet/withdraw")
    async def wallet_withdraw(room_id: int, payload: dict = Body(...),
                              role: str = Depends(rw)):
        """Reference wallet.ts:162-230 validation, then the send path."""
        import re as _re
        to = str(payload.get("to", "")).strip()
        amount = str(payload.get("amount", "")).strip()
        if not to or not amount:
            raise HTTPException(400, "Missing required fields: to, amount")
        if not _re.fullmatch(r"0x[0-9a-fA-F]{40}", to):
            raise HTTPException(400, "Invalid address")
        try:
            parsed = float(amount)
        except ValueError:
            raise HTTPException(400, "Invalid amount")
        if not (parsed > 0):
            raise HTTPException(400, "Invalid amount")
        with ldb as db:
            if q.get_room(db, room_id) is None:
                raise HTTPException(404, "Room not found")
            if q.get_room_wallet(db, room_id) is None:
                raise HTTPException(400, "Room has no wallet")
            try:
                return wallet_mod.send_token(
                    db, room_id, to, amount,
                    chain=payload.get("chain", "base"),
                    token=payload.get("token", "usdc"),
                    description="keeper withdrawal")
            except ValueError as e:
                raise HTTPException(422, str(e))

    # --------------------------------------------------- clerk (UI contract)

    @app.post("/api/clerk/api-key")
    async def clerk_api_key(payload: dict = Body(...), role: str = Depends(rw)):
        provider = str(payload.get("provider", "")).strip()
        key = str(payload.get("key", "")).strip()
        if provider not in ("openai_api", "anthropic_api", "gemini_api"):
            raise HTTPException(
                400, "provider must be openai_api, anthropic_api, or gemini_api")
        if not key:
            raise HTTPException(400, "key is required")
        with ldb as db:
            q.set_setting(db, f"clerk_api_key_{provider}", encrypt_secret(key))
        # no egress: stored without provider-side validation
        return {"ok": True, "validated": False,
                "note": "stored; provider validation requires network"}

    @app.put("/api/clerk/settings")
    async def clerk_settings(payload: dict = Body(...), role: str = Depends(rw)):
        """clerk.ts:561-596: clerk model doubles as default queen model."""
        with ldb as db:
            if "model" in payload:
                model = str(payload["model"])
                q.set_setting(db, "clerk_model", model)
                q.set_setting(db, "queen_model", model)
            if "commentary" in payload:
                q.set_setting(db, "clerk_commentary",
                              "on" if payload["commentary"] else "off")
            if "pace" in payload:
                q.set_setting(db, "clerk_commentary_pace", str(payload["pace"]))
        return {"ok": True}

    # ------------------------------------------------------- update checker

    @app.post("/api/status/check-update")
    async def check_update(role: str = Depends(rw)):
        uc = getattr(state, "update_checker", None)
        if uc is None:
            from ..core.update_checker import UpdateChecker
            uc = state.update_checker = UpdateChecker("0.1.0")
        return uc.check()

    # ------------------------------------- provider / installer (UI contract)
    # The reference manages external CLI/Ollama installs with streamed
    # sessions (provider-install.ts, local-model.ts). This build's engine is
    # in-process — these endpoints preserve the UI flow with sessions that
    # complete immediately and a status that reports the built-in engine.

    def _session_registry() -> dict:
        reg = getattr(state, "provider_sessions", None)
        if reg is None:
            reg = state.provider_sessions = {"next_id": 1, "sessions": {}}
        return reg

    def _mk_session(kind: str, provider: str) -> dict:
        reg = _session_registry()
        sid = reg["next_id"]
        reg["next_id"] += 1
        sess = {"id": sid, "kind": kind, "provider": provider,
                "status": "completed",
                "lines": [{"id": 1, "stream": "system",
                           "text": "in-process engine: nothing to install",
                           "timestamp": time.strftime("%Y-%m-%dT%H:%M:%SZ")}],
                "exitCode": 0}
        reg["sessions"][sid] = sess
        return sess

    @app.post("/api/providers/{provider}/connect")
    async def provider_connect(provider: str, role: str = Depends(rw)):
        return {"session": _mk_session("connect", provider)}

    @app.post("/api/providers/{provider}/disconnect")
    async def provider_disconnect(provider: str, role: str = Depends(rw)):
        return {"ok": True}

    @app.post("/api/providers/{provider}/install")
    async def provider_install(provider: str, role: str = Depends(rw)):
        return {"session": _mk_session("install", provider)}

    @app.get("/api/providers/{provider}/session")
    async def provider_session(provider: str, role: str = Depends(get_role)):
        reg = _session_registry()
        for s in reversed(list(reg["sessions"].values())):
            if s["provider"] == provider and s["kind"] == "connect":
                return {"session": s}
        return {"session": None}

    @app.get("/api/providers/{provider}/install-session")
    async def provider_install_session(provider: str,
                                       role: str = Depends(get_role)):
        reg = _session_registry()
        for s in reversed(list(reg["sessions"].values())):
            if s["provider"] == provider and s["kind"] == "install":
                return {"session": s}
        return {"session": None}

    @app.get("/api/providers/sessions/{sid}")
    async def provider_session_by_id(sid: int, role: str = Depends(get_role)):
        s = _session_registry()["sessions"].get(sid)
        if s is None:
            raise HTTPException(404, "session not found")
        return {"session": s}

    @app.get("/api/providers/install-sessions/{sid}")
    async def provider_install_session_by_id(sid: int,
                                             role: str = Depends(get_role)):
        return await provider_session_by_id(sid, role)  # type: ignore

    @app.post("/api/providers/sessions/{sid}/cancel")
    async def provider_session_cancel(sid: int, role: str = Depends(rw)):
        s = _session_registry()["sessions"].get(sid)
        if s is None:
            raise HTTPException(404, "session not found")
        if s["status"] in ("starting", "running"):
            s["status"] = "canceled"
        return {"session": s}

    @app.post("/api/providers/install-sessions/{sid}/cancel")
    async def provider_install_cancel(sid: int, role: str = Depends(rw)):
        return await provider_session_cancel(sid, role)  # type: ignore

    @app.post("/api/local-model/install")
    async def local_model_install(role: str = Depends(rw)):
        return {"started": False, "alreadyInstalled": True,
                "session": _mk_session("install", "local")}

    @app.get("/api/local-model/install-session")
    async def local_model_install_session(role: str = Depends(get_role)):
        reg = _session_registry()
        for s in reversed(list(reg["sessions"].values())):
            if s["provider"] == "local":
                return {"session": s}
        return {"session": None}

    @app.post("/api/local-model/install-sessions/{sid}/cancel")
    async def local_model_install_cancel(sid: int, role: str = Depends(rw)):
        return await provider_session_cancel(sid, role)  # type: ignore

    @app.post("/api/local-model/apply-all")
    async def local_model_apply_all(role: str = Depends(rw)):
        """Set every room's worker_model to the in-process engine default."""
        with ldb as db:
            rooms = q.list_rooms(db)
            for r in rooms:
                q.update_room(db, r["id"], worker_model="qwen3-coder-30b")
        return {"applied": len(rooms), "model": "qwen3-coder-30b"}

    @app.get("/", response_class=HTMLResponse)
    async def index():
        from .dashboard import DASHBOARD_HTML
        return DASHBOARD_HTML

    # ------------------------------------------------------------ websocket

    @app.websocket("/ws")
    async def ws_endpoint(ws: WebSocket, token: str = Query(default="")):
        if auth.role_for(token) is None:
            await ws.close(code=4401)
            return
        await ws.accept()
        subscribed: set[str] = set()
        queue: asyncio.Queue = asyncio.Queue(maxsize=1000)
        loop = asyncio.get_running_loop()

        def on_event(channel: str, event: dict) -> None:
            if channel in subscribed or "*" in subscribed:
                try:
                    loop.call_soon_threadsafe(queue.put_nowait, event)
                except RuntimeError:
                    pass

        unsub = bus.on("*", lambda ch, ev: on_event(ch, ev))

        async def sender():
            while True:
                try:
                    event = await asyncio.wait_for(queue.get(), timeout=30)
                    await ws.send_json(event)
                except asyncio.TimeoutError:
                    await ws.send_json({"type": "ping", "channel": "system",
                                        "data": None,
                                        "timestamp": int(time.time() * 1000)})

        send_task = asyncio.create_task(sender())
        try:
            while True:
                msg = await ws.receive_json()
                if msg.get("type") == "subscribe":
                    subscribed.add(msg.get("channel", ""))
                elif msg.get("type") == "unsubscribe":
                    subscribed.discard(msg.get("channel", ""))
                elif msg.get("type") == "ping":
                    await ws.send_json({"type": "pong", "channel": "system",
                                        "data": None,
                                        "timestamp": int(time.time() * 1000)})
        except WebSocketDisconnect:
            pass
        finally:
            send_task.cancel()
            unsub()

    # Starlette matches routes in registration order; float the late-added
    # literal paths above their parameterized siblings (/api/rooms/{room_id}
    # would otherwise shadow /api/rooms/queen-states with a 422).
    literals = [r for r in app.router.routes
                if getattr(r, "path", "") in ("/api/rooms/queen-states", "/api/settings/referral")]
    for r in literals:
        app.router.routes.remove(r)
        app.router.routes.insert(0, r)

    return app
