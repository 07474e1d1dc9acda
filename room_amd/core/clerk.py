"""Clerk — the system-wide assistant (reference: src/server/clerk-*.ts,
src/shared/clerk-tools.ts).

Chat with in-process tool execution (room CRUD/lifecycle, task creation,
messaging) plus the commentary engine that narrates cycle activity from the
event bus (clerk-commentary.ts: buffered entries, paced narration).
"""
from __future__ import annotations

import json
import time
from typing import Optional

from ..db import LockedDb
from ..db import queries as q
from ..engine.providers import execute_agent
from ..engine.types import AgentExecutionOptions, ToolCall, ToolDef
from . import room as room_mod

_S = {"type": "string"}
_I = {"type": "integer"}


def _obj(props, required=None):
    return {"type": "object", "properties": props, "required": required or []}


CLERK_TOOLS = [
    ToolDef("clerk_list_rooms", "List all rooms with status.", _obj({})),
    ToolDef("clerk_create_room", "Create a new room.",
            _obj({"name": _S, "goal": _S}, ["name"])),
    ToolDef("clerk_pause_room", "Pause a room.", _obj({"room_id": _I}, ["room_id"])),
    ToolDef("clerk_resume_room", "Resume a room.", _obj({"room_id": _I}, ["room_id"])),
    ToolDef("clerk_create_task", "Create a scheduled task.",
            _obj({"name": _S, "prompt": _S, "cron_expression": _S, "room_id": _I},
                 ["name", "prompt"])),
    ToolDef("clerk_send_message", "Send a message to a room's queen.",
            _obj({"room_id": _I, "body": _S}, ["room_id", "body"])),
    ToolDef("clerk_room_status", "Get a room's status summary.",
            _obj({"room_id": _I}, ["room_id"])),
    ToolDef("clerk_restart_room", "Restart a room (clears goals/decisions).",
            _obj({"room_id": _I}, ["room_id"])),
    ToolDef("clerk_delete_room", "Permanently delete a room.",
            _obj({"room_id": _I}, ["room_id"])),
    ToolDef("clerk_list_tasks", "List scheduled tasks (optionally per room).",
            _obj({"room_id": _I})),
    ToolDef("clerk_keeper_vote", "Cast the keeper's override vote on a "
            "decision.", _obj({"decision_id": _I, "vote": _S},
                              ["decision_id", "vote"])),
    ToolDef("clerk_answer_escalation", "Answer a pending escalation on the "
            "keeper's behalf.", _obj({"escalation_id": _I, "answer": _S},
                                     ["escalation_id", "answer"])),
]


def execute_clerk_tool(ldb: LockedDb, call: ToolCall) -> str:
    name, args = call.name, call.arguments
    try:
        with ldb as db:
            if name == "clerk_list_rooms":
                rooms = q.list_rooms(db)
                return json.dumps([{"id": r["id"], "name": r["name"],
                                    "status": r["status"]} for r in rooms])
            if name == "clerk_create_room":
                r = room_mod.create_room(db, args["name"], goal=args.get("goal"))
                return json.dumps({"room_id": r["id"]})
            if name == "clerk_pause_room":
                room_mod.pause_room(db, args["room_id"])
                return json.dumps({"paused": True})
            if name == "clerk_resume_room":
                room_mod.resume_room(db, args["room_id"])
                return json.dumps({"resumed": True})
            if name == "clerk_create_task":
                t = q.create_task(db, args["name"], args["prompt"],
                                  cron_expression=args.get("cron_expression"),
                                  room_id=args.get("room_id"))
                return json.dumps({"task_id": t["id"]})
            if name == "clerk_send_message":
                e = q.create_escalation(db, args["room_id"], args["body"])
                return json.dumps({"escalation_id": e["id"]})
            if name == "clerk_restart_room":
                room_mod.restart_room(db, args["room_id"])
                return json.dumps({"restarted": True})
            if name == "clerk_delete_room":
                room_mod.delete_room(db, args["room_id"])
                return json.dumps({"deleted": True})
            if name == "clerk_list_tasks":
                ts = q.list_tasks(db, room_id=args.get("room_id"))
                return json.dumps([{"id": t["id"], "name": t["name"],
                                    "status": t["status"],
                                    "trigger": t["trigger_type"]}
                                   for t in ts])
            if name == "clerk_keeper_vote":
                from . import quorum as quorum_mod
                d = quorum_mod.keeper_vote(db, args["decision_id"],
                                           args["vote"])
                return json.dumps({"decision_id": d["id"],
                                   "status": d["status"]})
            if name == "clerk_answer_escalation":
                q.answer_escalation(db, args["escalation_id"], args["answer"])
                return json.dumps({"answered": True})
            if name == "clerk_room_status":
                s = room_mod.get_room_status(db, args["room_id"])
                return json.dumps({"room": s["room"]["name"],
                                   "status": s["room"]["status"],
                                   "workers": len(s["workers"]),
                                   "active_goals": len(s["active_goals"])})
            return json.dumps({"error": f"unknown tool {name}"})
    except Exception as e:
        return json.dumps({"error": str(e)})


# Project-doc sync (reference: src/shared/clerk-profile-config.ts
# CLERK_PROJECT_DOC_SPECS + hash setting keys): configured docs are mirrored
# into clerk memory entities when their content hash changes, so the clerk
# can answer questions about the system from its own docs.
CLERK_DOC_SPECS = [
    {"entity": "Project README", "rel": "README.md",
     "hash_key": "clerk_project_doc_hash_readme"},
    {"entity": "Project Dashboard Source", "rel": "room_amd/server/dashboard.py",
     "hash_key": "clerk_project_doc_hash_dashboard"},
]
CLERK_DOC_CONTENT_MAX = 200_000
CLERK_DOC_SYNC_MIN_S = 60.0


def sync_project_docs(ldb: LockedDb, root: str | None = None,
                      min_interval_s: float = CLERK_DOC_SYNC_MIN_S) -> int:
    """Sync changed project docs into clerk memory entities; returns the
    number of docs refreshed. Hash-gated (no-op when nothing changed) and
    rate-limited by the last-sync timestamp setting."""
    import hashlib
    import time as _time
    from pathlib import Path

    base = Path(root) if root else Path(__file__).resolve().parents[2]
    now = _time.time()
    with ldb as db:
        last = q.get_setting(db, "clerk_project_doc_last_sync")
        if last and now - float(last) < min_interval_s:
            return 0
        q.set_setting(db, "clerk_project_doc_last_sync", str(now))
        synced = 0
        for spec in CLERK_DOC_SPECS:
            path = base / spec["rel"]
            try:
                content = path.read_text()[:CLERK_DOC_CONTENT_MAX]
            except OSError:
                continue
            digest = hashlib.sha256(content.encode()).hexdigest()[:16]
            if q.get_setting(db, spec["hash_key"]) == digest:
                continue
            ent = q.get_entity_by_name(db, spec["entity"])
            if ent is None:
                ent = q.create_entity(db, spec["entity"], entity_type="document",
                                      category="project_doc")
            q.add_observation(db, ent["id"], content, source="doc_sync")
            q.set_setting(db, spec["hash_key"], digest)
            synced += 1
    return synced


def clerk_model_chain(ldb: LockedDb, model: str = "stub") -> list[str]:
    """Model preference chain (reference clerk-profile-config.ts:
    DEFAULT_CLERK_MODEL → subscription → OpenAI → Anthropic fallbacks):
    clerk_model setting → requested model → stub (always available)."""
    with ldb as db:
        setting = q.get_setting(db, "clerk_model")
    chain = []
    for m in (setting, model, "stub"):
        if m and m not in chain:
            chain.append(m)
    return chain


def clerk_chat(ldb: LockedDb, content: str, memory=None,
               model: str = "stub") -> str:
    """One clerk chat turn with tool execution + usage accounting; walks
    the model fallback chain when a model is unavailable (reference
    clerk-profile.ts model fallback)."""
    with ldb as db:
        q.add_clerk_message(db, "user", content)

    chain = clerk_model_chain(ldb, model)
    result = None
    for m in chain:
        model = m
        try:
            result = execute_agent(AgentExecutionOptions(
                prompt=content, model=m,
                system_prompt="You are the Clerk — the keeper's system-wide "
                              "assistant. Manage rooms and tasks via tools; "
                              "answer concisely.",
                max_turns=5, tools=CLERK_TOOLS,
                tool_executor=lambda call: execute_clerk_tool(ldb, call)))
        except Exception as e:
            result = None
            last_err = str(e)
            continue
        if result.success or not _model_unavailable(result.error):
            break
    if result is None:
        from ..engine.types import AgentExecutionResult
        result = AgentExecutionResult(success=False, text="",
                                      error=last_err)
    reply = result.text or result.error or ""
    with ldb as db:
        q.add_clerk_message(db, "assistant", reply, source="chat")
        q.log_clerk_usage(db, "chat", model, result.input_tokens,
                          result.output_tokens, success=result.success)
    return reply


def _model_unavailable(error: str | None) -> bool:
    if not error:
        return False
    e = error.lower()
    return any(s in e for s in ("model not found", "no api key", "unavailable",
                                "not registered", "connection", "unreachable",
                                "no egress", "api key"))


class CommentaryEngine:
    """Narrates swarm activity (reference: clerk-commentary.ts).

    Full pacing semantics (clerk-commentary.ts:21-33):
    - active pace: a random 8–30 s interval between lines while the keeper
      is present (presence heartbeat within the last 90 s);
    - light pace: 2–3 h intervals when the keeper is away or the
      `clerk_commentary_mode` setting is 'light';
    - keeper-pause: any keeper chat message silences commentary until 60 s
      of keeper silence have passed;
    - `clerk_commentary` setting 'off' disables generation entirely;
    - buffer capped at 200 entries.
    """

    ACTIVE_MIN_S, ACTIVE_MAX_S = 8.0, 30.0
    LIGHT_MIN_S, LIGHT_MAX_S = 2 * 3600.0, 3 * 3600.0
    PRESENCE_TIMEOUT_S = 90.0
    SILENCE_THRESHOLD_S = 60.0

    def __init__(self, ldb: LockedDb, bus, model: str = "stub",
                 buffer_cap: int = 200, time_source=time.time):
        import random
        self.ldb = ldb
        self.bus = bus
        self.model = model
        self.buffer: list[str] = []
        self.buffer_cap = buffer_cap
        self.time = time_source
        self._rng = random.Random(0xC0FFEE)
        self._last_presence = 0.0
        self._last_keeper_msg = 0.0
        self._next_due = self.time() + self.ACTIVE_MIN_S
        self._unsub = bus.on("*", self._on_event)
        # back-compat: the runtime loop polls on this cadence
        self.pace_s = self.ACTIVE_MIN_S

    def stop(self) -> None:
        if self._unsub:
            self._unsub()
            self._unsub = None

    def _on_event(self, channel: str, event: dict) -> None:
        etype = event.get("type")
        if channel == "clerk":
            if etype == "presence":
                self._last_presence = self.time()
            elif etype in ("keeper_message", "typing"):
                self._last_keeper_msg = self.time()
            return
        if etype in ("cycle_finished", "decision", "escalation",
                     "run_finished"):
            self.buffer.append(f"{channel}: {etype}")
            if len(self.buffer) > self.buffer_cap:
                self.buffer = self.buffer[-self.buffer_cap:]

    # ------------------------------------------------------------- pacing

    def current_pace(self) -> str:
        """'active' | 'light' (mode setting + presence window)."""
        with self.ldb as db:
            mode = q.get_setting(db, "clerk_commentary_mode") or "auto"
        if mode == "light":
            return "light"
        present = self.time() - self._last_presence < self.PRESENCE_TIMEOUT_S
        return "active" if present else "light"

    def _schedule_next(self) -> None:
        if self.current_pace() == "active":
            lo, hi = self.ACTIVE_MIN_S, self.ACTIVE_MAX_S
        else:
            lo, hi = self.LIGHT_MIN_S, self.LIGHT_MAX_S
        self._next_due = self.time() + self._rng.uniform(lo, hi)

    def tick(self) -> Optional[str]:
        """Generate one commentary line if due (call from a runtime loop)."""
        now = self.time()
        with self.ldb as db:
            if (q.get_setting(db, "clerk_commentary") or "on") == "off":
                return None
        # keeper-pause: stay silent until 60 s after the last keeper message
        if now - self._last_keeper_msg < self.SILENCE_THRESHOLD_S:
            return None
        if not self.buffer or now < self._next_due:
            return None
        events = self.buffer[-20:]
        self.buffer = []
        self._schedule_next()
        result = execute_agent(AgentExecutionOptions(
            prompt="Recent swarm events:\n" + "\n".join(events)
                   + "\nGive one short, lively commentary line.",
            model=self.model,
            system_prompt="You are a sports commentator narrating an AI agent "
                          "swarm. One sentence.",
            max_turns=1, max_new_tokens=64))
        line = result.text or ""
        with self.ldb as db:
            q.add_clerk_message(db, "commentary", line, source="commentary")
            q.log_clerk_usage(db, "commentary", self.model, result.input_tokens,
                              result.output_tokens, success=result.success)
        self.bus.emit("clerk", "commentary", {"content": line})
        return line

    def stop(self) -> None:
        self._unsub()
