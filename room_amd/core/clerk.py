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


def clerk_chat(ldb: LockedDb, content: str, memory=None,
               model: str = "stub") -> str:
    """One clerk chat turn with tool execution + usage accounting."""
    with ldb as db:
        q.add_clerk_message(db, "user", content)
        model_setting = q.get_setting(db, "clerk_model")
        if model_setting:
            model = model_setting

    result = execute_agent(AgentExecutionOptions(
        prompt=content, model=model,
        system_prompt="You are the Clerk — the keeper's system-wide assistant. "
                      "Manage rooms and tasks via tools; answer concisely.",
        max_turns=5, tools=CLERK_TOOLS,
        tool_executor=lambda call: execute_clerk_tool(ldb, call)))
    reply = result.text or result.error or ""
    with ldb as db:
        q.add_clerk_message(db, "assistant", reply, source="chat")
        q.log_clerk_usage(db, "chat", model, result.input_tokens,
                          result.output_tokens, success=result.success)
    return reply


class CommentaryEngine:
    """Narrates swarm activity (reference: clerk-commentary.ts — subscribes to
    cycle events on the bus, buffers up to 200 entries, paced generation)."""

    def __init__(self, ldb: LockedDb, bus, model: str = "stub",
                 pace_s: float = 8.0, buffer_cap: int = 200):
        self.ldb = ldb
        self.bus = bus
        self.model = model
        self.pace_s = pace_s
        self.buffer: list[str] = []
        self.buffer_cap = buffer_cap
        self._last_emit = 0.0
        self._unsub = bus.on("*", self._on_event)

    def _on_event(self, channel: str, event: dict) -> None:
        if event.get("type") in ("cycle_finished", "decision", "escalation",
                                 "run_finished"):
            self.buffer.append(f"{channel}: {event['type']}")
            if len(self.buffer) > self.buffer_cap:
                self.buffer = self.buffer[-self.buffer_cap:]

    def tick(self) -> Optional[str]:
        """Generate one commentary line if due (call from a runtime loop)."""
        now = time.time()
        if not self.buffer or now - self._last_emit < self.pace_s:
            return None
        events = self.buffer[-20:]
        self.buffer = []
        self._last_emit = now
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
