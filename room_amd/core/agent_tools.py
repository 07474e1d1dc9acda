"""Agent tool registry and in-process tool executor.

Role-split sets mirroring the reference's QUEEN_TOOLS (17) vs WORKER_TOOLS (10)
(src/shared/queen-tools.ts:348-369) and the in-process `executeQueenTool`
switch (queen-tools.ts:394-649). Tools run in-process against SQLite; no
subprocess/HTTP hop — the model generating the tool call runs on this GPU.
"""
from __future__ import annotations

import json
import sqlite3
from typing import Callable, Optional

from ..db import queries as q
from ..engine.types import ToolCall, ToolDef
from . import goals as goals_mod
from . import quorum as quorum_mod
from . import skills as skills_mod
from . import wallet as wallet_mod

# Wake callback registry: the agent-loop manager registers trigger_agent here
# so tools can wake workers without a circular import.
_wake_worker: Optional[Callable[[int], None]] = None
_wake_room: Optional[Callable[[int, Optional[int]], None]] = None


def register_wake_callbacks(wake_worker: Callable[[int], None],
                            wake_room: Callable[[int, Optional[int]], None]) -> None:
    global _wake_worker, _wake_room
    _wake_worker = wake_worker
    _wake_room = wake_room


def wake_worker(worker_id: int) -> None:
    if _wake_worker:
        _wake_worker(worker_id)


def wake_room_workers(room_id: int, exclude: int | None = None) -> None:
    if _wake_room:
        _wake_room(room_id, exclude)


def _obj(props: dict, required: list[str] | None = None) -> dict:
    return {"type": "object", "properties": props, "required": required or []}


_S = {"type": "string"}
_I = {"type": "integer"}
_N = {"type": "number"}
_B = {"type": "boolean"}

_COMMON_TOOLS = [
    ToolDef("room_remember", "Store a fact in room memory for teammates.",
            _obj({"name": _S, "content": _S, "category": _S}, ["name", "content"])),
    ToolDef("room_recall", "Search room memory (hybrid keyword + semantic).",
            _obj({"query": _S, "limit": _I}, ["query"])),
    ToolDef("room_save_wip", "Save work-in-progress to continue next cycle.",
            _obj({"wip": _S}, ["wip"])),
    ToolDef("room_object", "Object to an announced decision before it takes effect.",
            _obj({"decision_id": _I, "reason": _S}, ["decision_id", "reason"])),
    ToolDef("room_vote", "Cast a vote on an open decision.",
            _obj({"decision_id": _I, "vote": {"type": "string",
                  "enum": ["yes", "no", "abstain"]}, "reasoning": _S},
                 ["decision_id", "vote"])),
    ToolDef("room_send_message", "Send a message to the keeper or another room.",
            _obj({"to": _S, "subject": _S, "body": _S}, ["to", "body"])),
    ToolDef("room_complete_goal", "Mark a goal as completed.",
            _obj({"goal_id": _I, "observation": _S}, ["goal_id"])),
    ToolDef("room_update_goal_progress", "Report progress (0..1) on a goal.",
            _obj({"goal_id": _I, "progress": _N, "observation": _S},
                 ["goal_id", "progress"])),
    ToolDef("room_create_skill", "Save a reusable skill recipe.",
            _obj({"name": _S, "content": _S, "activation_context": _S,
                  "auto_activate": _B}, ["name", "content"])),
    ToolDef("room_escalate", "Escalate a question to the keeper.",
            _obj({"question": _S}, ["question"])),
]

_QUEEN_ONLY_TOOLS = [
    ToolDef("room_set_goal", "Create a goal (optionally under a parent goal).",
            _obj({"description": _S, "parent_goal_id": _I, "assigned_worker_id": _I},
                 ["description"])),
    ToolDef("room_delegate_task", "Assign a goal to a worker and wake them.",
            _obj({"description": _S, "worker_id": _I, "parent_goal_id": _I},
                 ["description", "worker_id"])),
    ToolDef("room_announce", "Announce a decision (auto-effective in 10 min "
            "unless a worker objects).",
            _obj({"proposal": _S, "decision_type": {
                "type": "string",
                "enum": ["strategy", "resource", "personnel", "rule_change",
                         "low_impact"]}}, ["proposal"])),
    ToolDef("room_create_worker", "Create a new worker in the room.",
            _obj({"name": _S, "role": _S, "system_prompt": _S}, ["name", "role"])),
    ToolDef("room_configure_room", "Update room configuration fields.",
            _obj({"goal": _S, "autonomy_mode": _S, "max_concurrent_tasks": _I})),
    ToolDef("room_wallet_balance", "Check room wallet balance.",
            _obj({"chain": _S, "token": _S})),
    ToolDef("room_send_token", "Send tokens from the room wallet.",
            _obj({"to_address": _S, "amount": _S, "chain": _S, "token": _S},
                 ["to_address", "amount"])),
    ToolDef("room_update_worker", "Update a worker's prompt, role or pacing.",
            _obj({"worker_id": _I, "system_prompt": _S, "role": _S,
                  "cycle_gap_ms": _I, "max_turns": _I}, ["worker_id"])),
    ToolDef("room_web_search", "Keyless web search for research.",
            _obj({"query": _S}, ["query"])),
    ToolDef("room_web_fetch", "Fetch a URL as readable text.",
            _obj({"url": _S}, ["url"])),
    ToolDef("room_browser", "Persistent browser session action "
            "(navigate/snapshot/click/type).",
            _obj({"session_id": _S, "action": _S, "url": _S, "selector": _S,
                  "text": _S}, ["session_id", "action"])),
]

QUEEN_TOOLS: list[ToolDef] = _QUEEN_ONLY_TOOLS + _COMMON_TOOLS
WORKER_TOOLS: list[ToolDef] = list(_COMMON_TOOLS)


def tools_for_role(role: str | None) -> list[ToolDef]:
    return QUEEN_TOOLS if role == "queen" else WORKER_TOOLS


def execute_agent_tool(db: sqlite3.Connection, room_id: int, worker_id: int,
                       call: ToolCall,
                       embed_fn: Callable[[str], list[float]] | None = None,
                       memory=None) -> str:
    """In-process tool dispatch. Returns a string result fed back to the model."""
    name, args = call.name, call.arguments
    try:
        if name == "room_update_worker":
            fields = {k: v for k, v in args.items()
                      if k in ("system_prompt", "role", "cycle_gap_ms",
                               "max_turns") and v is not None}
            w = q.update_worker(db, args["worker_id"], **fields)
            if w is None:
                return json.dumps({"error": "worker not found"})
            return json.dumps({"updated": w["id"], "fields": list(fields)})

        if name == "room_web_search":
            from .web_tools import web_search
            return json.dumps(web_search(args["query"]))[:4000]

        if name == "room_web_fetch":
            from .web_tools import web_fetch
            return json.dumps(web_fetch(args["url"]))[:6000]

        if name == "room_browser":
            from .web_tools import browser_action
            return json.dumps(browser_action(
                args["session_id"], args["action"], url=args.get("url"),
                selector=args.get("selector"), text=args.get("text")))[:6000]

        if name == "room_set_goal":
            g = q.create_goal(db, room_id, args["description"],
                              parent_goal_id=args.get("parent_goal_id"),
                              assigned_worker_id=args.get("assigned_worker_id"))
            return json.dumps({"goal_id": g["id"], "status": g["status"]})

        if name == "room_delegate_task":
            g = q.create_goal(db, room_id, args["description"],
                              parent_goal_id=args.get("parent_goal_id"),
                              assigned_worker_id=args["worker_id"])
            q.update_goal(db, g["id"], status="in_progress")
            q.log_room_activity(db, room_id, "delegation",
                                f"Delegated to worker #{args['worker_id']}: "
                                f"{args['description']}", actor_id=worker_id)
            wake_worker(args["worker_id"])
            return json.dumps({"goal_id": g["id"], "delegated_to": args["worker_id"]})

        if name == "room_announce":
            # duplicate check mirrors queen-tools.ts:439-457
            recent = q.list_room_decisions(db, room_id, limit=20)
            if any(d["proposal"] == args["proposal"]
                   and d["status"] in ("announced", "voting") for d in recent):
                return json.dumps({"error": "duplicate announcement"})
            d = quorum_mod.announce(db, room_id, worker_id, args["proposal"],
                                    args.get("decision_type", "low_impact"))
            wake_room_workers(room_id, exclude=worker_id)
            return json.dumps({"decision_id": d["id"], "status": d["status"],
                               "effective_at": d.get("effective_at")})

        if name == "room_object":
            d = quorum_mod.object_to(db, args["decision_id"], worker_id, args["reason"])
            return json.dumps({"decision_id": d["id"], "status": d["status"]})

        if name == "room_vote":
            v = quorum_mod.vote(db, args["decision_id"], worker_id, args["vote"],
                                args.get("reasoning"))
            return json.dumps({"vote_id": v["id"], "vote": v["vote"]})

        if name == "room_create_worker":
            from .constants import WORKER_ROLE_PRESETS
            preset = WORKER_ROLE_PRESETS.get(args.get("role", ""), {})
            prompt = args.get("system_prompt") or preset.get(
                "systemPromptPrefix", f"You are a {args.get('role','worker')}.")
            w = q.create_worker(db, args["name"], prompt, role=args.get("role"),
                                room_id=room_id,
                                cycle_gap_ms=preset.get("cycleGapMs"),
                                max_turns=preset.get("maxTurns"))
            q.log_room_activity(db, room_id, "worker",
                                f"Worker '{w['name']}' created ({w['role']})",
                                actor_id=worker_id)
            return json.dumps({"worker_id": w["id"], "name": w["name"]})

        if name == "room_remember":
            if memory is not None:
                # embeds + upserts the GPU store and the durable SQLite copy
                ent = q.get_entity_by_name(db, args["name"], room_id)
                if ent is None:
                    ent = q.create_entity(db, args["name"],
                                          category=args.get("category"),
                                          room_id=room_id)
                q.add_observation(db, ent["id"], args["content"],
                                  source=f"worker:{worker_id}")
                vec = memory.embed(f"{args['name']} {args['content']}")
                q.upsert_embedding(db, ent["id"], vec,
                                   memory.embedder.text_hash(args["content"]))
                memory.store.upsert(ent["id"], vec)
                return json.dumps({"entity_id": ent["id"]})
            ent = q.get_entity_by_name(db, args["name"], room_id)
            if ent is None:
                ent = q.create_entity(db, args["name"], category=args.get("category"),
                                      room_id=room_id)
            q.add_observation(db, ent["id"], args["content"], source=f"worker:{worker_id}")
            if embed_fn is not None:
                vec = embed_fn(f"{args['name']} {args['content']}")
                import hashlib
                h = hashlib.sha256(args["content"].encode()).hexdigest()[:16]
                q.upsert_embedding(db, ent["id"], vec, h)
            return json.dumps({"entity_id": ent["id"]})

        if name == "room_recall":
            if memory is not None:
                qvec = memory.embed(args["query"])
                semantic = memory.store.search(qvec, k=20)
                hits = q.hybrid_search(db, args["query"], qvec,
                                       limit=args.get("limit", 5), room_id=room_id,
                                       semantic_hits=semantic)
            else:
                vec = embed_fn(args["query"]) if embed_fn else None
                hits = q.hybrid_search(db, args["query"], vec,
                                       limit=args.get("limit", 5), room_id=room_id)
            return json.dumps([{"name": h["name"], "score": round(h["score"], 4),
                                "observations": h["observations"][:3]} for h in hits])

        if name == "room_save_wip":
            q.set_worker_wip(db, worker_id, args["wip"])
            return json.dumps({"saved": True})

        if name == "room_send_message":
            to = args.get("to", "keeper")
            if to == "keeper":
                e = q.create_escalation(db, room_id, args["body"],
                                        from_agent_id=worker_id)
                return json.dumps({"escalation_id": e["id"]})
            m = q.create_room_message(db, room_id, "outbound",
                                      args.get("subject", ""), args["body"],
                                      to_room_id=to)
            return json.dumps({"message_id": m["id"]})

        if name == "room_escalate":
            e = q.create_escalation(db, room_id, args["question"],
                                    from_agent_id=worker_id)
            q.log_room_activity(db, room_id, "escalation",
                                f"Escalated: {args['question']}", actor_id=worker_id)
            return json.dumps({"escalation_id": e["id"]})

        if name == "room_complete_goal":
            g = goals_mod.complete_goal(db, args["goal_id"],
                                        observation=args.get("observation"),
                                        worker_id=worker_id)
            return json.dumps({"goal_id": g["id"], "status": g["status"]})

        if name == "room_update_goal_progress":
            g = goals_mod.update_goal_progress(db, args["goal_id"], args["progress"],
                                               observation=args.get("observation"),
                                               worker_id=worker_id)
            return json.dumps({"goal_id": g["id"], "progress": g["progress"]})

        if name == "room_create_skill":
            s = skills_mod.create_agent_skill(
                db, room_id, args["name"], args["content"],
                activation_context=args.get("activation_context"),
                auto_activate=args.get("auto_activate", False),
                created_by_worker_id=worker_id)
            return json.dumps({"skill_id": s["id"], "version": s["version"]})

        if name == "room_configure_room":
            fields = {k: v for k, v in args.items()
                      if k in ("goal", "autonomy_mode", "max_concurrent_tasks")}
            if fields:
                q.update_room(db, room_id, **fields)
            return json.dumps({"updated": sorted(fields)})

        if name == "room_wallet_balance":
            bal = wallet_mod.get_on_chain_balance(
                db, room_id, chain=args.get("chain", "base"),
                token=args.get("token", "usdc"))
            return json.dumps(bal)

        if name == "room_send_token":
            res = wallet_mod.send_token(db, room_id, args["to_address"],
                                        args["amount"], chain=args.get("chain", "base"),
                                        token=args.get("token", "usdc"))
            return json.dumps(res)

        return json.dumps({"error": f"unknown tool: {name}"})
    except Exception as e:  # tool errors are fed back to the model, not raised
        return json.dumps({"error": str(e)})
