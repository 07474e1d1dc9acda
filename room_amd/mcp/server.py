"""MCP stdio server (reference: src/mcp/server.ts + 17 tool modules, 76 tools).

Implements the Model Context Protocol over stdio as JSON-RPC 2.0
(initialize / tools/list / tools/call). Tools share the same SQLite file as
the HTTP server (ROOMAMD_DB_PATH env, reference QUOROOM_DB_PATH); agent wakes
cross the process boundary via an HTTP nudge using the token in
~/.roomamd/api.token (reference mcp/nudge.ts:14-43).
"""
from __future__ import annotations

import json
import sys
from typing import Any, Callable

from ..core import goals as goals_mod
from ..core import quorum as quorum_mod
from ..core import room as room_mod
from ..core import self_mod, skills as skills_mod, wallet as wallet_mod
from ..db import LockedDb, connect
from ..db import queries as q

PROTOCOL_VERSION = "2024-11-05"
SERVER_INFO = {"name": "room-amd", "version": "0.1.0"}


def nudge_worker(worker_id: int) -> bool:
    """Cross-process wake: POST /api/workers/{id}/start on the local server."""
    import urllib.request

    from ..server.auth import data_dir
    try:
        port = int((data_dir() / "api.port").read_text().strip())
        token = (data_dir() / "api.token").read_text().strip()
        req = urllib.request.Request(
            f"http://127.0.0.1:{port}/api/workers/{worker_id}/start",
            method="POST", headers={"Authorization": f"Bearer {token}"},
            data=b"{}")
        urllib.request.urlopen(req, timeout=3)
        return True
    except Exception:
        return False  # fail-silent like the reference nudge


def _server_port() -> int | None:
    from ..server.auth import data_dir
    try:
        return int((data_dir() / "api.port").read_text().strip())
    except Exception:
        return None


def _server_post(path: str) -> bool:
    """Authenticated POST to the local HTTP server; False if unreachable."""
    import urllib.request

    from ..server.auth import data_dir
    try:
        port = _server_port()
        token = (data_dir() / "api.token").read_text().strip()
        req = urllib.request.Request(
            f"http://127.0.0.1:{port}{path}", method="POST",
            headers={"Authorization": f"Bearer {token}"}, data=b"{}")
        urllib.request.urlopen(req, timeout=3)
        return True
    except Exception:
        return False


class McpServer:
    def __init__(self, ldb: LockedDb, memory=None, nudge=nudge_worker):
        self.ldb = ldb
        self.memory = memory
        self.nudge = nudge
        self.tools: dict[str, tuple[dict, Callable[[dict], Any]]] = {}
        self._register_all()

    # ------------------------------------------------------------ registry

    def tool(self, name: str, description: str, schema: dict):
        def deco(fn):
            self.tools[name] = (
                {"name": name, "description": description,
                 "inputSchema": {"type": "object", "properties": schema.get("p", {}),
                                 "required": schema.get("r", [])}}, fn)
            return fn
        return deco

    def _register_all(self) -> None:
        S = {"type": "string"}
        I = {"type": "integer"}
        B = {"type": "boolean"}
        N = {"type": "number"}
        t = self.tool
        ldb = self.ldb

        # ---- rooms
        @t("room_list_rooms", "List all rooms.", {})
        def _(args):
            with ldb as db:
                return [{"id": r["id"], "name": r["name"], "status": r["status"],
                         "goal": r["goal"]} for r in q.list_rooms(db)]

        @t("room_create_room", "Create a room with a queen worker.",
           {"p": {"name": S, "goal": S, "worker_model": S}, "r": ["name"]})
        def _(args):
            with ldb as db:
                r = room_mod.create_room(db, args["name"], goal=args.get("goal"),
                                         worker_model=args.get("worker_model",
                                                               "qwen3-coder-30b"))
            return {"room_id": r["id"], "queen_worker_id": r["queen_worker_id"]}

        @t("room_get_status", "Room status summary.",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                s = room_mod.get_room_status(db, args["room_id"])
            return {"room": s["room"]["name"], "status": s["room"]["status"],
                    "workers": [{"id": w["id"], "name": w["name"],
                                 "state": w["agent_state"]} for w in s["workers"]],
                    "active_goals": len(s["active_goals"]),
                    "pending_decisions": len(s["pending_decisions"])}

        @t("room_pause_room", "Pause a room.", {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                room_mod.pause_room(db, args["room_id"])
            return {"paused": True}

        @t("room_resume_room", "Resume a room.", {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                room_mod.resume_room(db, args["room_id"])
            return {"resumed": True}

        @t("room_restart_room", "Restart a room (clears goals/decisions).",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                room_mod.restart_room(db, args["room_id"])
            return {"restarted": True}

        # ---- workers
        @t("room_list_workers", "List a room's workers.",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                return q.list_room_workers(db, args["room_id"])

        @t("room_create_worker", "Create a worker in a room.",
           {"p": {"room_id": I, "name": S, "role": S, "system_prompt": S},
            "r": ["room_id", "name"]})
        def _(args):
            from ..core.constants import WORKER_ROLE_PRESETS
            preset = WORKER_ROLE_PRESETS.get(args.get("role", ""), {})
            with ldb as db:
                w = q.create_worker(db, args["name"],
                                    args.get("system_prompt")
                                    or preset.get("systemPromptPrefix", ""),
                                    role=args.get("role"), room_id=args["room_id"],
                                    cycle_gap_ms=preset.get("cycleGapMs"),
                                    max_turns=preset.get("maxTurns"))
            return {"worker_id": w["id"]}

        @t("room_start_worker", "Wake a worker (cross-process nudge).",
           {"p": {"worker_id": I}, "r": ["worker_id"]})
        def _(args):
            ok = self.nudge(args["worker_id"])
            return {"nudged": ok}

        @t("room_save_wip", "Save a worker's work-in-progress.",
           {"p": {"worker_id": I, "wip": S}, "r": ["worker_id", "wip"]})
        def _(args):
            with ldb as db:
                q.set_worker_wip(db, args["worker_id"], args["wip"])
            return {"saved": True}

        # ---- goals
        @t("room_set_goal", "Create a goal in a room.",
           {"p": {"room_id": I, "description": S, "parent_goal_id": I,
                  "assigned_worker_id": I}, "r": ["room_id", "description"]})
        def _(args):
            with ldb as db:
                g = q.create_goal(db, args["room_id"], args["description"],
                                  parent_goal_id=args.get("parent_goal_id"),
                                  assigned_worker_id=args.get("assigned_worker_id"))
            return {"goal_id": g["id"]}

        @t("room_delegate_task", "Assign a goal to a worker and wake them.",
           {"p": {"room_id": I, "description": S, "worker_id": I},
            "r": ["room_id", "description", "worker_id"]})
        def _(args):
            with ldb as db:
                g = q.create_goal(db, args["room_id"], args["description"],
                                  assigned_worker_id=args["worker_id"])
                q.update_goal(db, g["id"], status="in_progress")
            self.nudge(args["worker_id"])
            return {"goal_id": g["id"]}

        @t("room_goal_tree", "The room's hierarchical goal tree.",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                return goals_mod.get_goal_tree(db, args["room_id"])

        @t("room_complete_goal", "Mark a goal completed.",
           {"p": {"goal_id": I, "observation": S}, "r": ["goal_id"]})
        def _(args):
            with ldb as db:
                g = goals_mod.complete_goal(db, args["goal_id"],
                                            observation=args.get("observation"))
            return {"goal_id": g["id"], "status": g["status"]}

        @t("room_update_goal_progress", "Report goal progress 0..1.",
           {"p": {"goal_id": I, "progress": N}, "r": ["goal_id", "progress"]})
        def _(args):
            with ldb as db:
                g = goals_mod.update_goal_progress(db, args["goal_id"],
                                                   args["progress"])
            return {"goal_id": g["id"], "progress": g["progress"]}

        # ---- quorum
        @t("room_announce", "Announce a decision (auto-effective unless objected).",
           {"p": {"room_id": I, "proposer_id": I, "proposal": S,
                  "decision_type": S}, "r": ["room_id", "proposal"]})
        def _(args):
            with ldb as db:
                d = quorum_mod.announce(db, args["room_id"],
                                        args.get("proposer_id"), args["proposal"],
                                        args.get("decision_type", "low_impact"))
            return {"decision_id": d["id"], "status": d["status"]}

        @t("room_object", "Object to an announced decision.",
           {"p": {"decision_id": I, "worker_id": I, "reason": S},
            "r": ["decision_id", "worker_id", "reason"]})
        def _(args):
            with ldb as db:
                d = quorum_mod.object_to(db, args["decision_id"],
                                         args["worker_id"], args["reason"])
            return {"status": d["status"]}

        @t("room_vote", "Vote on an open decision.",
           {"p": {"decision_id": I, "worker_id": I, "vote": S},
            "r": ["decision_id", "worker_id", "vote"]})
        def _(args):
            with ldb as db:
                v = quorum_mod.vote(db, args["decision_id"], args["worker_id"],
                                    args["vote"])
            return {"vote_id": v["id"]}

        @t("room_keeper_vote", "Keeper override vote on a decision.",
           {"p": {"decision_id": I, "vote": S}, "r": ["decision_id", "vote"]})
        def _(args):
            with ldb as db:
                d = quorum_mod.keeper_vote(db, args["decision_id"], args["vote"])
            return {"status": d["status"]}

        @t("room_list_decisions", "List a room's decisions.",
           {"p": {"room_id": I, "status": S}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                return q.list_room_decisions(db, args["room_id"],
                                             status=args.get("status"))

        # ---- memory
        @t("room_remember", "Store a fact in room memory.",
           {"p": {"room_id": I, "name": S, "content": S, "category": S},
            "r": ["name", "content"]})
        def _(args):
            if self.memory is not None:
                eid = self.memory.remember(args.get("room_id"), args["name"],
                                           args["content"],
                                           category=args.get("category"))
                return {"entity_id": eid}
            with ldb as db:
                ent = q.create_entity(db, args["name"],
                                      category=args.get("category"),
                                      room_id=args.get("room_id"),
                                      observations=[args["content"]])
            return {"entity_id": ent["id"]}

        @t("room_recall", "Hybrid search over room memory.",
           {"p": {"room_id": I, "query": S, "limit": I}, "r": ["query"]})
        def _(args):
            if self.memory is not None:
                hits = self.memory.recall(args.get("room_id"), args["query"],
                                          limit=args.get("limit", 5))
            else:
                with ldb as db:
                    hits = q.hybrid_search(db, args["query"], None,
                                           limit=args.get("limit", 5),
                                           room_id=args.get("room_id"))
            return [{"name": h["name"], "score": round(h["score"], 4),
                     "observations": h["observations"][:3]} for h in hits]

        # ---- skills
        @t("room_create_skill", "Save a reusable skill.",
           {"p": {"room_id": I, "name": S, "content": S, "activation_context": S,
                  "auto_activate": B}, "r": ["name", "content"]})
        def _(args):
            with ldb as db:
                s = skills_mod.create_agent_skill(
                    db, args.get("room_id"), args["name"], args["content"],
                    activation_context=args.get("activation_context"),
                    auto_activate=args.get("auto_activate", False))
            return {"skill_id": s["id"]}

        @t("room_list_skills", "List a room's skills.",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                return [{"id": s["id"], "name": s["name"], "version": s["version"]}
                        for s in q.list_room_skills(db, args["room_id"])]

        @t("room_modify_skill", "Audited self-modification of a skill.",
           {"p": {"room_id": I, "worker_id": I, "skill_id": I, "content": S,
                  "reason": S}, "r": ["worker_id", "skill_id", "content"]})
        def _(args):
            with ldb as db:
                out = self_mod.perform_skill_modification(
                    db, args.get("room_id"), args["worker_id"], args["skill_id"],
                    args["content"], reason=args.get("reason"))
            return {"audit_id": out["audit_id"]}

        @t("room_revert_modification", "Revert an audited self-modification.",
           {"p": {"audit_id": I}, "r": ["audit_id"]})
        def _(args):
            with ldb as db:
                return self_mod.revert_modification(db, args["audit_id"])

        # ---- tasks / scheduler
        @t("room_create_task", "Create a scheduled task (cron/once/manual/webhook).",
           {"p": {"name": S, "prompt": S, "trigger_type": S, "cron_expression": S,
                  "scheduled_at": S, "room_id": I, "max_runs": I},
            "r": ["name", "prompt"]})
        def _(args):
            with ldb as db:
                t_ = q.create_task(db, args["name"], args["prompt"],
                                   trigger_type=args.get("trigger_type", "cron"),
                                   cron_expression=args.get("cron_expression"),
                                   scheduled_at=args.get("scheduled_at"),
                                   room_id=args.get("room_id"),
                                   max_runs=args.get("max_runs"))
            return {"task_id": t_["id"]}

        @t("room_list_tasks", "List tasks.", {"p": {"room_id": I}})
        def _(args):
            with ldb as db:
                return [{"id": t_["id"], "name": t_["name"], "status": t_["status"],
                         "trigger_type": t_["trigger_type"]}
                        for t_ in q.list_tasks(db, room_id=args.get("room_id"))]

        @t("room_task_runs", "Recent runs of a task.",
           {"p": {"task_id": I}, "r": ["task_id"]})
        def _(args):
            with ldb as db:
                return q.list_task_runs(db, args["task_id"])

        # ---- inbox / escalations
        @t("room_escalate", "Escalate a question to the keeper.",
           {"p": {"room_id": I, "question": S, "from_agent_id": I},
            "r": ["room_id", "question"]})
        def _(args):
            with ldb as db:
                e = q.create_escalation(db, args["room_id"], args["question"],
                                        from_agent_id=args.get("from_agent_id"))
            return {"escalation_id": e["id"]}

        @t("room_answer_escalation", "Answer a pending escalation.",
           {"p": {"escalation_id": I, "answer": S},
            "r": ["escalation_id", "answer"]})
        def _(args):
            with ldb as db:
                q.answer_escalation(db, args["escalation_id"], args["answer"])
            return {"answered": True}

        @t("room_list_escalations", "List escalations for a room.",
           {"p": {"room_id": I, "status": S}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                return q.list_escalations(db, args["room_id"],
                                          status=args.get("status"))

        @t("room_send_room_message", "Send an inter-room message.",
           {"p": {"room_id": I, "to_room_id": S, "subject": S, "body": S},
            "r": ["room_id", "body"]})
        def _(args):
            with ldb as db:
                m = q.create_room_message(db, args["room_id"], "outbound",
                                          args.get("subject", ""), args["body"],
                                          to_room_id=args.get("to_room_id"))
            return {"message_id": m["id"]}

        # ---- wallet / identity
        @t("room_wallet_address", "The room wallet's address.",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                addr = wallet_mod.get_wallet_address(db, args["room_id"])
            return {"address": addr}

        @t("room_wallet_send", "Send tokens from the room wallet.",
           {"p": {"room_id": I, "to_address": S, "amount": S, "chain": S,
                  "token": S}, "r": ["room_id", "to_address", "amount"]})
        def _(args):
            with ldb as db:
                return wallet_mod.send_token(db, args["room_id"],
                                             args["to_address"], args["amount"],
                                             chain=args.get("chain", "base"),
                                             token=args.get("token", "usdc"))

        @t("room_payment_audit", "Wallet transaction history.",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                w = q.get_room_wallet(db, args["room_id"])
                if w is None:
                    return []
                return q.list_wallet_txs(db, w["id"])

        @t("room_register_identity", "Register the room's on-chain identity "
           "(ERC-8004 metadata prepared; broadcast requires RPC).",
           {"p": {"room_id": I, "chain": S}, "r": ["room_id"]})
        def _(args):
            from ..core.identity import register_identity
            with ldb as db:
                return register_identity(db, args["room_id"],
                                         chain=args.get("chain", "base"))

        # ---- settings / resources
        @t("room_get_setting", "Read a settings key.", {"p": {"key": S}, "r": ["key"]})
        def _(args):
            with ldb as db:
                return {"key": args["key"], "value": q.get_setting(db, args["key"])}

        @t("room_set_setting", "Write a settings key.",
           {"p": {"key": S, "value": S}, "r": ["key", "value"]})
        def _(args):
            with ldb as db:
                q.set_setting(db, args["key"], args["value"])
            return {"ok": True}

        @t("room_activity_feed", "Recent room activity.",
           {"p": {"room_id": I, "limit": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                return q.get_room_activity(db, args["room_id"],
                                           limit=args.get("limit", 20))

        # ---- watches
        @t("room_watch_path", "Watch a filesystem path (validated allowlist).",
           {"p": {"path": S, "action_prompt": S, "room_id": I}, "r": ["path"]})
        def _(args):
            from ..core.watch_path import validate_watch_path
            ok, why = validate_watch_path(args["path"])
            if not ok:
                return {"error": why}
            with ldb as db:
                w = q.create_watch(db, args["path"],
                                   action_prompt=args.get("action_prompt"),
                                   room_id=args.get("room_id"))
            return {"watch_id": w["id"]}

        @t("room_list_watches", "List watches.", {"p": {"room_id": I}})
        def _(args):
            with ldb as db:
                return q.list_watches(db, room_id=args.get("room_id"))

        # ---- additional surface (workers/goals/tasks/credentials/etc.)
        @t("room_get_worker", "Read one worker.", {"p": {"worker_id": I}, "r": ["worker_id"]})
        def _(args):
            with ldb as db:
                return q.get_worker(db, args["worker_id"]) or {"error": "not found"}

        @t("room_update_worker", "Update worker fields (prompt/model/pacing).",
           {"p": {"worker_id": I, "system_prompt": S, "model": S,
                  "cycle_gap_ms": I, "max_turns": I}, "r": ["worker_id"]})
        def _(args):
            fields = {k: v for k, v in args.items() if k != "worker_id"}
            with ldb as db:
                return q.update_worker(db, args["worker_id"], **fields) or {}

        @t("room_delete_worker", "Delete a worker.", {"p": {"worker_id": I}, "r": ["worker_id"]})
        def _(args):
            with ldb as db:
                q.delete_worker(db, args["worker_id"])
            return {"deleted": True}

        @t("room_get_wip", "Read a worker's saved work-in-progress.",
           {"p": {"worker_id": I}, "r": ["worker_id"]})
        def _(args):
            with ldb as db:
                w = q.get_worker(db, args["worker_id"])
            return {"wip": (w or {}).get("wip")}

        @t("room_clear_wip", "Clear a worker's WIP.", {"p": {"worker_id": I}, "r": ["worker_id"]})
        def _(args):
            with ldb as db:
                q.set_worker_wip(db, args["worker_id"], None)
            return {"cleared": True}

        @t("room_abandon_goal", "Abandon a goal with a reason.",
           {"p": {"goal_id": I, "reason": S}, "r": ["goal_id"]})
        def _(args):
            with ldb as db:
                g = goals_mod.abandon_goal(db, args["goal_id"], reason=args.get("reason"))
            return {"goal_id": g["id"], "status": g["status"]}

        @t("room_pause_task", "Pause a scheduled task.", {"p": {"task_id": I}, "r": ["task_id"]})
        def _(args):
            with ldb as db:
                q.update_task(db, args["task_id"], status="paused")
            return {"paused": True}

        @t("room_resume_task", "Resume a paused task.", {"p": {"task_id": I}, "r": ["task_id"]})
        def _(args):
            with ldb as db:
                q.update_task(db, args["task_id"], status="active")
            return {"resumed": True}

        @t("room_delete_task", "Delete a task.", {"p": {"task_id": I}, "r": ["task_id"]})
        def _(args):
            with ldb as db:
                q.delete_task(db, args["task_id"])
            return {"deleted": True}

        @t("room_run_logs", "Console logs of a task run.",
           {"p": {"run_id": I, "after_seq": I}, "r": ["run_id"]})
        def _(args):
            with ldb as db:
                return q.get_console_logs(db, args["run_id"],
                                          after_seq=args.get("after_seq", -1))

        @t("room_set_credential", "Store an encrypted room credential.",
           {"p": {"room_id": I, "name": S, "value": S, "type": S},
            "r": ["room_id", "name", "value"]})
        def _(args):
            from ..core.secret_store import encrypt_secret
            with ldb as db:
                row = q.set_credential(db, args["room_id"], args["name"],
                                       encrypt_secret(args["value"]),
                                       cred_type=args.get("type", "other"))
            return {"name": row["name"], "type": row["type"]}

        @t("room_list_credentials", "List credential names (values hidden).",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                return q.list_credentials(db, args["room_id"])

        @t("room_delete_credential", "Delete a credential.",
           {"p": {"room_id": I, "name": S}, "r": ["room_id", "name"]})
        def _(args):
            with ldb as db:
                q.delete_credential(db, args["room_id"], args["name"])
            return {"deleted": True}

        @t("room_cycles", "Recent agent cycles of a room.",
           {"p": {"room_id": I, "limit": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                return q.list_room_cycles(db, args["room_id"],
                                          limit=args.get("limit", 20))

        @t("room_cycle_logs", "Streamed logs of one agent cycle.",
           {"p": {"cycle_id": I, "after_seq": I}, "r": ["cycle_id"]})
        def _(args):
            with ldb as db:
                return q.get_cycle_logs(db, args["cycle_id"],
                                        after_seq=args.get("after_seq", -1))

        @t("room_token_usage", "Token usage rollup for a room.",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                return q.get_room_token_usage(db, args["room_id"])

        @t("room_list_templates", "Available room/worker templates.", {})
        def _(args):
            from ..core.templates import list_templates
            return list_templates()

        @t("room_create_from_template", "Instantiate a room template.",
           {"p": {"template": S, "name": S, "worker_model": S},
            "r": ["template", "name"]})
        def _(args):
            from ..core.templates import instantiate_room_template
            with ldb as db:
                r = instantiate_room_template(
                    db, args["template"], args["name"],
                    worker_model=args.get("worker_model", "qwen3-coder-30b"))
            return {"room_id": r["id"]}

        @t("room_export_prompts", "Export worker prompts as markdown files.",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            from ..core.prompt_sync import export_worker_prompts
            with ldb as db:
                return {"files": export_worker_prompts(db, args["room_id"])}

        @t("room_import_prompts", "Import worker prompts from markdown files.",
           {"p": {"room_id": I, "force": B}, "r": ["room_id"]})
        def _(args):
            from ..core.prompt_sync import import_worker_prompts
            with ldb as db:
                return import_worker_prompts(db, args["room_id"],
                                             force=args.get("force", False))

        @t("room_public_feed", "Curated public activity feed.", {"p": {"limit": I}})
        def _(args):
            from ..core.public_feed import get_public_feed
            with ldb as db:
                return get_public_feed(db, limit=args.get("limit", 20))

        @t("room_public_profile", "Public profile of a public room.",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            from ..core.public_feed import get_public_room_profile
            with ldb as db:
                return get_public_room_profile(db, args["room_id"]) or                     {"error": "room not public"}

        @t("room_notify_keeper", "Deliver a keeper notification (outbox/email).",
           {"p": {"subject": S, "body": S, "room_id": I}, "r": ["subject", "body"]})
        def _(args):
            from ..core.notifications import notify_keeper
            return notify_keeper(args["subject"], args["body"],
                                 room_id=args.get("room_id"))

        @t("room_read_outbox", "Read recent keeper notifications.", {"p": {"limit": I}})
        def _(args):
            from ..core.notifications import read_outbox
            return read_outbox(limit=args.get("limit", 20))

        @t("room_web_fetch", "Fetch a URL as readable text.",
           {"p": {"url": S}, "r": ["url"]})
        def _(args):
            from ..core.web_tools import web_fetch
            return web_fetch(args["url"])

        @t("room_web_search", "Keyless web search.", {"p": {"query": S}, "r": ["query"]})
        def _(args):
            from ..core.web_tools import web_search
            return web_search(args["query"])

        @t("room_browser_action", "Persistent browser-session action.",
           {"p": {"session_id": S, "action": S, "url": S},
            "r": ["session_id", "action"]})
        def _(args):
            from ..core.web_tools import browser_action
            return browser_action(args["session_id"], args["action"],
                                  url=args.get("url"))

        @t("room_clerk_chat", "Chat with the system-wide Clerk assistant.",
           {"p": {"content": S}, "r": ["content"]})
        def _(args):
            from ..core.clerk import clerk_chat
            return {"reply": clerk_chat(ldb, args["content"],
                                        memory=self.memory)}

        @t("room_delete_watch", "Remove a filesystem watch.",
           {"p": {"watch_id": I}, "r": ["watch_id"]})
        def _(args):
            with ldb as db:
                q.delete_watch(db, args["watch_id"])
            return {"deleted": True}

        @t("room_list_messages", "Inter-room messages of a room.",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                return q.list_room_messages(db, args["room_id"])

        @t("room_self_mod_audit", "Self-modification audit log.",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                return q.list_self_mod_audit(db, args["room_id"])

        # ---- parity surface (reference mcp/tools: quorum/skills/room/memory/
        # credentials/identity/invite/scheduler/wallet/resources details)

        @t("room_decision_detail", "A decision with its votes.",
           {"p": {"decision_id": I}, "r": ["decision_id"]})
        def _(args):
            with ldb as db:
                d = q.get_decision(db, args["decision_id"])
                if d is None:
                    return {"error": f"decision {args['decision_id']} not found"}
                votes = q.get_votes(db, args["decision_id"])
            return {**d, "votes": [{"worker_id": v["worker_id"], "vote": v["vote"],
                                    "reasoning": v["reasoning"]} for v in votes]}

        @t("room_activate_skill", "Enable auto-activation for a skill.",
           {"p": {"skill_id": I}, "r": ["skill_id"]})
        def _(args):
            with ldb as db:
                s = q.update_skill(db, args["skill_id"], auto_activate=True)
            return {"activated": s is not None}

        @t("room_deactivate_skill", "Disable auto-activation for a skill.",
           {"p": {"skill_id": I}, "r": ["skill_id"]})
        def _(args):
            with ldb as db:
                s = q.update_skill(db, args["skill_id"], auto_activate=False)
            return {"deactivated": s is not None}

        @t("room_delete_skill", "Delete a skill.",
           {"p": {"skill_id": I}, "r": ["skill_id"]})
        def _(args):
            with ldb as db:
                q.delete_skill(db, args["skill_id"])
            return {"deleted": True}

        @t("room_configure_room", "Update a room's quorum/pacing config "
           "(threshold, timeoutMinutes, tieBreaker, autoApprove, minCycleGapMs...).",
           {"p": {"room_id": I, "config": {"type": "object"}},
            "r": ["room_id", "config"]})
        def _(args):
            with ldb as db:
                room = q.get_room(db, args["room_id"])
                if room is None:
                    return {"error": "room not found"}
                raw = room.get("config") or {}
                cfg = json.loads(raw) if isinstance(raw, str) else dict(raw)
                cfg.update(args["config"])
                q.update_room(db, args["room_id"], config=json.dumps(cfg))
            return {"config": cfg}

        @t("room_delete_room", "Permanently delete a room and its data.",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                room_mod.delete_room(db, args["room_id"])
            return {"deleted": True}

        @t("room_get_credential", "Retrieve a credential's decrypted value "
           "(room_list_credentials shows masked names only).",
           {"p": {"room_id": I, "name": S}, "r": ["room_id", "name"]})
        def _(args):
            from ..core.secret_store import decrypt_secret
            with ldb as db:
                row = q.get_credential(db, args["room_id"], args["name"])
            if row is None:
                return {"error": f"credential '{args['name']}' not found"}
            return {"name": row["name"], "type": row["type"],
                    "value": decrypt_secret(row["value_encrypted"])}

        @t("room_forget", "Delete a memory entity (and its observations/"
           "relations/embedding).", {"p": {"entity_id": I}, "r": ["entity_id"]})
        def _(args):
            with ldb as db:
                ent = q.get_entity(db, args["entity_id"])
                if ent is None:
                    return {"error": f"memory {args['entity_id']} not found"}
                q.delete_entity(db, args["entity_id"])
            if self.memory is not None:
                try:
                    self.memory.store.remove(args["entity_id"])
                except Exception:
                    pass  # GPU index rebuilds from SQLite
            return {"forgot": ent["name"]}

        @t("room_memory_list", "List a room's memory entities (newest first).",
           {"p": {"room_id": I, "limit": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                rows = db.execute(
                    "SELECT id, name, type, category, created_at FROM entities"
                    " WHERE room_id = ? ORDER BY id DESC LIMIT ?",
                    (args["room_id"], args.get("limit", 50))).fetchall()
            return rows

        @t("room_identity_get", "The room's on-chain identity (ERC-8004).",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                w = q.get_room_wallet(db, args["room_id"])
            if w is None:
                return {"error": "room has no wallet"}
            return {"address": w["address"],
                    "agent_id": w.get("identity_agent_id"),
                    "registered": bool(w.get("identity_agent_id"))}

        @t("room_identity_update", "Rebuild and re-register the identity "
           "metadata URI from current room state.",
           {"p": {"room_id": I, "chain": S}, "r": ["room_id"]})
        def _(args):
            from ..core import identity as identity_mod
            with ldb as db:
                return identity_mod.register_identity(
                    db, args["room_id"], chain=args.get("chain", "base"))

        @t("room_invite_create", "Create a cloud invite link for this room's "
           "network.", {"p": {"room_id": I, "max_uses": I, "expires_in_days": I},
                        "r": ["room_id"]})
        def _(args):
            from ..core import cloud_sync
            if cloud_sync.cloud_api() is None:
                return {"error": "Cloud is not configured (ROOMAMD_CLOUD_API); "
                                 "invites need the cloud relay."}
            token = cloud_sync.register_with_cloud(ldb, args["room_id"])
            out = cloud_sync._post(
                f"/rooms/{args['room_id']}/invites",
                {"maxUses": args.get("max_uses"),
                 "expiresInDays": args.get("expires_in_days")}, token)
            return out or {"error": "Failed to create invite (cloud unavailable)."}

        @t("room_invite_list", "List this room's invite links.",
           {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            from ..core import cloud_sync
            if cloud_sync.cloud_api() is None:
                return {"error": "Cloud is not configured (ROOMAMD_CLOUD_API)."}
            token = cloud_sync.load_room_tokens().get(str(args["room_id"]))
            out = cloud_sync._post(f"/rooms/{args['room_id']}/invites/list",
                                   {}, token)
            return out or {"error": "Cloud unavailable."}

        @t("room_invite_network", "Rooms that joined through this room's "
           "invites.", {"p": {"room_id": I}, "r": ["room_id"]})
        def _(args):
            from ..core import cloud_sync
            if cloud_sync.cloud_api() is None:
                return {"error": "Cloud is not configured (ROOMAMD_CLOUD_API)."}
            token = cloud_sync.load_room_tokens().get(str(args["room_id"]))
            out = cloud_sync._post(f"/rooms/{args['room_id']}/network", {}, token)
            return out or {"error": "Cloud unavailable."}

        @t("room_pause_watch", "Pause a filesystem watch.",
           {"p": {"watch_id": I}, "r": ["watch_id"]})
        def _(args):
            with ldb as db:
                db.execute("UPDATE watches SET status = 'paused' WHERE id = ?",
                           (args["watch_id"],))
            return {"paused": True}

        @t("room_resume_watch", "Resume a paused filesystem watch.",
           {"p": {"watch_id": I}, "r": ["watch_id"]})
        def _(args):
            with ldb as db:
                db.execute("UPDATE watches SET status = 'active' WHERE id = ?",
                           (args["watch_id"],))
            return {"resumed": True}

        @t("room_reset_session", "Clear a task's session so the next run "
           "starts a fresh conversation.", {"p": {"task_id": I}, "r": ["task_id"]})
        def _(args):
            with ldb as db:
                task = q.get_task(db, args["task_id"])
                if task is None:
                    return {"error": f"task {args['task_id']} not found"}
                q.update_task(db, args["task_id"], session_id=None)
            return {"reset": task["name"]}

        @t("room_run_task", "Execute a task immediately (returns right away; "
           "use room_task_runs for status).", {"p": {"task_id": I}, "r": ["task_id"]})
        def _(args):
            with ldb as db:
                task = q.get_task(db, args["task_id"])
                if task is None:
                    return {"error": f"task {args['task_id']} not found"}
            # cross-process: ask the HTTP server to run it now; fall back to
            # marking it due so the 15 s runtime loop picks it up
            if _server_post(f"/api/tasks/{args['task_id']}/run"):
                return {"started": task["name"]}
            with ldb as db:
                q.update_task(db, args["task_id"], status="active",
                              scheduled_at=q.now_iso())
            return {"queued": task["name"],
                    "note": "server offline; task due on next scheduler pass"}

        @t("room_wallet_balance", "On-chain token balance of the room wallet "
           "(USDC/USDT across Base, Ethereum, Arbitrum, Optimism, Polygon).",
           {"p": {"room_id": I, "network": S, "token": S}, "r": ["room_id"]})
        def _(args):
            with ldb as db:
                return wallet_mod.get_on_chain_balance(
                    db, args["room_id"], chain=args.get("network", "base"),
                    token=args.get("token", "usdc"))

        @t("room_wallet_topup", "A top-up URL the keeper can use to fund the "
           "room wallet by card (USDC on Base).",
           {"p": {"room_id": I, "amount": N}, "r": ["room_id"]})
        def _(args):
            from ..core import cloud_sync
            with ldb as db:
                w = q.get_room_wallet(db, args["room_id"])
            if w is None:
                return {"error": "room has no wallet"}
            if cloud_sync.cloud_api() is not None:
                token = cloud_sync.load_room_tokens().get(str(args["room_id"]))
                out = cloud_sync._post(f"/rooms/{args['room_id']}/onramp",
                                       {"address": w["address"],
                                        "amount": args.get("amount")}, token)
                if out and out.get("onrampUrl"):
                    return {"onramp_url": out["onrampUrl"]}
            return {"address": w["address"],
                    "note": "Cloud on-ramp unavailable; send USDC on Base "
                            "directly to this address."}

        @t("room_webhook_url", "Webhook URL for a task (external trigger) or "
           "room (message + queen wake).",
           {"p": {"task_id": I, "room_id": I, "generate_if_missing": B}})
        def _(args):
            import secrets as _secrets
            port = _server_port()
            base = f"http://127.0.0.1:{port}" if port else "http://<server>"
            with ldb as db:
                if args.get("task_id"):
                    task = q.get_task(db, args["task_id"])
                    if task is None:
                        return {"error": f"task {args['task_id']} not found"}
                    tok = task.get("webhook_token")
                    if not tok and args.get("generate_if_missing"):
                        tok = _secrets.token_hex(16)
                        q.update_task(db, args["task_id"], webhook_token=tok)
                    if not tok:
                        return {"error": "task has no webhook token "
                                         "(pass generate_if_missing)"}
                    return {"url": f"{base}/api/hooks/task/{tok}"}
                if args.get("room_id"):
                    room = q.get_room(db, args["room_id"])
                    if room is None:
                        return {"error": "room not found"}
                    tok = room.get("webhook_token")
                    if not tok and args.get("generate_if_missing"):
                        tok = _secrets.token_hex(16)
                        q.update_room(db, args["room_id"], webhook_token=tok)
                    if not tok:
                        return {"error": "room has no webhook token "
                                         "(pass generate_if_missing)"}
                    return {"url": f"{base}/api/hooks/queen/{tok}"}
            return {"error": "pass task_id or room_id"}

        @t("room_resources", "Local machine + GPU resource usage (decide "
           "whether the swarm needs more capacity).", {})
        def _(args):
            import os as _os
            load1, load5, _ = _os.getloadavg()
            cpus = _os.cpu_count() or 1
            mem = {}
            try:
                with open("/proc/meminfo") as f:
                    for line in f:
                        k, v = line.split(":", 1)
                        if k in ("MemTotal", "MemAvailable"):
                            mem[k] = int(v.strip().split()[0]) * 1024
            except OSError:
                pass
            used_pct = (round((1 - mem["MemAvailable"] / mem["MemTotal"]) * 100)
                        if mem else None)
            out = {"cpu_load_1m": load1, "cpu_load_5m": load5, "cpus": cpus,
                   "cpu_pct_of_capacity": round(load1 / cpus * 100),
                   "ram_used_pct": used_pct}
            try:
                import torch
                if torch.cuda.is_available():
                    free_b, total_b = torch.cuda.mem_get_info()
                    out["gpu_hbm_used_pct"] = round((1 - free_b / total_b) * 100)
                    out["gpu_count"] = torch.cuda.device_count()
            except Exception:
                pass
            with ldb as db:
                tasks = q.list_tasks(db)
                out["active_tasks"] = sum(1 for tk in tasks
                                          if tk["status"] == "active")
            high = (load1 / cpus > 0.8) or (used_pct or 0) > 85
            out["summary"] = ("HIGH LOAD — consider scaling swarm capacity"
                              if high else "Normal load")
            return out

        # ---- reference-name aliases: every quoroom_* tool of the reference
        # MCP registry (src/mcp/server.ts, 76 tools) resolves here too, so
        # clients configured against the reference's tool names work
        # unchanged. Trivial renames map quoroom_X → room_X; the rest map to
        # the semantically-equivalent tool below.
        REF_ALIAS = {
            "browser": "room_browser_action",
            "create_subgoal": "room_set_goal",
            "credentials_get": "room_get_credential",
            "credentials_list": "room_list_credentials",
            "edit_skill": "room_modify_skill",
            "export_worker_prompts": "room_export_prompts",
            "import_worker_prompts": "room_import_prompts",
            "identity_register": "room_register_identity",
            "inbox_list": "room_list_escalations",
            "inbox_reply": "room_answer_escalation",
            "inbox_send_room": "room_send_room_message",
            "list_goals": "room_goal_tree",
            "propose": "room_announce",
            "resources_get": "room_resources",
            "room_activity": "room_activity_feed",
            "room_status": "room_get_status",
            "schedule": "room_create_task",
            "self_mod_edit": "room_modify_skill",
            "self_mod_history": "room_self_mod_audit",
            "self_mod_revert": "room_revert_modification",
            "send_message": "room_send_room_message",
            "task_history": "room_task_runs",
            "task_progress": "room_task_runs",
            "unwatch": "room_delete_watch",
            "update_progress": "room_update_goal_progress",
            "wallet_create": "room_wallet_address",
            "wallet_history": "room_payment_audit",
            "watch": "room_watch_path",
        }
        REF_SUFFIXES = [
            "abandon_goal", "activate_skill", "browser", "complete_goal",
            "configure_room", "create_room", "create_skill", "create_subgoal",
            "create_worker", "credentials_get", "credentials_list",
            "deactivate_skill", "decision_detail", "delegate_task",
            "delete_room", "delete_skill", "delete_task", "delete_worker",
            "edit_skill", "export_worker_prompts", "forget", "get_setting",
            "identity_get", "identity_register", "identity_update",
            "import_worker_prompts", "inbox_list", "inbox_reply",
            "inbox_send_room", "invite_create", "invite_list",
            "invite_network", "list_decisions", "list_goals", "list_rooms",
            "list_skills", "list_tasks", "list_watches", "list_workers",
            "memory_list", "pause_room", "pause_task", "pause_watch",
            "propose", "recall", "remember", "reset_session",
            "resources_get", "restart_room", "resume_task", "resume_watch",
            "room_activity", "room_status", "run_task", "save_wip",
            "schedule", "self_mod_edit", "self_mod_history",
            "self_mod_revert", "send_message", "set_goal", "set_setting",
            "task_history", "task_progress", "unwatch", "update_progress",
            "update_worker", "vote", "wallet_address", "wallet_balance",
            "wallet_create", "wallet_history", "wallet_send", "wallet_topup",
            "watch", "webhook_url",
        ]
        for suffix in REF_SUFFIXES:
            target = REF_ALIAS.get(suffix, f"room_{suffix}")
            if target not in self.tools:
                continue  # defensive; covered by tests
            schema, fn = self.tools[target]
            alias = f"quoroom_{suffix}"
            self.tools[alias] = ({**schema, "name": alias,
                                  "description": schema["description"]
                                  + f" (alias of {target})"}, fn)

    # ------------------------------------------------------------ JSON-RPC

    def handle(self, msg: dict) -> dict | None:
        mid = msg.get("id")
        method = msg.get("method", "")
        if method == "initialize":
            return self._result(mid, {
                "protocolVersion": PROTOCOL_VERSION,
                "capabilities": {"tools": {}},
                "serverInfo": SERVER_INFO})
        if method == "notifications/initialized":
            return None
        if method == "tools/list":
            return self._result(mid, {"tools": [t[0] for t in self.tools.values()]})
        if method == "tools/call":
            params = msg.get("params", {})
            name = params.get("name", "")
            args = params.get("arguments", {}) or {}
            if name not in self.tools:
                return self._error(mid, -32602, f"unknown tool: {name}")
            try:
                out = self.tools[name][1](args)
                return self._result(mid, {"content": [
                    {"type": "text", "text": json.dumps(out, default=str)}]})
            except Exception as e:
                return self._result(mid, {"content": [
                    {"type": "text", "text": json.dumps({"error": str(e)})}],
                    "isError": True})
        if method == "ping":
            return self._result(mid, {})
        if mid is None:
            return None
        return self._error(mid, -32601, f"unknown method: {method}")

    @staticmethod
    def _result(mid, result) -> dict:
        return {"jsonrpc": "2.0", "id": mid, "result": result}

    @staticmethod
    def _error(mid, code, message) -> dict:
        return {"jsonrpc": "2.0", "id": mid,
                "error": {"code": code, "message": message}}

    def run_stdio(self) -> None:
        """Blocking stdio loop: newline-delimited JSON-RPC."""
        for line in sys.stdin:
            line = line.strip()
            if not line:
                continue
            try:
                msg = json.loads(line)
            except ValueError:
                continue
            resp = self.handle(msg)
            if resp is not None:
                sys.stdout.write(json.dumps(resp) + "\n")
                sys.stdout.flush()


def main() -> None:
    import os

    from ..memory.vector_store import GpuVectorStore, MemoryService
    from ..server.auth import data_dir

    db_path = os.environ.get("ROOMAMD_DB_PATH", str(data_dir() / "data.db"))
    ldb = LockedDb(connect(db_path))
    memory = MemoryService(ldb, store=GpuVectorStore(capacity=100_000,
                                                     device="cpu"))
    McpServer(ldb, memory=memory).run_stdio()


if __name__ == "__main__":
    main()
