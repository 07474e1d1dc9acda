"""Room lifecycle (reference semantics: src/shared/room.ts).

createRoom = room row + queen worker + root goal + deterministic wallet
(room.ts:33-70); pause/restart clears goals/decisions/escalations
(room.ts:87-112).
"""
from __future__ import annotations

import hashlib
import secrets
import sqlite3

from ..db import queries as q
from .constants import DEFAULT_QUEEN_CYCLE_GAP_MS, LOCAL_MODEL_TAG

DEFAULT_QUEEN_PROMPT = """You are the Queen of this room — the control plane.
You coordinate, you do not execute. Decompose the room objective into goals,
delegate tasks to workers, announce decisions for quorum review, and escalate
to the keeper when blocked. Never do worker-level execution yourself; create
or wake an executor worker instead."""


def create_room(db: sqlite3.Connection, name: str, goal: str | None = None,
                worker_model: str = LOCAL_MODEL_TAG,
                queen_cycle_gap_ms: int | None = None,
                config: dict | None = None) -> dict:
    """Create a room with its queen worker, root goal, webhook token and wallet."""
    room = q.create_room_row(db, name, goal=goal, worker_model=worker_model,
                             config=config)
    queen = q.create_worker(
        db, name=f"Queen of {name}", role="queen",
        system_prompt=DEFAULT_QUEEN_PROMPT, model=worker_model,
        room_id=room["id"], is_default=True,
    )
    q.update_room(
        db, room["id"],
        queen_worker_id=queen["id"],
        webhook_token=secrets.token_hex(16),
        queen_cycle_gap_ms=queen_cycle_gap_ms or DEFAULT_QUEEN_CYCLE_GAP_MS
        if queen_cycle_gap_ms else DEFAULT_QUEEN_CYCLE_GAP_MS,
    )
    if queen_cycle_gap_ms:
        db.execute("UPDATE rooms SET queen_cycle_gap_ms = ? WHERE id = ?",
                   (queen_cycle_gap_ms, room["id"]))
    if goal:
        q.create_goal(db, room["id"], goal)
    # Auto wallet with a deterministic key (reference: room.ts:52-62 uses a
    # SHA-256-derived key); actual keygen/encryption lives in core.wallet.
    from . import wallet
    wallet.create_room_wallet(db, room["id"], deterministic_seed=f"room-{room['id']}-{name}")
    q.log_room_activity(db, room["id"], "room", f"Room '{name}' created")
    return q.get_room(db, room["id"])


def pause_room(db: sqlite3.Connection, room_id: int) -> dict:
    q.update_room(db, room_id, status="paused")
    q.log_room_activity(db, room_id, "room", "Room paused")
    return q.get_room(db, room_id)


def resume_room(db: sqlite3.Connection, room_id: int) -> dict:
    q.update_room(db, room_id, status="active")
    q.log_room_activity(db, room_id, "room", "Room resumed")
    return q.get_room(db, room_id)


def stop_room(db: sqlite3.Connection, room_id: int) -> dict:
    q.update_room(db, room_id, status="stopped")
    q.log_room_activity(db, room_id, "room", "Room stopped")
    return q.get_room(db, room_id)


def restart_room(db: sqlite3.Connection, room_id: int) -> dict:
    """Fresh start: delete goals, decisions, escalations; keep workers, memory,
    wallet (reference: room.ts:87-112)."""
    db.execute("DELETE FROM goals WHERE room_id = ?", (room_id,))
    db.execute("DELETE FROM quorum_decisions WHERE room_id = ?", (room_id,))
    db.execute("DELETE FROM escalations WHERE room_id = ?", (room_id,))
    room = q.get_room(db, room_id)
    if room and room.get("goal"):
        q.create_goal(db, room_id, room["goal"])
    q.update_room(db, room_id, status="active")
    q.log_room_activity(db, room_id, "room", "Room restarted")
    return q.get_room(db, room_id)


def delete_room(db: sqlite3.Connection, room_id: int) -> None:
    q.delete_room(db, room_id)


def get_room_status(db: sqlite3.Connection, room_id: int) -> dict:
    """Aggregate status snapshot (reference: room.ts:135-146)."""
    room = q.get_room(db, room_id)
    if room is None:
        raise ValueError(f"Room {room_id} not found")
    workers = q.list_room_workers(db, room_id)
    goals = q.list_room_goals(db, room_id)
    decisions = q.list_room_decisions(db, room_id, limit=10)
    pending = [d for d in decisions if d["status"] in ("voting", "announced")]
    escalations = q.list_escalations(db, room_id, status="pending")
    usage = q.get_room_token_usage(db, room_id)
    return {
        "room": room,
        "workers": workers,
        "goals": goals,
        "active_goals": [g for g in goals if g["status"] in ("active", "in_progress")],
        "pending_decisions": pending,
        "pending_escalations": escalations,
        "token_usage": usage,
    }


def deterministic_key_material(seed: str) -> bytes:
    return hashlib.sha256(seed.encode()).digest()
