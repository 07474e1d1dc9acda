"""Curated public activity feed (reference: src/shared/public-feed.ts —
is_public rows only, details stripped) + public room profile aggregate."""
from __future__ import annotations

import sqlite3

from ..db import queries as q


def get_public_feed(db: sqlite3.Connection, limit: int = 50) -> list[dict]:
    rows = db.execute(
        "SELECT a.id, a.room_id, r.name AS room_name, a.event_type, a.summary,"
        " a.created_at FROM room_activity a JOIN rooms r ON r.id = a.room_id"
        " WHERE a.is_public = 1 AND r.visibility = 'public'"
        " ORDER BY a.id DESC LIMIT ?", (limit,)).fetchall()
    return rows  # details column deliberately omitted


def get_public_room_profile(db: sqlite3.Connection, room_id: int) -> dict | None:
    room = q.get_room(db, room_id)
    if room is None or room.get("visibility") != "public":
        return None
    goals = q.list_room_goals(db, room_id)
    workers = q.list_room_workers(db, room_id)
    usage = q.get_room_token_usage(db, room_id)
    return {
        "id": room["id"],
        "name": room["name"],
        "goal": room["goal"],
        "status": room["status"],
        "worker_count": len(workers),
        "goals_total": len(goals),
        "goals_completed": sum(1 for g in goals if g["status"] == "completed"),
        "cycles": usage["cycles"],
    }
