"""Typed query layer over the preserved Quoroom schema.

Function-per-query style mirroring the seam the reference exposes
(src/shared/db-queries.ts, ~190 functions) — every engine module goes through
here, never raw SQL. Newly written for this runtime; semantics cross-checked
against the reference's behavior (citations inline).
"""
from __future__ import annotations

import json
import sqlite3
import struct
import time
from datetime import datetime, timedelta
from typing import Any, Iterable, Optional

from ..core.constants import (DEFAULT_ROOM_CONFIG, HYBRID_FTS_WEIGHT,
                              HYBRID_RRF_K, HYBRID_SEMANTIC_WEIGHT)

# ---------------------------------------------------------------- helpers


def now_iso() -> str:
    return datetime.now().strftime("%Y-%m-%d %H:%M:%S")


def _iso_in_ms(ms: int) -> str:
    return (datetime.now() + timedelta(milliseconds=ms)).strftime("%Y-%m-%d %H:%M:%S")


# ---------------------------------------------------------------- settings


def get_setting(db: sqlite3.Connection, key: str) -> Optional[str]:
    row = db.execute("SELECT value FROM settings WHERE key = ?", (key,)).fetchone()
    return row["value"] if row else None


def set_setting(db: sqlite3.Connection, key: str, value: str) -> None:
    db.execute(
        "INSERT INTO settings (key, value, updated_at) VALUES (?, ?, ?) "
        "ON CONFLICT(key) DO UPDATE SET value = excluded.value, updated_at = excluded.updated_at",
        (key, value, now_iso()),
    )


def list_settings(db: sqlite3.Connection) -> list[dict]:
    return db.execute("SELECT * FROM settings ORDER BY key").fetchall()


# ---------------------------------------------------------------- workers


def create_worker(
    db: sqlite3.Connection,
    name: str,
    system_prompt: str,
    role: str | None = None,
    description: str | None = None,
    model: str | None = None,
    room_id: int | None = None,
    cycle_gap_ms: int | None = None,
    max_turns: int | None = None,
    is_default: bool = False,
) -> dict:
    cur = db.execute(
        "INSERT INTO workers (name, role, system_prompt, description, model, room_id,"
        " cycle_gap_ms, max_turns, is_default) VALUES (?,?,?,?,?,?,?,?,?)",
        (name, role, system_prompt, description, model, room_id, cycle_gap_ms,
         max_turns, int(is_default)),
    )
    return get_worker(db, cur.lastrowid)


def get_worker(db: sqlite3.Connection, worker_id: int) -> Optional[dict]:
    return db.execute("SELECT * FROM workers WHERE id = ?", (worker_id,)).fetchone()


def list_workers(db: sqlite3.Connection) -> list[dict]:
    return db.execute("SELECT * FROM workers ORDER BY id").fetchall()


def list_room_workers(db: sqlite3.Connection, room_id: int) -> list[dict]:
    return db.execute(
        "SELECT * FROM workers WHERE room_id = ? ORDER BY id", (room_id,)
    ).fetchall()


def update_worker(db: sqlite3.Connection, worker_id: int, **fields: Any) -> Optional[dict]:
    allowed = {"name", "role", "system_prompt", "description", "model", "room_id",
               "cycle_gap_ms", "max_turns", "agent_state", "wip", "is_default",
               "task_count", "votes_cast", "votes_missed"}
    sets, vals = [], []
    for k, v in fields.items():
        if k not in allowed:
            raise ValueError(f"unknown worker field: {k}")
        sets.append(f"{k} = ?")
        vals.append(v)
    if not sets:
        return get_worker(db, worker_id)
    vals += [now_iso(), worker_id]
    db.execute(f"UPDATE workers SET {', '.join(sets)}, updated_at = ? WHERE id = ?", vals)
    return get_worker(db, worker_id)


def delete_worker(db: sqlite3.Connection, worker_id: int) -> None:
    db.execute("DELETE FROM workers WHERE id = ?", (worker_id,))


def set_worker_state(db: sqlite3.Connection, worker_id: int, state: str) -> None:
    db.execute(
        "UPDATE workers SET agent_state = ?, updated_at = ? WHERE id = ?",
        (state, now_iso(), worker_id),
    )


def set_worker_wip(db: sqlite3.Connection, worker_id: int, wip: str | None) -> None:
    db.execute(
        "UPDATE workers SET wip = ?, updated_at = ? WHERE id = ?",
        (wip, now_iso(), worker_id),
    )


def ensure_worker_room_mapping(db: sqlite3.Connection, worker_id: int, room_id: int) -> None:
    """Re-attach a worker to its room if the mapping was lost (agent-loop.ts behavior)."""
    db.execute(
        "UPDATE workers SET room_id = ? WHERE id = ? AND (room_id IS NULL OR room_id != ?)",
        (room_id, worker_id, room_id),
    )


# ---------------------------------------------------------------- rooms


def create_room_row(db: sqlite3.Connection, name: str, goal: str | None = None,
                    worker_model: str = "local", config: dict | None = None) -> dict:
    cur = db.execute(
        "INSERT INTO rooms (name, goal, worker_model, config) VALUES (?,?,?,?)",
        (name, goal, worker_model, json.dumps(config) if config else None),
    )
    return get_room(db, cur.lastrowid)


def get_room(db: sqlite3.Connection, room_id: int) -> Optional[dict]:
    row = db.execute("SELECT * FROM rooms WHERE id = ?", (room_id,)).fetchone()
    if row is None:
        return None
    cfg = dict(DEFAULT_ROOM_CONFIG)
    if row.get("config"):
        try:
            cfg.update(json.loads(row["config"]))
        except (ValueError, TypeError):
            pass
    row["config"] = cfg
    return row


def get_room_by_webhook_token(db: sqlite3.Connection, token: str) -> Optional[dict]:
    row = db.execute("SELECT id FROM rooms WHERE webhook_token = ?", (token,)).fetchone()
    return get_room(db, row["id"]) if row else None


def list_rooms(db: sqlite3.Connection) -> list[dict]:
    rows = db.execute("SELECT id FROM rooms ORDER BY id").fetchall()
    return [get_room(db, r["id"]) for r in rows]


def update_room(db: sqlite3.Connection, room_id: int, **fields: Any) -> Optional[dict]:
    allowed = {"name", "goal", "status", "visibility", "autonomy_mode",
               "max_concurrent_tasks", "worker_model", "queen_cycle_gap_ms",
               "queen_max_turns", "queen_quiet_from", "queen_quiet_until", "config",
               "webhook_token", "queen_nickname", "chat_session_id", "allowed_tools",
               "queen_worker_id"}
    sets, vals = [], []
    for k, v in fields.items():
        if k not in allowed:
            raise ValueError(f"unknown room field: {k}")
        if k == "config" and isinstance(v, dict):
            v = json.dumps(v)
        sets.append(f"{k} = ?")
        vals.append(v)
    if sets:
        vals += [now_iso(), room_id]
        db.execute(f"UPDATE rooms SET {', '.join(sets)}, updated_at = ? WHERE id = ?", vals)
    return get_room(db, room_id)


def delete_room(db: sqlite3.Connection, room_id: int) -> None:
    db.execute("DELETE FROM workers WHERE room_id = ?", (room_id,))
    db.execute("DELETE FROM rooms WHERE id = ?", (room_id,))


# ---------------------------------------------------------------- activity


def log_room_activity(db: sqlite3.Connection, room_id: int, event_type: str,
                      summary: str, details: str | None = None,
                      actor_id: int | None = None, is_public: bool = True) -> int:
    cur = db.execute(
        "INSERT INTO room_activity (room_id, event_type, actor_id, summary, details, is_public)"
        " VALUES (?,?,?,?,?,?)",
        (room_id, event_type, actor_id, summary, details, int(is_public)),
    )
    return cur.lastrowid


def get_room_activity(db: sqlite3.Connection, room_id: int, limit: int = 50) -> list[dict]:
    return db.execute(
        "SELECT * FROM room_activity WHERE room_id = ? ORDER BY id DESC LIMIT ?",
        (room_id, limit),
    ).fetchall()


# ---------------------------------------------------------------- quorum


def create_decision(db: sqlite3.Connection, room_id: int, proposer_id: int | None,
                    proposal: str, decision_type: str, threshold: str = "majority",
                    timeout_minutes: int = 60, min_voters: int = 0,
                    sealed: bool = False) -> dict:
    # reference db-queries.ts:1266-1274 (min_voters / sealed on the row)
    timeout_at = _iso_in_ms(timeout_minutes * 60_000)
    cur = db.execute(
        "INSERT INTO quorum_decisions (room_id, proposer_id, proposal, decision_type,"
        " status, threshold, timeout_at, min_voters, sealed)"
        " VALUES (?,?,?,?,'voting',?,?,?,?)",
        (room_id, proposer_id, proposal, decision_type, threshold, timeout_at,
         min_voters, int(sealed)),
    )
    return get_decision(db, cur.lastrowid)


def create_announcement(db: sqlite3.Connection, room_id: int, proposer_id: int | None,
                        proposal: str, decision_type: str, effective_at: str) -> dict:
    cur = db.execute(
        "INSERT INTO quorum_decisions (room_id, proposer_id, proposal, decision_type,"
        " status, effective_at) VALUES (?,?,?,?,'announced',?)",
        (room_id, proposer_id, proposal, decision_type, effective_at),
    )
    return get_decision(db, cur.lastrowid)


def get_decision(db: sqlite3.Connection, decision_id: int) -> Optional[dict]:
    return db.execute(
        "SELECT * FROM quorum_decisions WHERE id = ?", (decision_id,)
    ).fetchone()


def list_room_decisions(db: sqlite3.Connection, room_id: int, status: str | None = None,
                        limit: int = 100) -> list[dict]:
    if status:
        return db.execute(
            "SELECT * FROM quorum_decisions WHERE room_id = ? AND status = ?"
            " ORDER BY id DESC LIMIT ?", (room_id, status, limit)).fetchall()
    return db.execute(
        "SELECT * FROM quorum_decisions WHERE room_id = ? ORDER BY id DESC LIMIT ?",
        (room_id, limit)).fetchall()


def resolve_decision(db: sqlite3.Connection, decision_id: int, status: str,
                     result: str | None = None) -> None:
    db.execute(
        "UPDATE quorum_decisions SET status = ?, result = ?, resolved_at = ? WHERE id = ?",
        (status, result, now_iso(), decision_id),
    )


def get_announced_decisions_past_effective(db: sqlite3.Connection) -> list[dict]:
    return db.execute(
        "SELECT * FROM quorum_decisions WHERE status = 'announced'"
        " AND effective_at <= ?", (now_iso(),)).fetchall()


def get_expired_voting_decisions(db: sqlite3.Connection) -> list[dict]:
    return db.execute(
        "SELECT * FROM quorum_decisions WHERE status = 'voting' AND timeout_at <= ?",
        (now_iso(),)).fetchall()


def cast_vote(db: sqlite3.Connection, decision_id: int, worker_id: int, vote: str,
              reasoning: str | None = None) -> dict:
    db.execute(
        "INSERT INTO quorum_votes (decision_id, worker_id, vote, reasoning) VALUES (?,?,?,?)"
        " ON CONFLICT(decision_id, worker_id) DO UPDATE SET vote=excluded.vote,"
        " reasoning=excluded.reasoning",
        (decision_id, worker_id, vote, reasoning),
    )
    db.execute("UPDATE workers SET votes_cast = votes_cast + 1 WHERE id = ?", (worker_id,))
    return db.execute(
        "SELECT * FROM quorum_votes WHERE decision_id = ? AND worker_id = ?",
        (decision_id, worker_id)).fetchone()


def get_voter_health(db: sqlite3.Connection, room_id: int,
                     threshold: float = 0.5) -> list[dict]:
    """Per-worker quorum participation (reference db-queries.ts:1368-1383;
    unlike the reference, votes_cast/votes_missed are actually incremented
    on vote cast / decision resolution here)."""
    out = []
    for w in list_room_workers(db, room_id):
        total = (w["votes_cast"] or 0) + (w["votes_missed"] or 0)
        rate = 1.0 if total == 0 else w["votes_cast"] / total
        out.append({"worker_id": w["id"], "worker_name": w["name"],
                    "votes_cast": w["votes_cast"],
                    "votes_missed": w["votes_missed"],
                    "total_decisions": total,
                    "participation_rate": rate,
                    "is_healthy": rate >= threshold})
    return out


def get_votes(db: sqlite3.Connection, decision_id: int) -> list[dict]:
    return db.execute(
        "SELECT * FROM quorum_votes WHERE decision_id = ? ORDER BY id", (decision_id,)
    ).fetchall()


def set_keeper_vote(db: sqlite3.Connection, decision_id: int, vote: str) -> None:
    db.execute("UPDATE quorum_decisions SET keeper_vote = ? WHERE id = ?",
               (vote, decision_id))


# ---------------------------------------------------------------- goals


def create_goal(db: sqlite3.Connection, room_id: int, description: str,
                parent_goal_id: int | None = None,
                assigned_worker_id: int | None = None) -> dict:
    cur = db.execute(
        "INSERT INTO goals (room_id, description, parent_goal_id, assigned_worker_id)"
        " VALUES (?,?,?,?)",
        (room_id, description, parent_goal_id, assigned_worker_id),
    )
    return get_goal(db, cur.lastrowid)


def get_goal(db: sqlite3.Connection, goal_id: int) -> Optional[dict]:
    return db.execute("SELECT * FROM goals WHERE id = ?", (goal_id,)).fetchone()


def list_room_goals(db: sqlite3.Connection, room_id: int,
                    status: str | None = None) -> list[dict]:
    if status:
        return db.execute(
            "SELECT * FROM goals WHERE room_id = ? AND status = ? ORDER BY id",
            (room_id, status)).fetchall()
    return db.execute("SELECT * FROM goals WHERE room_id = ? ORDER BY id",
                      (room_id,)).fetchall()


def list_worker_goals(db: sqlite3.Connection, worker_id: int) -> list[dict]:
    return db.execute(
        "SELECT * FROM goals WHERE assigned_worker_id = ? AND status IN"
        " ('active','in_progress') ORDER BY id", (worker_id,)).fetchall()


def update_goal(db: sqlite3.Connection, goal_id: int, **fields: Any) -> Optional[dict]:
    allowed = {"description", "status", "assigned_worker_id", "progress", "parent_goal_id"}
    sets, vals = [], []
    for k, v in fields.items():
        if k not in allowed:
            raise ValueError(f"unknown goal field: {k}")
        sets.append(f"{k} = ?")
        vals.append(v)
    if sets:
        vals += [now_iso(), goal_id]
        db.execute(f"UPDATE goals SET {', '.join(sets)}, updated_at = ? WHERE id = ?", vals)
    return get_goal(db, goal_id)


def add_goal_update(db: sqlite3.Connection, goal_id: int, observation: str,
                    worker_id: int | None = None,
                    metric_value: float | None = None) -> int:
    cur = db.execute(
        "INSERT INTO goal_updates (goal_id, worker_id, observation, metric_value)"
        " VALUES (?,?,?,?)", (goal_id, worker_id, observation, metric_value))
    return cur.lastrowid


def recalc_goal_progress(db: sqlite3.Connection, goal_id: int) -> float:
    """Parent progress = mean of children progress, completed children = 1.0
    (reference behavior: db-queries.ts:1488)."""
    children = db.execute(
        "SELECT status, progress FROM goals WHERE parent_goal_id = ?", (goal_id,)
    ).fetchall()
    if not children:
        row = get_goal(db, goal_id)
        return row["progress"] if row else 0.0
    total = sum(1.0 if c["status"] == "completed" else (c["progress"] or 0.0)
                for c in children)
    progress = total / len(children)
    db.execute("UPDATE goals SET progress = ?, updated_at = ? WHERE id = ?",
               (progress, now_iso(), goal_id))
    return progress


def get_goal_tree(db: sqlite3.Connection, room_id: int) -> list[dict]:
    goals = list_room_goals(db, room_id)
    by_parent: dict[Optional[int], list[dict]] = {}
    for g in goals:
        by_parent.setdefault(g["parent_goal_id"], []).append(g)

    def attach(g: dict) -> dict:
        g = dict(g)
        g["children"] = [attach(c) for c in by_parent.get(g["id"], [])]
        return g

    return [attach(g) for g in by_parent.get(None, [])]


# ---------------------------------------------------------------- skills


def create_skill(db: sqlite3.Connection, room_id: int | None, name: str, content: str,
                 activation_context: str | None = None, auto_activate: bool = False,
                 agent_created: bool = False,
                 created_by_worker_id: int | None = None) -> dict:
    cur = db.execute(
        "INSERT INTO skills (room_id, name, content, activation_context, auto_activate,"
        " agent_created, created_by_worker_id) VALUES (?,?,?,?,?,?,?)",
        (room_id, name, content, activation_context, int(auto_activate),
         int(agent_created), created_by_worker_id),
    )
    return get_skill(db, cur.lastrowid)


def get_skill(db: sqlite3.Connection, skill_id: int) -> Optional[dict]:
    return db.execute("SELECT * FROM skills WHERE id = ?", (skill_id,)).fetchone()


def list_room_skills(db: sqlite3.Connection, room_id: int | None) -> list[dict]:
    if room_id is None:
        return db.execute("SELECT * FROM skills WHERE room_id IS NULL ORDER BY id").fetchall()
    return db.execute("SELECT * FROM skills WHERE room_id = ? ORDER BY id",
                      (room_id,)).fetchall()


def update_skill(db: sqlite3.Connection, skill_id: int, content: str | None = None,
                 activation_context: str | None = None,
                 auto_activate: bool | None = None) -> Optional[dict]:
    skill = get_skill(db, skill_id)
    if skill is None:
        return None
    db.execute(
        "UPDATE skills SET content = COALESCE(?, content),"
        " activation_context = COALESCE(?, activation_context),"
        " auto_activate = COALESCE(?, auto_activate),"
        " version = version + 1, updated_at = ? WHERE id = ?",
        (content, activation_context,
         None if auto_activate is None else int(auto_activate), now_iso(), skill_id),
    )
    return get_skill(db, skill_id)


def delete_skill(db: sqlite3.Connection, skill_id: int) -> None:
    db.execute("DELETE FROM skills WHERE id = ?", (skill_id,))


def get_active_skills_for_context(db: sqlite3.Connection, room_id: int,
                                  context: str) -> list[dict]:
    """Keyword activation: a skill activates if auto_activate or any of its
    comma-separated activation_context keywords appears in the cycle context
    (reference: db-queries.ts:1577, skills.ts:5-35)."""
    skills = db.execute(
        "SELECT * FROM skills WHERE room_id = ? OR room_id IS NULL ORDER BY id",
        (room_id,)).fetchall()
    ctx = context.lower()
    out = []
    for s in skills:
        if s["auto_activate"]:
            out.append(s)
            continue
        ac = (s["activation_context"] or "").strip()
        if ac and any(kw.strip().lower() in ctx for kw in ac.split(",") if kw.strip()):
            out.append(s)
    return out


# ---------------------------------------------------------------- memory


def create_entity(db: sqlite3.Connection, name: str, entity_type: str = "fact",
                  category: str | None = None, room_id: int | None = None,
                  observations: Iterable[str] = ()) -> dict:
    cur = db.execute(
        "INSERT INTO entities (name, type, category, room_id) VALUES (?,?,?,?)",
        (name, entity_type, category, room_id),
    )
    eid = cur.lastrowid
    for content in observations:
        add_observation(db, eid, content)
    return get_entity(db, eid)


def get_entity(db: sqlite3.Connection, entity_id: int) -> Optional[dict]:
    return db.execute("SELECT * FROM entities WHERE id = ?", (entity_id,)).fetchone()


def get_entity_by_name(db: sqlite3.Connection, name: str,
                       room_id: int | None = None) -> Optional[dict]:
    if room_id is not None:
        return db.execute(
            "SELECT * FROM entities WHERE name = ? AND room_id = ?", (name, room_id)
        ).fetchone()
    return db.execute("SELECT * FROM entities WHERE name = ?", (name,)).fetchone()


def delete_entity(db: sqlite3.Connection, entity_id: int) -> None:
    db.execute("DELETE FROM entities WHERE id = ?", (entity_id,))


def add_observation(db: sqlite3.Connection, entity_id: int, content: str,
                    source: str = "agent") -> int:
    cur = db.execute(
        "INSERT INTO observations (entity_id, content, source) VALUES (?,?,?)",
        (entity_id, content, source),
    )
    # keep FTS content column in sync (entities row carries name; observations
    # are searchable through the content column)
    db.execute(
        "INSERT INTO memory_fts(memory_fts, rowid, name, content, category)"
        " SELECT 'delete', e.id, e.name, '', e.category FROM entities e WHERE e.id = ?",
        (entity_id,),
    )
    db.execute(
        "INSERT INTO memory_fts(rowid, name, content, category)"
        " SELECT e.id, e.name,"
        " (SELECT group_concat(o.content, ' ') FROM observations o WHERE o.entity_id = e.id),"
        " e.category FROM entities e WHERE e.id = ?",
        (entity_id,),
    )
    return cur.lastrowid


def get_observations(db: sqlite3.Connection, entity_id: int) -> list[dict]:
    return db.execute(
        "SELECT * FROM observations WHERE entity_id = ? ORDER BY id", (entity_id,)
    ).fetchall()


def create_relation(db: sqlite3.Connection, from_entity: int, to_entity: int,
                    relation_type: str) -> int:
    cur = db.execute(
        "INSERT INTO relations (from_entity, to_entity, relation_type) VALUES (?,?,?)",
        (from_entity, to_entity, relation_type),
    )
    return cur.lastrowid


def fts_search(db: sqlite3.Connection, query: str, limit: int = 20,
               room_id: int | None = None) -> list[dict]:
    """FTS5 keyword search returning entities ranked by bm25."""
    # sanitize: quote each term to avoid FTS syntax errors on user input
    terms = [t for t in query.replace('"', " ").split() if t]
    if not terms:
        return []
    match = " OR ".join(f'"{t}"' for t in terms)
    sql = (
        "SELECT e.*, bm25(memory_fts) AS rank FROM memory_fts f"
        " JOIN entities e ON e.id = f.rowid WHERE memory_fts MATCH ?"
    )
    params: list[Any] = [match]
    if room_id is not None:
        sql += " AND (e.room_id = ? OR e.room_id IS NULL)"
        params.append(room_id)
    sql += " ORDER BY rank LIMIT ?"
    params.append(limit)
    try:
        return db.execute(sql, params).fetchall()
    except sqlite3.OperationalError:
        return []


# --- embeddings ---


def vector_to_blob(vec: Iterable[float]) -> bytes:
    """Float32 little-endian blob — same codec as the reference
    (embeddings.ts:116-122: Float32Array buffer)."""
    v = list(vec)
    return struct.pack(f"<{len(v)}f", *v)


def blob_to_vector(blob: bytes) -> list[float]:
    n = len(blob) // 4
    return list(struct.unpack(f"<{n}f", blob))


def upsert_embedding(db: sqlite3.Connection, entity_id: int, vector: Iterable[float],
                     text_hash: str, source_type: str = "entity",
                     source_id: int | None = None,
                     model: str = "all-MiniLM-L6-v2") -> int:
    blob = vector_to_blob(vector)
    dims = len(blob) // 4
    sid = source_id if source_id is not None else entity_id
    db.execute(
        "INSERT INTO embeddings (entity_id, source_type, source_id, text_hash, vector,"
        " model, dimensions) VALUES (?,?,?,?,?,?,?)"
        " ON CONFLICT(source_type, source_id, model) DO UPDATE SET"
        " vector = excluded.vector, text_hash = excluded.text_hash,"
        " dimensions = excluded.dimensions",
        (entity_id, source_type, sid, text_hash, blob, model, dims),
    )
    db.execute("UPDATE entities SET embedded_at = ? WHERE id = ?", (now_iso(), entity_id))
    row = db.execute(
        "SELECT id FROM embeddings WHERE source_type = ? AND source_id = ? AND model = ?",
        (source_type, sid, model)).fetchone()
    return row["id"]


def get_embedding(db: sqlite3.Connection, entity_id: int) -> Optional[dict]:
    return db.execute(
        "SELECT * FROM embeddings WHERE entity_id = ? LIMIT 1", (entity_id,)
    ).fetchone()


def get_unembedded_entities(db: sqlite3.Connection, limit: int = 64) -> list[dict]:
    return db.execute(
        "SELECT e.* FROM entities e LEFT JOIN embeddings m ON m.entity_id = e.id"
        " WHERE m.id IS NULL ORDER BY e.id LIMIT ?", (limit,)).fetchall()


def all_embeddings(db: sqlite3.Connection, room_id: int | None = None) -> list[dict]:
    if room_id is not None:
        return db.execute(
            "SELECT m.id, m.entity_id, m.vector, m.dimensions FROM embeddings m"
            " JOIN entities e ON e.id = m.entity_id"
            " WHERE e.room_id = ? OR e.room_id IS NULL", (room_id,)).fetchall()
    return db.execute(
        "SELECT id, entity_id, vector, dimensions FROM embeddings").fetchall()


def semantic_search_host(db: sqlite3.Connection, query_vec: list[float], limit: int = 10,
                         room_id: int | None = None) -> list[tuple[int, float]]:
    """Host-side brute-force cosine over stored blobs — durable fallback path.
    The hot path is the GPU store (room_amd.memory.vector_store). Returns
    (entity_id, similarity) sorted desc."""
    import math

    qn = math.sqrt(sum(x * x for x in query_vec)) or 1.0
    scored: list[tuple[int, float]] = []
    for row in all_embeddings(db, room_id):
        v = blob_to_vector(row["vector"])
        if len(v) != len(query_vec):
            continue
        dot = sum(a * b for a, b in zip(query_vec, v))
        vn = math.sqrt(sum(x * x for x in v)) or 1.0
        scored.append((row["entity_id"], dot / (qn * vn)))
    scored.sort(key=lambda t: -t[1])
    return scored[:limit]


def hybrid_search(db: sqlite3.Connection, query: str, query_vec: list[float] | None,
                  limit: int = 5, room_id: int | None = None,
                  semantic_hits: list[tuple[int, float]] | None = None) -> list[dict]:
    """Hybrid = FTS reciprocal-rank (k=60) × 0.4 + cosine similarity × 0.6
    (reference fusion: db-queries.ts:1021-1059). `semantic_hits` lets the GPU
    vector store supply the cosine side; otherwise host cosine is used."""
    fts_hits = fts_search(db, query, limit=20, room_id=room_id)
    if semantic_hits is None:
        semantic_hits = (semantic_search_host(db, query_vec, limit=20, room_id=room_id)
                         if query_vec else [])
    scores: dict[int, float] = {}
    for rank, row in enumerate(fts_hits):
        scores[row["id"]] = scores.get(row["id"], 0.0) + \
            HYBRID_FTS_WEIGHT * (1.0 / (HYBRID_RRF_K + rank + 1))
    for eid, sim in semantic_hits:
        scores[eid] = scores.get(eid, 0.0) + HYBRID_SEMANTIC_WEIGHT * sim
    ranked = sorted(scores.items(), key=lambda t: -t[1])[:limit]
    out = []
    for eid, score in ranked:
        e = get_entity(db, eid)
        if e:
            e = dict(e)
            e["score"] = score
            e["observations"] = [o["content"] for o in get_observations(db, eid)]
            out.append(e)
    return out


# ---------------------------------------------------------------- escalations


def create_escalation(db: sqlite3.Connection, room_id: int, question: str,
                      from_agent_id: int | None = None,
                      to_agent_id: int | None = None) -> dict:
    cur = db.execute(
        "INSERT INTO escalations (room_id, from_agent_id, to_agent_id, question)"
        " VALUES (?,?,?,?)", (room_id, from_agent_id, to_agent_id, question))
    return db.execute("SELECT * FROM escalations WHERE id = ?", (cur.lastrowid,)).fetchone()


def answer_escalation(db: sqlite3.Connection, escalation_id: int, answer: str) -> None:
    db.execute(
        "UPDATE escalations SET answer = ?, status = 'answered', resolved_at = ?"
        " WHERE id = ?", (answer, now_iso(), escalation_id))


def list_escalations(db: sqlite3.Connection, room_id: int,
                     status: str | None = None) -> list[dict]:
    if status:
        return db.execute(
            "SELECT * FROM escalations WHERE room_id = ? AND status = ? ORDER BY id DESC",
            (room_id, status)).fetchall()
    return db.execute(
        "SELECT * FROM escalations WHERE room_id = ? ORDER BY id DESC", (room_id,)
    ).fetchall()


def get_pending_keeper_answers(db: sqlite3.Connection, room_id: int) -> list[dict]:
    """Answered escalations not yet consumed by the queen."""
    return db.execute(
        "SELECT * FROM escalations WHERE room_id = ? AND status = 'answered'"
        " ORDER BY id", (room_id,)).fetchall()


def mark_escalation_consumed(db: sqlite3.Connection, escalation_id: int) -> None:
    db.execute("UPDATE escalations SET status = 'closed' WHERE id = ?", (escalation_id,))


# ---------------------------------------------------------------- room messages


def create_room_message(db: sqlite3.Connection, room_id: int, direction: str,
                        subject: str, body: str, from_room_id: str | None = None,
                        to_room_id: str | None = None) -> dict:
    cur = db.execute(
        "INSERT INTO room_messages (room_id, direction, from_room_id, to_room_id,"
        " subject, body) VALUES (?,?,?,?,?,?)",
        (room_id, direction, from_room_id, to_room_id, subject, body))
    return db.execute("SELECT * FROM room_messages WHERE id = ?",
                      (cur.lastrowid,)).fetchone()


def get_unread_room_messages(db: sqlite3.Connection, room_id: int) -> list[dict]:
    return db.execute(
        "SELECT * FROM room_messages WHERE room_id = ? AND direction = 'inbound'"
        " AND status = 'unread' ORDER BY id", (room_id,)).fetchall()


def mark_room_message_read(db: sqlite3.Connection, message_id: int) -> None:
    db.execute("UPDATE room_messages SET status = 'read' WHERE id = ?", (message_id,))


def list_room_messages(db: sqlite3.Connection, room_id: int, limit: int = 50) -> list[dict]:
    return db.execute(
        "SELECT * FROM room_messages WHERE room_id = ? ORDER BY id DESC LIMIT ?",
        (room_id, limit)).fetchall()


# ---------------------------------------------------------------- cycles & logs


def create_worker_cycle(db: sqlite3.Connection, worker_id: int, room_id: int,
                        model: str | None = None) -> int:
    cur = db.execute(
        "INSERT INTO worker_cycles (worker_id, room_id, model) VALUES (?,?,?)",
        (worker_id, room_id, model))
    return cur.lastrowid


def complete_worker_cycle(db: sqlite3.Connection, cycle_id: int, status: str,
                          error_message: str | None = None,
                          duration_ms: int | None = None,
                          input_tokens: int | None = None,
                          output_tokens: int | None = None) -> None:
    db.execute(
        "UPDATE worker_cycles SET status = ?, error_message = ?, finished_at = ?,"
        " duration_ms = ?, input_tokens = ?, output_tokens = ? WHERE id = ?",
        (status, error_message, now_iso(), duration_ms, input_tokens, output_tokens,
         cycle_id))


def get_worker_cycle(db: sqlite3.Connection, cycle_id: int) -> Optional[dict]:
    return db.execute("SELECT * FROM worker_cycles WHERE id = ?", (cycle_id,)).fetchone()


def list_room_cycles(db: sqlite3.Connection, room_id: int, limit: int = 50) -> list[dict]:
    return db.execute(
        "SELECT * FROM worker_cycles WHERE room_id = ? ORDER BY id DESC LIMIT ?",
        (room_id, limit)).fetchall()


def add_cycle_logs(db: sqlite3.Connection, cycle_id: int,
                   entries: list[tuple[int, str, str]]) -> None:
    """Batched insert: entries = [(seq, entry_type, content)] — the 1s flush
    cadence batching lives in core.log_buffer."""
    db.executemany(
        "INSERT INTO cycle_logs (cycle_id, seq, entry_type, content) VALUES (?,?,?,?)",
        [(cycle_id, s, t, c) for s, t, c in entries])


def get_cycle_logs(db: sqlite3.Connection, cycle_id: int,
                   after_seq: int = -1) -> list[dict]:
    return db.execute(
        "SELECT * FROM cycle_logs WHERE cycle_id = ? AND seq > ? ORDER BY seq",
        (cycle_id, after_seq)).fetchall()


def prune_old_cycles(db: sqlite3.Connection, room_id: int, keep: int = 200) -> int:
    cur = db.execute(
        "DELETE FROM worker_cycles WHERE room_id = ? AND id NOT IN"
        " (SELECT id FROM worker_cycles WHERE room_id = ? ORDER BY id DESC LIMIT ?)",
        (room_id, room_id, keep))
    return cur.rowcount


def cleanup_stale_cycles(db: sqlite3.Connection) -> int:
    """Mark running cycles failed on boot ('Server restarted',
    reference: db-queries.ts:2388-2393)."""
    cur = db.execute(
        "UPDATE worker_cycles SET status = 'failed', error_message = 'Server restarted',"
        " finished_at = ? WHERE status = 'running'", (now_iso(),))
    return cur.rowcount


def get_room_token_usage(db: sqlite3.Connection, room_id: int) -> dict:
    row = db.execute(
        "SELECT COALESCE(SUM(input_tokens),0) AS input_tokens,"
        " COALESCE(SUM(output_tokens),0) AS output_tokens,"
        " COUNT(*) AS cycles FROM worker_cycles WHERE room_id = ?", (room_id,)
    ).fetchone()
    return row


# ---------------------------------------------------------------- sessions


def get_agent_session(db: sqlite3.Connection, worker_id: int) -> Optional[dict]:
    return db.execute(
        "SELECT * FROM agent_sessions WHERE worker_id = ?", (worker_id,)).fetchone()


def save_agent_session(db: sqlite3.Connection, worker_id: int,
                       session_id: str | None = None,
                       messages_json: str | None = None, model: str = "",
                       turn_count: int = 0) -> None:
    db.execute(
        "INSERT INTO agent_sessions (worker_id, session_id, messages_json, model,"
        " turn_count, updated_at) VALUES (?,?,?,?,?,?)"
        " ON CONFLICT(worker_id) DO UPDATE SET session_id=excluded.session_id,"
        " messages_json=excluded.messages_json, model=excluded.model,"
        " turn_count=excluded.turn_count, updated_at=excluded.updated_at",
        (worker_id, session_id, messages_json, model, turn_count, now_iso()))


def clear_agent_session(db: sqlite3.Connection, worker_id: int) -> None:
    db.execute("DELETE FROM agent_sessions WHERE worker_id = ?", (worker_id,))


# ---------------------------------------------------------------- tasks


def create_task(db: sqlite3.Connection, name: str, prompt: str,
                trigger_type: str = "cron", cron_expression: str | None = None,
                scheduled_at: str | None = None, room_id: int | None = None,
                worker_id: int | None = None, executor: str = "local",
                session_continuity: bool = False, max_runs: int | None = None,
                description: str | None = None, webhook_token: str | None = None,
                timeout_minutes: int | None = None,
                max_turns: int | None = None) -> dict:
    cur = db.execute(
        "INSERT INTO tasks (name, description, prompt, cron_expression, trigger_type,"
        " webhook_token, executor, scheduled_at, max_runs, worker_id,"
        " session_continuity, timeout_minutes, max_turns, room_id)"
        " VALUES (?,?,?,?,?,?,?,?,?,?,?,?,?,?)",
        (name, description, prompt, cron_expression, trigger_type, webhook_token,
         executor, scheduled_at, max_runs, worker_id, int(session_continuity),
         timeout_minutes, max_turns, room_id))
    return get_task(db, cur.lastrowid)


def get_task(db: sqlite3.Connection, task_id: int) -> Optional[dict]:
    return db.execute("SELECT * FROM tasks WHERE id = ?", (task_id,)).fetchone()


def get_task_by_webhook_token(db: sqlite3.Connection, token: str) -> Optional[dict]:
    return db.execute("SELECT * FROM tasks WHERE webhook_token = ?", (token,)).fetchone()


def list_tasks(db: sqlite3.Connection, room_id: int | None = None,
               status: str | None = None) -> list[dict]:
    sql, params = "SELECT * FROM tasks", []
    conds = []
    if room_id is not None:
        conds.append("room_id = ?")
        params.append(room_id)
    if status is not None:
        conds.append("status = ?")
        params.append(status)
    if conds:
        sql += " WHERE " + " AND ".join(conds)
    sql += " ORDER BY id"
    return db.execute(sql, params).fetchall()


def update_task(db: sqlite3.Connection, task_id: int, **fields: Any) -> Optional[dict]:
    allowed = {"name", "description", "prompt", "cron_expression", "trigger_type",
               "status", "last_run", "last_result", "error_count", "scheduled_at",
               "max_runs", "run_count", "worker_id", "session_continuity",
               "session_id", "timeout_minutes", "max_turns", "learned_context",
               "room_id", "webhook_token"}
    sets, vals = [], []
    for k, v in fields.items():
        if k not in allowed:
            raise ValueError(f"unknown task field: {k}")
        sets.append(f"{k} = ?")
        vals.append(v)
    if sets:
        vals += [now_iso(), task_id]
        db.execute(f"UPDATE tasks SET {', '.join(sets)}, updated_at = ? WHERE id = ?", vals)
    return get_task(db, task_id)


def delete_task(db: sqlite3.Connection, task_id: int) -> None:
    db.execute("DELETE FROM tasks WHERE id = ?", (task_id,))


def get_due_once_tasks(db: sqlite3.Connection) -> list[dict]:
    return db.execute(
        "SELECT * FROM tasks WHERE trigger_type = 'once' AND status = 'active'"
        " AND scheduled_at <= ?", (now_iso(),)).fetchall()


def create_task_run(db: sqlite3.Connection, task_id: int,
                    session_id: str | None = None) -> int:
    cur = db.execute(
        "INSERT INTO task_runs (task_id, session_id) VALUES (?,?)",
        (task_id, session_id))
    return cur.lastrowid


def finish_task_run(db: sqlite3.Connection, run_id: int, status: str,
                    result: str | None = None, error_message: str | None = None,
                    result_file: str | None = None,
                    duration_ms: int | None = None) -> None:
    db.execute(
        "UPDATE task_runs SET status = ?, result = ?, error_message = ?,"
        " result_file = ?, duration_ms = ?, finished_at = ? WHERE id = ?",
        (status, result, error_message, result_file, duration_ms, now_iso(), run_id))


def get_task_run(db: sqlite3.Connection, run_id: int) -> Optional[dict]:
    return db.execute("SELECT * FROM task_runs WHERE id = ?", (run_id,)).fetchone()


def get_latest_task_run(db: sqlite3.Connection, task_id: int) -> Optional[dict]:
    return db.execute(
        "SELECT * FROM task_runs WHERE task_id = ? ORDER BY id DESC LIMIT 1",
        (task_id,)).fetchone()


def list_task_runs(db: sqlite3.Connection, task_id: int, limit: int = 20) -> list[dict]:
    return db.execute(
        "SELECT * FROM task_runs WHERE task_id = ? ORDER BY id DESC LIMIT ?",
        (task_id, limit)).fetchall()


def cleanup_stale_runs(db: sqlite3.Connection, max_age_hours: float = 2.0) -> int:
    cutoff = (datetime.now() - timedelta(hours=max_age_hours)).strftime("%Y-%m-%d %H:%M:%S")
    cur = db.execute(
        "UPDATE task_runs SET status = 'failed', error_message = 'Stale run cleaned up',"
        " finished_at = ? WHERE status = 'running' AND started_at < ?",
        (now_iso(), cutoff))
    return cur.rowcount


def cleanup_all_running_runs(db: sqlite3.Connection) -> int:
    cur = db.execute(
        "UPDATE task_runs SET status = 'failed', error_message = 'Server restarted',"
        " finished_at = ? WHERE status = 'running'", (now_iso(),))
    return cur.rowcount


def add_console_logs(db: sqlite3.Connection, run_id: int,
                     entries: list[tuple[int, str, str]]) -> None:
    db.executemany(
        "INSERT INTO console_logs (run_id, seq, entry_type, content) VALUES (?,?,?,?)",
        [(run_id, s, t, c) for s, t, c in entries])


def get_console_logs(db: sqlite3.Connection, run_id: int,
                     after_seq: int = -1) -> list[dict]:
    return db.execute(
        "SELECT * FROM console_logs WHERE run_id = ? AND seq > ? ORDER BY seq",
        (run_id, after_seq)).fetchall()


# ---------------------------------------------------------------- credentials


def set_credential(db: sqlite3.Connection, room_id: int, name: str,
                   value_encrypted: str, cred_type: str = "other",
                   provided_by: str = "keeper") -> dict:
    db.execute(
        "INSERT INTO credentials (room_id, name, type, value_encrypted, provided_by)"
        " VALUES (?,?,?,?,?) ON CONFLICT(room_id, name) DO UPDATE SET"
        " value_encrypted = excluded.value_encrypted, type = excluded.type",
        (room_id, name, cred_type, value_encrypted, provided_by))
    return db.execute(
        "SELECT * FROM credentials WHERE room_id = ? AND name = ?",
        (room_id, name)).fetchone()


def get_credential(db: sqlite3.Connection, room_id: int, name: str) -> Optional[dict]:
    return db.execute(
        "SELECT * FROM credentials WHERE room_id = ? AND name = ?",
        (room_id, name)).fetchone()


def list_credentials(db: sqlite3.Connection, room_id: int) -> list[dict]:
    return db.execute(
        "SELECT id, room_id, name, type, provided_by, created_at FROM credentials"
        " WHERE room_id = ? ORDER BY name", (room_id,)).fetchall()


def delete_credential(db: sqlite3.Connection, room_id: int, name: str) -> None:
    db.execute("DELETE FROM credentials WHERE room_id = ? AND name = ?", (room_id, name))


# ---------------------------------------------------------------- wallets


def create_wallet_row(db: sqlite3.Connection, room_id: int, address: str,
                      private_key_encrypted: str, chain: str = "base") -> dict:
    cur = db.execute(
        "INSERT INTO wallets (room_id, address, private_key_encrypted, chain)"
        " VALUES (?,?,?,?)", (room_id, address, private_key_encrypted, chain))
    return db.execute("SELECT * FROM wallets WHERE id = ?", (cur.lastrowid,)).fetchone()


def get_room_wallet(db: sqlite3.Connection, room_id: int) -> Optional[dict]:
    return db.execute(
        "SELECT * FROM wallets WHERE room_id = ? ORDER BY id LIMIT 1", (room_id,)
    ).fetchone()


def log_wallet_tx(db: sqlite3.Connection, wallet_id: int, tx_type: str, amount: str,
                  counterparty: str | None = None, tx_hash: str | None = None,
                  description: str | None = None, status: str = "confirmed",
                  category: str | None = None) -> int:
    cur = db.execute(
        "INSERT INTO wallet_transactions (wallet_id, type, amount, counterparty,"
        " tx_hash, description, status, category) VALUES (?,?,?,?,?,?,?,?)",
        (wallet_id, tx_type, amount, counterparty, tx_hash, description, status,
         category))
    return cur.lastrowid


def list_wallet_txs(db: sqlite3.Connection, wallet_id: int, limit: int = 50) -> list[dict]:
    return db.execute(
        "SELECT * FROM wallet_transactions WHERE wallet_id = ? ORDER BY id DESC LIMIT ?",
        (wallet_id, limit)).fetchall()


def set_wallet_identity(db: sqlite3.Connection, wallet_id: int, agent_id: str) -> None:
    db.execute("UPDATE wallets SET erc8004_agent_id = ? WHERE id = ?",
               (agent_id, wallet_id))


# ---------------------------------------------------------------- self-mod


def create_self_mod_audit(db: sqlite3.Connection, room_id: int | None,
                          worker_id: int | None, file_path: str,
                          old_hash: str | None, new_hash: str | None,
                          reason: str | None, reversible: bool = True) -> int:
    cur = db.execute(
        "INSERT INTO self_mod_audit (room_id, worker_id, file_path, old_hash,"
        " new_hash, reason, reversible) VALUES (?,?,?,?,?,?,?)",
        (room_id, worker_id, file_path, old_hash, new_hash, reason, int(reversible)))
    return cur.lastrowid


def create_self_mod_snapshot(db: sqlite3.Connection, audit_id: int, target_type: str,
                             target_id: int | None, old_content: str | None,
                             new_content: str | None) -> None:
    db.execute(
        "INSERT INTO self_mod_snapshots (audit_id, target_type, target_id,"
        " old_content, new_content) VALUES (?,?,?,?,?)",
        (audit_id, target_type, target_id, old_content, new_content))


def get_self_mod_audit(db: sqlite3.Connection, audit_id: int) -> Optional[dict]:
    return db.execute("SELECT * FROM self_mod_audit WHERE id = ?", (audit_id,)).fetchone()


def get_self_mod_snapshot(db: sqlite3.Connection, audit_id: int) -> Optional[dict]:
    return db.execute(
        "SELECT * FROM self_mod_snapshots WHERE audit_id = ?", (audit_id,)).fetchone()


def mark_self_mod_reverted(db: sqlite3.Connection, audit_id: int) -> None:
    db.execute("UPDATE self_mod_audit SET reverted = 1 WHERE id = ?", (audit_id,))


def list_self_mod_audit(db: sqlite3.Connection, room_id: int | None = None,
                        limit: int = 50) -> list[dict]:
    if room_id is not None:
        return db.execute(
            "SELECT * FROM self_mod_audit WHERE room_id = ? ORDER BY id DESC LIMIT ?",
            (room_id, limit)).fetchall()
    return db.execute(
        "SELECT * FROM self_mod_audit ORDER BY id DESC LIMIT ?", (limit,)).fetchall()


def last_self_mod_time(db: sqlite3.Connection, worker_id: int) -> Optional[str]:
    row = db.execute(
        "SELECT created_at FROM self_mod_audit WHERE worker_id = ?"
        " ORDER BY id DESC LIMIT 1", (worker_id,)).fetchone()
    return row["created_at"] if row else None


# ---------------------------------------------------------------- chat / clerk


def add_chat_message(db: sqlite3.Connection, room_id: int, role: str, content: str) -> int:
    cur = db.execute(
        "INSERT INTO chat_messages (room_id, role, content) VALUES (?,?,?)",
        (room_id, role, content))
    return cur.lastrowid


def list_chat_messages(db: sqlite3.Connection, room_id: int, limit: int = 50) -> list[dict]:
    return db.execute(
        "SELECT * FROM chat_messages WHERE room_id = ? ORDER BY id DESC LIMIT ?",
        (room_id, limit)).fetchall()


def add_clerk_message(db: sqlite3.Connection, role: str, content: str,
                      source: str | None = None) -> int:
    cur = db.execute(
        "INSERT INTO clerk_messages (role, content, source) VALUES (?,?,?)",
        (role, content, source))
    return cur.lastrowid


def list_clerk_messages(db: sqlite3.Connection, limit: int = 50) -> list[dict]:
    return db.execute(
        "SELECT * FROM clerk_messages ORDER BY id DESC LIMIT ?", (limit,)).fetchall()


def log_clerk_usage(db: sqlite3.Connection, source: str, model: str,
                    input_tokens: int, output_tokens: int, success: bool = True,
                    used_fallback: bool = False, attempts: int = 1) -> None:
    db.execute(
        "INSERT INTO clerk_usage (source, model, input_tokens, output_tokens,"
        " total_tokens, success, used_fallback, attempts) VALUES (?,?,?,?,?,?,?,?)",
        (source, model, input_tokens, output_tokens, input_tokens + output_tokens,
         int(success), int(used_fallback), attempts))


# ---------------------------------------------------------------- watches


def create_watch(db: sqlite3.Connection, path: str, action_prompt: str | None = None,
                 description: str | None = None, room_id: int | None = None) -> dict:
    cur = db.execute(
        "INSERT INTO watches (path, description, action_prompt, room_id) VALUES (?,?,?,?)",
        (path, description, action_prompt, room_id))
    return db.execute("SELECT * FROM watches WHERE id = ?", (cur.lastrowid,)).fetchone()


def list_watches(db: sqlite3.Connection, room_id: int | None = None) -> list[dict]:
    if room_id is not None:
        return db.execute("SELECT * FROM watches WHERE room_id = ? ORDER BY id",
                          (room_id,)).fetchall()
    return db.execute("SELECT * FROM watches ORDER BY id").fetchall()


def delete_watch(db: sqlite3.Connection, watch_id: int) -> None:
    db.execute("DELETE FROM watches WHERE id = ?", (watch_id,))
