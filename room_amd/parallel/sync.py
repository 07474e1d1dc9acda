"""Control-plane sync across swarm shards: goal / skill / WIP / announcement
refresh rides an RCCL broadcast from the queen rank.

In the multi-GPU swarm each rank owns a room shard in its own SQLite file.
The queen (rank 0) is the control plane — reference semantics: the queen
sets objectives, decomposes goals, announces decisions (room.ts:9-24,
quorum.ts:17-48). Worker shards on other ranks mirror that control state so
their cycle prompts observe the same objective/goals/skills the queen set,
without any cross-process SQLite access: one broadcast_blob per swarm step.

Announcements propagate too, so workers on any GPU can object locally; their
shard's objection rows are summed into the global tally by
core.quorum.tally's all-reduce.
"""
from __future__ import annotations

from .swarm import SwarmContext


def build_control_digest(db, room_id: int) -> dict:
    """Queen-rank snapshot of the control plane (small: KBs)."""
    from ..db import queries as q
    room = q.get_room(db, room_id)
    goals = [{"description": g["description"], "status": g["status"],
              "progress": g["progress"]}
             for g in q.list_room_goals(db, room_id)]
    skills = [{"name": s["name"], "content": s["content"],
               "activation_context": s["activation_context"],
               "version": s["version"],
               "auto_activate": s["auto_activate"]}
              for s in q.list_room_skills(db, room_id)]
    queen = q.get_worker(db, room["queen_worker_id"]) if room.get(
        "queen_worker_id") else None
    announced = [{"proposal": d["proposal"], "decision_type": d["decision_type"],
                  "effective_at": d["effective_at"]}
                 for d in q.list_room_decisions(db, room_id, status="announced")]
    return {"goal": room["goal"], "config": room["config"], "goals": goals,
            "skills": skills, "queen_wip": (queen or {}).get("wip"),
            "announced": announced}


def apply_control_digest(db, room_id: int, digest: dict) -> None:
    """Worker-rank upsert of the queen's control state into the local shard.
    Local rows the queen doesn't know about (worker-created goals, local
    votes/objections) are left untouched."""
    from ..db import queries as q
    room = q.get_room(db, room_id)
    if room is None:
        return
    if digest.get("goal") != room["goal"]:
        q.update_room(db, room_id, goal=digest.get("goal"))
    if digest.get("config") and digest["config"] != room["config"]:
        q.update_room(db, room_id, config=digest["config"])
    have_goals = {g["description"] for g in q.list_room_goals(db, room_id)}
    for g in digest.get("goals", []):
        if g["description"] not in have_goals:
            created = q.create_goal(db, room_id, g["description"])
            q.update_goal(db, created["id"], status=g["status"],
                          progress=g["progress"])
    have_skills = {s["name"]: s for s in q.list_room_skills(db, room_id)}
    for s in digest.get("skills", []):
        mine = have_skills.get(s["name"])
        if mine is None:
            q.create_skill(db, room_id, s["name"], s["content"],
                           activation_context=s["activation_context"])
        elif s["version"] > mine["version"]:
            q.update_skill(db, mine["id"], content=s["content"])
    have_props = {d["proposal"]
                  for d in q.list_room_decisions(db, room_id, status="announced")}
    for d in digest.get("announced", []):
        if d["proposal"] not in have_props:
            q.create_announcement(db, room_id, None, d["proposal"],
                                  d["decision_type"], d["effective_at"])


class SwarmSync:
    """Per-step control-plane refresh: rank 0 builds the digest, all ranks
    receive it over xGMI, worker ranks apply it to their shard."""

    def __init__(self, ctx: SwarmContext, ldb):
        self.ctx = ctx
        self.ldb = ldb

    def step(self, room_id: int) -> dict | None:
        ctx = self.ctx
        if not ctx.is_distributed:
            return None
        digest = None
        if ctx.rank == 0:
            with self.ldb as db:
                digest = build_control_digest(db, room_id)
        digest = ctx.broadcast_blob(digest, src=0)
        if ctx.rank != 0 and digest:
            with self.ldb as db:
                apply_control_digest(db, room_id, digest)
        return digest
