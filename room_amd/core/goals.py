"""Hierarchical goal tree (reference: src/shared/goals.ts + db-queries.ts:1401-1520)."""
from __future__ import annotations

import sqlite3

from ..db import queries as q


def set_room_objective(db: sqlite3.Connection, room_id: int, objective: str) -> dict:
    q.update_room(db, room_id, goal=objective)
    root = q.create_goal(db, room_id, objective)
    q.log_room_activity(db, room_id, "goal", f"Objective set: {objective}")
    return root


def decompose_goal(db: sqlite3.Connection, goal_id: int,
                   subgoals: list[str],
                   assigned_worker_ids: list[int | None] | None = None) -> list[dict]:
    parent = q.get_goal(db, goal_id)
    if parent is None:
        raise ValueError(f"Goal {goal_id} not found")
    created = []
    for i, desc in enumerate(subgoals):
        wid = assigned_worker_ids[i] if assigned_worker_ids and i < len(assigned_worker_ids) else None
        created.append(q.create_goal(db, parent["room_id"], desc,
                                     parent_goal_id=goal_id, assigned_worker_id=wid))
    q.update_goal(db, goal_id, status="in_progress")
    q.log_room_activity(db, parent["room_id"], "goal",
                        f"Goal #{goal_id} decomposed into {len(subgoals)} subgoals")
    return created


def assign_goal(db: sqlite3.Connection, goal_id: int, worker_id: int) -> dict:
    g = q.update_goal(db, goal_id, assigned_worker_id=worker_id, status="in_progress")
    if g:
        q.log_room_activity(db, g["room_id"], "goal",
                            f"Goal #{goal_id} assigned to worker #{worker_id}")
    return g


def complete_goal(db: sqlite3.Connection, goal_id: int,
                  observation: str | None = None,
                  worker_id: int | None = None) -> dict:
    g = q.update_goal(db, goal_id, status="completed", progress=1.0)
    if g is None:
        raise ValueError(f"Goal {goal_id} not found")
    if observation:
        q.add_goal_update(db, goal_id, observation, worker_id=worker_id)
    if g["parent_goal_id"]:
        q.recalc_goal_progress(db, g["parent_goal_id"])
    q.log_room_activity(db, g["room_id"], "goal",
                        f"Goal completed: {g['description']}", actor_id=worker_id)
    return g


def abandon_goal(db: sqlite3.Connection, goal_id: int,
                 reason: str | None = None) -> dict:
    g = q.update_goal(db, goal_id, status="abandoned")
    if g is None:
        raise ValueError(f"Goal {goal_id} not found")
    if reason:
        q.add_goal_update(db, goal_id, f"Abandoned: {reason}")
    if g["parent_goal_id"]:
        q.recalc_goal_progress(db, g["parent_goal_id"])
    q.log_room_activity(db, g["room_id"], "goal",
                        f"Goal abandoned: {g['description']}")
    return g


def update_goal_progress(db: sqlite3.Connection, goal_id: int, progress: float,
                         observation: str | None = None,
                         worker_id: int | None = None) -> dict:
    progress = max(0.0, min(1.0, progress))
    g = q.update_goal(db, goal_id, progress=progress,
                      status="in_progress" if progress < 1.0 else "completed")
    if g is None:
        raise ValueError(f"Goal {goal_id} not found")
    if observation:
        q.add_goal_update(db, goal_id, observation, worker_id=worker_id,
                          metric_value=progress)
    if g["parent_goal_id"]:
        q.recalc_goal_progress(db, g["parent_goal_id"])
    return g


def get_goal_tree(db: sqlite3.Connection, room_id: int) -> list[dict]:
    return q.get_goal_tree(db, room_id)
