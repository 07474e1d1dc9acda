"""Versioned reusable skills with keyword activation and a per-cycle budget
(reference: src/shared/skills.ts — 8 skills / 6000 chars per cycle)."""
from __future__ import annotations

import sqlite3

from ..db import queries as q
from .constants import SKILLS_CHAR_BUDGET, SKILLS_MAX_PER_CYCLE


def create_agent_skill(db: sqlite3.Connection, room_id: int | None, name: str,
                       content: str, activation_context: str | None = None,
                       auto_activate: bool = False,
                       created_by_worker_id: int | None = None) -> dict:
    return q.create_skill(db, room_id, name, content,
                          activation_context=activation_context,
                          auto_activate=auto_activate, agent_created=True,
                          created_by_worker_id=created_by_worker_id)


def load_skills_for_agent(db: sqlite3.Connection, room_id: int,
                          context: str) -> tuple[str, list[dict]]:
    """Select activated skills under the cycle budget and render the prompt
    block. Returns (skills_block, skills_used)."""
    active = q.get_active_skills_for_context(db, room_id, context)
    used: list[dict] = []
    parts: list[str] = []
    chars = 0
    for s in active[:SKILLS_MAX_PER_CYCLE]:
        block = f"### Skill: {s['name']} (v{s['version']})\n{s['content']}\n"
        if chars + len(block) > SKILLS_CHAR_BUDGET:
            break
        parts.append(block)
        chars += len(block)
        used.append(s)
    if not parts:
        return "", []
    return "## Active skills\n" + "\n".join(parts), used
