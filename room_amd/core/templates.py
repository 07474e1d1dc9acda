"""Prebuilt room and worker templates (reference: src/shared/room-templates.ts
+ worker-templates.ts — named configurations instantiated at room creation)."""
from __future__ import annotations

import sqlite3

from ..db import queries as q
from . import room as room_mod
from .constants import WORKER_ROLE_PRESETS

WORKER_TEMPLATES = {
    "market-researcher": {
        "role": "researcher",
        "system_prompt": "Research markets and competitors. Be data-driven: "
                         "real numbers, URLs, pricing. Store findings with "
                         "room_remember.",
    },
    "content-writer": {
        "role": "writer",
        "system_prompt": "Produce publishable written output. Draft, revise, "
                         "finish. Save WIP before cycle end.",
    },
    "code-executor": {
        "role": "executor",
        "system_prompt": "Execute engineering tasks end to end. Do, don't plan. "
                         "Record results in room memory.",
    },
    "qa-guardian": {
        "role": "guardian",
        "system_prompt": "Monitor outputs and detect anomalies or regressions. "
                         "Object to risky decisions.",
    },
    "data-analyst": {
        "role": "analyst",
        "system_prompt": "Perform deep analysis with concrete numbers. Work to "
                         "completion over long cycles.",
    },
}

ROOM_TEMPLATES = {
    "saas-builder": {
        "goal": "Build and launch a small SaaS product end to end",
        "workers": ["market-researcher", "code-executor", "content-writer",
                    "qa-guardian"],
    },
    "freelancer": {
        "goal": "Find, win and deliver freelance work",
        "workers": ["market-researcher", "content-writer", "code-executor"],
    },
    "research-lab": {
        "goal": "Continuously research a topic and publish digests",
        "workers": ["market-researcher", "data-analyst", "content-writer"],
    },
    "ops-monitor": {
        "goal": "Monitor systems and respond to incidents",
        "workers": ["qa-guardian", "code-executor"],
    },
}


def list_templates() -> dict:
    return {"rooms": sorted(ROOM_TEMPLATES), "workers": sorted(WORKER_TEMPLATES)}


def instantiate_room_template(db: sqlite3.Connection, template: str, name: str,
                              worker_model: str = "qwen3-coder-30b") -> dict:
    tpl = ROOM_TEMPLATES.get(template)
    if tpl is None:
        raise ValueError(f"unknown room template: {template}")
    room = room_mod.create_room(db, name, goal=tpl["goal"],
                                worker_model=worker_model)
    for wt_name in tpl["workers"]:
        wt = WORKER_TEMPLATES[wt_name]
        preset = WORKER_ROLE_PRESETS.get(wt["role"], {})
        q.create_worker(db, wt_name, wt["system_prompt"], role=wt["role"],
                        room_id=room["id"], model=worker_model,
                        cycle_gap_ms=preset.get("cycleGapMs"),
                        max_turns=preset.get("maxTurns"))
    return q.get_room(db, room["id"])
