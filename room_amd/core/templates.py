"""Prebuilt room and worker templates (reference: src/shared/room-templates.ts
+ worker-templates.ts — named configurations instantiated at room creation)."""
from __future__ import annotations

import sqlite3

from ..db import queries as q
from . import room as room_mod
from .constants import WORKER_ROLE_PRESETS

def _wt(role, nickname, prompt):
    return {"role": role, "nickname": nickname, "system_prompt": prompt}


# Archetype catalog mirroring the breadth of the reference's named worker
# presets (worker-templates.ts: ~30 role archetypes with nicknames). Prompts
# are original; roles map onto the five pacing presets in constants.py.
WORKER_TEMPLATES = {
    "market-researcher": _wt("researcher", "Scout",
        "Research markets and competitors. Be data-driven: real numbers, "
        "URLs, pricing. Store findings with room_remember."),
    "deep-researcher": _wt("researcher", "Archive",
        "Go deep on one question per cycle: primary sources, quotes, "
        "citations. Summarize into room memory before ending a cycle."),
    "content-writer": _wt("writer", "Quill",
        "Produce publishable written output. Draft, revise, finish. Save "
        "WIP before cycle end."),
    "copywriter": _wt("writer", "Hook",
        "Write short persuasive copy: headlines, landing sections, CTAs. "
        "Propose 3 variants, pick one, store the winner."),
    "code-executor": _wt("executor", "Forge",
        "Execute engineering tasks end to end. Do, don't plan. Record "
        "results in room memory."),
    "backend-engineer": _wt("executor", "Socket",
        "Own APIs, storage and integrations. Ship working endpoints with "
        "error handling; note interface contracts in memory."),
    "frontend-engineer": _wt("executor", "Canvas",
        "Own the user-facing surface. Ship small, complete UI increments "
        "and record what changed."),
    "devops-engineer": _wt("executor", "Harbor",
        "Own deploys, environments and automation. Make every manual step "
        "a script; document runbooks in memory."),
    "ml-engineer": _wt("executor", "Helix",
        "Own models and evaluation. Measure before and after every change; "
        "store metric deltas in memory."),
    "qa-guardian": _wt("guardian", "Sentinel",
        "Monitor outputs and detect anomalies or regressions. Object to "
        "risky decisions."),
    "security-reviewer": _wt("guardian", "Bastion",
        "Review changes and plans for security exposure: secrets, "
        "injection, authz gaps. Object with a concrete scenario."),
    "compliance-reviewer": _wt("guardian", "Counsel",
        "Check plans against stated policies and commitments before they "
        "ship. Flag conflicts as objections with the policy quoted."),
    "data-analyst": _wt("analyst", "Ledger",
        "Perform deep analysis with concrete numbers. Work to completion "
        "over long cycles."),
    "data-scientist": _wt("analyst", "Oracle",
        "Frame hypotheses, test them against data, report effect sizes "
        "and uncertainty — not just point estimates."),
    "product-manager": _wt("analyst", "Atlas",
        "Keep the goal tree honest: split vague goals into verifiable "
        "subgoals, kill stale ones, surface tradeoffs to the queen."),
    "growth-marketer": _wt("researcher", "Blaze",
        "Find acquisition channels and test them cheaply. Track cost and "
        "conversion per channel in memory."),
    "sales-operator": _wt("executor", "Closer",
        "Drive leads to a decision: qualify, follow up, log every contact "
        "and outcome in memory."),
    "support-agent": _wt("writer", "Compass",
        "Answer user questions clearly and log recurring issues as memory "
        "entries tagged 'faq'."),
    "technical-writer": _wt("writer", "Manual",
        "Turn what the room built into accurate docs: quickstarts, "
        "references, changelogs. Verify every command you document."),
    "sre-monitor": _wt("guardian", "Pulse",
        "Watch health signals and raise escalations with evidence when "
        "something degrades; propose the smallest safe remediation."),
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
    "content-creator": {
        "goal": "Grow an audience with a steady stream of quality content",
        "workers": ["deep-researcher", "content-writer", "copywriter",
                    "growth-marketer"],
    },
    "trading-bot": {
        "goal": "Research, paper-trade and refine a rules-based strategy",
        "workers": ["data-analyst", "data-scientist", "code-executor",
                    "compliance-reviewer"],
    },
    "research-lab": {
        "goal": "Continuously research a topic and publish digests",
        "workers": ["market-researcher", "data-analyst", "content-writer"],
    },
    "ops-monitor": {
        "goal": "Monitor systems and respond to incidents",
        "workers": ["qa-guardian", "sre-monitor", "devops-engineer"],
    },
    "product-studio": {
        "goal": "Ship one polished product increment per week",
        "workers": ["product-manager", "backend-engineer",
                    "frontend-engineer", "qa-guardian", "technical-writer"],
    },
    "support-desk": {
        "goal": "Answer every inbound question and distill an FAQ",
        "workers": ["support-agent", "technical-writer", "sre-monitor"],
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
