"""Learned-context distillation (reference: src/shared/learned-context.ts).

After ≥3 runs (and every 3), a 1-turn model call distills run history into a
≤1500-char "methodology memo" stored on the task.
"""
from __future__ import annotations

import sqlite3

from ..db import queries as q


def should_distill(run_count: int) -> bool:
    return run_count >= 3 and run_count % 3 == 0


def distill_learned_context(db: sqlite3.Connection, task_id: int,
                            model: str = "stub") -> str | None:
    runs = q.list_task_runs(db, task_id, limit=6)
    if len(runs) < 3:
        return None
    history = "\n".join(
        f"- run {r['id']} [{r['status']}]: {(r['result'] or r['error_message'] or '')[:200]}"
        for r in runs)
    try:
        from ..engine.providers import resolve_engine
        from ..engine.types import AgentExecutionOptions
        engine = resolve_engine(model)
        text, _, _ = engine.chat(
            [{"role": "system",
              "content": "Distill this task's run history into a short "
                         "methodology memo (what works, what to avoid)."},
             {"role": "user", "content": history[:6000]}],
            [], AgentExecutionOptions(prompt="", model=model, max_new_tokens=256))
        return text[:1500]
    except Exception:
        return history[:1500]
