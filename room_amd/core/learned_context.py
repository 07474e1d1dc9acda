"""Learned-context distillation (reference: src/shared/learned-context.ts).

After ≥3 runs (and every 3), a 1-turn model call distills run history into a
≤1500-char "methodology memo" stored on the task.
"""
from __future__ import annotations

import sqlite3

from ..db import queries as q


def should_distill(run_count: int, has_context: bool = False) -> bool:
    """≥3 runs, then: first distillation as soon as eligible, refresh every
    3 runs after (learned-context.ts:20-31)."""
    if run_count < 3:
        return False
    return (not has_context) or run_count % 3 == 0


def distill_learned_context(db: sqlite3.Connection, task_id: int,
                            model: str = "stub") -> str | None:
    """Distill run history + tool/console activity into a ≤1500-char memo
    (learned-context.ts:40-104 includes the runs' tool logs, not just their
    final results — the methodology lives in the steps)."""
    runs = q.list_task_runs(db, task_id, limit=6)
    if len(runs) < 3:
        return None
    lines = []
    for r in runs:
        lines.append(f"- run {r['id']} [{r['status']}]: "
                     f"{(r['result'] or r['error_message'] or '')[:200]}")
        try:
            logs = q.get_console_logs(db, r["id"])
            tool_lines = [l["content"][:90] for l in logs
                          if l.get("entry_type") in ("tool", "tool_use",
                                                     "system")][:4]
            for tl in tool_lines:
                lines.append(f"    · {tl}")
        except Exception:
            pass
    history = "\n".join(lines)
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
