"""Scheduled-task runner (reference: src/shared/task-runner.ts, 659 LoC).

Semantics preserved: per-room concurrency slots 1-10 (:53-93), session
continuity with rotation after 20 runs (:33,399-412), learned-context +
memory injection into the prompt (:414-442), rate-limit retry (:137-209),
markdown result files (:631-659), terminal-error auto-pause (:609-624),
cross-process running check via the latest task_run row (:239-242).
The executor is the in-process GPU engine instead of a Claude CLI subprocess.
"""
from __future__ import annotations

import asyncio
import os
import re
import time
import uuid
from pathlib import Path
from typing import Optional

from ..db import LockedDb
from ..db import queries as q
from ..engine.providers import execute_agent
from ..engine.types import AgentExecutionOptions, ToolCall
from . import agent_tools
from .constants import (RATE_LIMIT_MAX_RETRIES, TASK_MAX_CONCURRENT,
                        TASK_MIN_CONCURRENT, TASK_SESSION_ROTATE_RUNS)
from .learned_context import distill_learned_context, should_distill
from .log_buffer import CycleLogBuffer
from .rate_limit import abortable_sleep, detect_rate_limit

TERMINAL_ERROR_PATTERNS = (
    re.compile(r"invalid api key", re.I),
    re.compile(r"authentication", re.I),
    re.compile(r"model not found", re.I),
)


def results_dir() -> Path:
    base = os.environ.get("ROOMAMD_RESULTS_DIR",
                          str(Path.home() / "RoomAMD" / "results"))
    p = Path(base)
    p.mkdir(parents=True, exist_ok=True)
    return p


class TaskRunner:
    def __init__(self, ldb: LockedDb, bus=None, memory=None,
                 default_model: str = "stub"):
        self.ldb = ldb
        self.bus = bus
        self.memory = memory
        self.default_model = default_model
        self.running_tasks: set[int] = set()
        self._room_slots: dict[int, int] = {}  # room_id → in-flight count

    def is_task_running(self, task_id: int) -> bool:
        if task_id in self.running_tasks:
            return True
        with self.ldb as db:  # cross-process check via latest run row
            run = q.get_latest_task_run(db, task_id)
        return bool(run and run["status"] == "running")

    def _acquire_slot(self, db, room_id: int | None) -> bool:
        if room_id is None:
            return True
        room = q.get_room(db, room_id)
        limit = max(TASK_MIN_CONCURRENT,
                    min(TASK_MAX_CONCURRENT,
                        (room or {}).get("max_concurrent_tasks") or 3))
        if self._room_slots.get(room_id, 0) >= limit:
            return False
        self._room_slots[room_id] = self._room_slots.get(room_id, 0) + 1
        return True

    def _release_slot(self, room_id: int | None) -> None:
        if room_id is not None and self._room_slots.get(room_id, 0) > 0:
            self._room_slots[room_id] -= 1

    async def execute_task(self, task_id: int) -> Optional[dict]:
        with self.ldb as db:
            task = q.get_task(db, task_id)
        if task is None or task["status"] != "active":
            return None
        if self.is_task_running(task_id):
            return None

        with self.ldb as db:
            if not self._acquire_slot(db, task["room_id"]):
                return None
        self.running_tasks.add(task_id)
        try:
            return await self._run(task)
        finally:
            self.running_tasks.discard(task_id)
            self._release_slot(task["room_id"])

    async def _run(self, task: dict) -> dict:
        task_id = task["id"]
        start = time.time()

        # built-in executors (reference task-runner.ts:256-329): these run
        # without a model call
        if task["executor"] in ("keeper_contact_check", "keeper_reminder"):
            return self._run_builtin(task)

        # session continuity: rotate after 20 runs
        session_id = task["session_id"]
        if task["session_continuity"] and task["run_count"] \
                and task["run_count"] % TASK_SESSION_ROTATE_RUNS == 0:
            session_id = None
        if task["session_continuity"] and not session_id:
            session_id = uuid.uuid4().hex

        with self.ldb as db:
            run_id = q.create_task_run(db, task_id, session_id=session_id)
        logs = CycleLogBuffer(self.ldb, run_id, bus=self.bus,
                              room_id=task["room_id"], table="console")
        if self.bus:
            self.bus.emit("runs", "run_started",
                          {"run_id": run_id, "task_id": task_id})

        # prompt augmentation: learned context + memory context (:414-442)
        prompt = task["prompt"]
        if task.get("learned_context"):
            prompt = (f"## Learned methodology (from prior runs)\n"
                      f"{task['learned_context']}\n\n{prompt}")
        if self.memory is not None and task["room_id"]:
            hits = self.memory.recall(task["room_id"], task["name"], limit=3)
            if hits:
                ctx = "\n".join(f"- {h['name']}: {'; '.join(h['observations'][:2])}"
                                for h in hits)
                prompt = f"## Relevant memory\n{ctx}\n\n{prompt}"

        model = self.default_model
        room_id = task["room_id"]
        worker_id = task["worker_id"]
        with self.ldb as db:
            if worker_id:
                w = q.get_worker(db, worker_id)
                if w and w.get("model"):
                    model = w["model"]
            elif room_id:
                room = q.get_room(db, room_id)
                if room:
                    model = room["worker_model"]

        def tool_executor(call: ToolCall) -> str:
            if room_id is None:
                return '{"error": "task has no room"}'
            with self.ldb as db:
                return agent_tools.execute_agent_tool(
                    db, room_id, worker_id or 0, call,
                    memory=self.memory)

        options = AgentExecutionOptions(
            prompt=prompt, model=model,
            system_prompt="You are a task executor. Complete the task and "
                          "report the result.",
            max_turns=task.get("max_turns") or 10,
            session_id=session_id,
            tools=agent_tools.WORKER_TOOLS if room_id else [],
            tool_executor=tool_executor if room_id else None,
            worker_id=worker_id, room_id=room_id,
            on_log=logs.append,
        )

        # rate-limit retry loop (:137-209)
        result = None
        for attempt in range(RATE_LIMIT_MAX_RETRIES + 1):
            result = await asyncio.to_thread(execute_agent, options)
            rl = detect_rate_limit(result.error or
                                   ("" if result.success else result.text))
            if not rl.detected:
                break
            logs.system(f"rate limited; retrying in {rl.wait_ms // 1000}s "
                        f"(attempt {attempt + 1})")
            await abortable_sleep(min(rl.wait_ms, 5_000))  # tests stay fast
            options.session_id = None  # resume-failure retry without session

        duration_ms = int((time.time() - start) * 1000)
        status = "completed" if result.success else "failed"

        # markdown result file (:631-659)
        result_file = None
        if result.success and result.text:
            safe = re.sub(r"[^a-zA-Z0-9_-]+", "-", task["name"])[:48]
            fname = f"{safe}-{int(time.time())}.md"
            path = results_dir() / fname
            try:
                path.write_text(f"# {task['name']}\n\n{result.text}\n")
                result_file = str(path)
            except OSError:
                pass

        with self.ldb as db:
            q.finish_task_run(db, run_id, status,
                              result=(result.text or "")[:10_000],
                              error_message=result.error,
                              result_file=result_file, duration_ms=duration_ms)
            updates = dict(last_run=q.now_iso(),
                           last_result=(result.text or result.error or "")[:1000],
                           run_count=task["run_count"] + 1,
                           session_id=session_id if task["session_continuity"] else None)
            if not result.success:
                updates["error_count"] = task["error_count"] + 1
                # terminal errors auto-pause the task (:609-624)
                if result.error and any(p.search(result.error)
                                        for p in TERMINAL_ERROR_PATTERNS):
                    updates["status"] = "paused"
            new_count = task["run_count"] + 1
            if task["max_runs"] and new_count >= task["max_runs"]:
                updates["status"] = "completed"
            if task["trigger_type"] == "once" and result.success:
                updates["status"] = "completed"
            q.update_task(db, task_id, **updates)

            # learned-context distillation every 3 runs (learned-context.ts:20-31)
            if result.success and should_distill(
                    new_count, has_context=bool(task.get("learned_context"))):
                memo = distill_learned_context(db, task_id, model=model)
                if memo:
                    q.update_task(db, task_id, learned_context=memo[:1500])

        logs.flush()
        if self.bus:
            self.bus.emit("runs", "run_finished",
                          {"run_id": run_id, "task_id": task_id, "status": status})
            self.bus.emit(f"run:{run_id}", "finished", {"status": status})
        return {"run_id": run_id, "status": status, "result": result}

    def _run_builtin(self, task: dict) -> dict:
        """keeper_contact_check: notify the keeper about pending escalations;
        keeper_reminder: deliver the task prompt as a keeper notification."""
        from .notifications import notify_keeper
        t0 = time.time()
        with self.ldb as db:
            run_id = q.create_task_run(db, task["id"])
        result = ""
        if task["executor"] == "keeper_contact_check":
            with self.ldb as db:
                pending = (q.list_escalations(db, task["room_id"], status="pending")
                           if task["room_id"] else [])
            if pending:
                notify_keeper(f"{len(pending)} pending escalation(s)",
                              "\n".join(e["question"][:200] for e in pending[:10]),
                              room_id=task["room_id"], bus=self.bus)
            result = f"checked: {len(pending)} pending"
        else:  # keeper_reminder
            notify_keeper(f"Reminder: {task['name']}", task["prompt"],
                          room_id=task["room_id"], bus=self.bus)
            result = "reminder delivered"
        dur = int((time.time() - t0) * 1000)
        with self.ldb as db:
            q.finish_task_run(db, run_id, "completed", result=result,
                              duration_ms=dur)
            q.update_task(db, task["id"], last_run=q.now_iso(),
                          last_result=result, run_count=task["run_count"] + 1)
        return {"run_id": run_id, "status": "completed", "result": result}

    def cancel_running_tasks_for_room(self, room_id: int) -> int:
        n = 0
        with self.ldb as db:
            for t in q.list_tasks(db, room_id=room_id):
                if t["id"] in self.running_tasks:
                    n += 1  # cooperative: runs finish their current turn
        return n
