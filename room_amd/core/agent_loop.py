"""Per-worker agent loop: observe → prompt → execute → persist.

Semantics preserved from the reference (src/shared/agent-loop.ts):
- runningLoops registry, pause/resume/trigger with abortable waits (:89,266-287)
- quiet hours (:30-51,138-155)
- adaptive 10s "momentum" gap when WIP exists (:204-217)
- stuck detector (:605-617)
- queen auto-creates an executor worker when she has none (:414-449)
- session rotation after 20 cycles / compression at 30 msgs (:462-532)
- cycle prompt parts 1-8 (:534-685)
- rate-limit backoff via typed error (:166-190)
- auto-WIP fallback + prune on persist (:837-867)

Control inversion for MI355X: cycles stay per-agent and synchronous in shape,
but every engine.chat() call funnels into the shared GPU scheduler which
batches concurrent agents' prefill/decode into fused kernels
(room_amd.engine.llm). Observable per-agent behavior is unchanged.
"""
from __future__ import annotations

import os
import asyncio
import json
import time
from dataclasses import dataclass, field
from datetime import datetime
from typing import Callable, Optional

from ..db import LockedDb
from ..db import queries as q
from ..engine.providers import compress_session, execute_agent
from ..engine.types import AgentExecutionOptions, ToolCall
from . import agent_tools, quorum
from .constants import (MEMORY_TOP_K, MIN_CYCLE_GAP_MS, MOMENTUM_GAP_MS,
                        SESSION_COMPRESS_AT_MSGS, SESSION_ROTATE_CYCLES)
from .events import EventBus
from .log_buffer import CycleLogBuffer
from .rate_limit import RateLimitError, abortable_sleep, detect_rate_limit
from .skills import load_skills_for_agent

# Model-B (soft) queen policy (reference agent-loop.ts:22-28): the queen is
# control-plane; direct web/browser execution is a logged deviation
QUEEN_EXECUTION_TOOLS = {"room_web_search", "room_web_fetch", "room_browser"}
QUEEN_POLICY_WIP_HINT = (
    "[policy] Queen control-plane mode: delegate execution tasks to workers "
    "with room_delegate_task, then monitor, unblock, and report outcomes. "
    "Avoid direct web/browser execution.")


@dataclass
class LoopState:
    worker_id: int
    room_id: int
    running: bool = True
    paused: bool = False
    cycle_count: int = 0
    consecutive_errors: int = 0
    wake_event: asyncio.Event = field(default_factory=asyncio.Event)
    task: Optional[asyncio.Task] = None
    last_summaries: list[str] = field(default_factory=list)  # stuck detector window


class AgentLoopManager:
    """Owns all agent loops in this process (one room shard / one GPU)."""

    def __init__(self, ldb: LockedDb, bus: EventBus | None = None,
                 embed_fn: Callable[[str], list[float]] | None = None,
                 memory=None,
                 time_source: Callable[[], float] = time.time):
        self.ldb = ldb
        self.bus = bus or EventBus()
        self.memory = memory  # MemoryService (GPU vector store) when available
        self.embed_fn = embed_fn or (memory.embed if memory is not None else None)
        self.running_loops: dict[int, LoopState] = {}
        self.time = time_source
        agent_tools.register_wake_callbacks(self.trigger_agent, self.wake_room_workers)

    # ------------------------------------------------------------- wakes

    def trigger_agent(self, worker_id: int) -> None:
        state = self.running_loops.get(worker_id)
        if state:
            state.wake_event.set()

    def wake_room_workers(self, room_id: int, exclude: int | None = None) -> None:
        for wid, state in self.running_loops.items():
            if state.room_id == room_id and wid != exclude:
                state.wake_event.set()

    def pause_agent(self, worker_id: int) -> None:
        state = self.running_loops.get(worker_id)
        if state:
            state.paused = True

    def resume_agent(self, worker_id: int) -> None:
        state = self.running_loops.get(worker_id)
        if state:
            state.paused = False
            state.wake_event.set()

    def stop_agent(self, worker_id: int) -> None:
        state = self.running_loops.pop(worker_id, None)
        if state:
            state.running = False
            state.wake_event.set()

    def stop_all(self) -> None:
        for wid in list(self.running_loops):
            self.stop_agent(wid)

    # ------------------------------------------------------------- loop

    async def start_agent_loop(self, room_id: int, worker_id: int) -> LoopState:
        if worker_id in self.running_loops:  # idempotent guard (:112-113)
            return self.running_loops[worker_id]
        state = LoopState(worker_id=worker_id, room_id=room_id)
        self.running_loops[worker_id] = state
        state.task = asyncio.create_task(self._loop(state))
        return state

    async def _loop(self, state: LoopState) -> None:
        while state.running:
            try:
                with self.ldb as db:
                    q.ensure_worker_room_mapping(db, state.worker_id, state.room_id)
                    room = q.get_room(db, state.room_id)
                    worker = q.get_worker(db, state.worker_id)
                if room is None or worker is None or room["status"] != "active" \
                        or state.paused:
                    state.wake_event.clear()
                    await abortable_sleep(2000, state.wake_event)
                    continue
                if self._in_quiet_hours(room):
                    state.wake_event.clear()
                    await abortable_sleep(60_000, state.wake_event)
                    continue

                await self.run_cycle(state.room_id, state.worker_id, state=state)
                state.consecutive_errors = 0
                gap = self._cycle_gap_ms(room, worker)
            except RateLimitError as e:
                with self.ldb as db:
                    q.set_worker_state(db, state.worker_id, "rate_limited")
                gap = e.wait_ms
            except Exception as e:
                state.consecutive_errors += 1
                with self.ldb as db:
                    q.log_room_activity(db, state.room_id, "error",
                                        f"Cycle error: {e}", actor_id=state.worker_id)
                gap = min(60_000 * state.consecutive_errors, 300_000)

            state.cycle_count += 1
            state.wake_event.clear()
            await abortable_sleep(gap, state.wake_event)

    def _in_quiet_hours(self, room: dict) -> bool:
        qf, qu = room.get("queen_quiet_from"), room.get("queen_quiet_until")
        if not qf or not qu:
            return False
        now = datetime.now().strftime("%H:%M")
        if qf <= qu:
            return qf <= now < qu
        return now >= qf or now < qu  # crosses midnight

    def _cycle_gap_ms(self, room: dict, worker: dict) -> int:
        if worker.get("wip"):
            return MOMENTUM_GAP_MS  # momentum: WIP exists, keep moving
        if worker["id"] == room.get("queen_worker_id"):
            gap = room.get("queen_cycle_gap_ms") or 1_800_000
        else:
            gap = worker.get("cycle_gap_ms") or 60_000
        floor = room["config"].get("minCycleGapMs", MIN_CYCLE_GAP_MS)
        return max(int(gap), int(floor))

    # ------------------------------------------------------------- cycle

    async def run_cycle(self, room_id: int, worker_id: int,
                        state: LoopState | None = None,
                        max_turns: int | None = None,
                        max_new_tokens: int | None = None) -> dict:
        """One full observe→prompt→execute→persist cycle. Async so many agents
        interleave; the blocking engine call runs in a worker thread and the
        GPU scheduler batches across agents."""
        start = time.time()
        with self.ldb as db:
            quorum.check_expired_decisions(db)
            room = q.get_room(db, room_id)
            worker = q.get_worker(db, worker_id)
            if room is None or worker is None:
                raise ValueError("room/worker not found")
            is_queen = worker_id == room.get("queen_worker_id")

            # queen auto-creates an executor when she has no workers (:414-449)
            if is_queen:
                others = [w for w in q.list_room_workers(db, room_id)
                          if w["id"] != worker_id]
                if not others:
                    from .constants import WORKER_ROLE_PRESETS
                    preset = WORKER_ROLE_PRESETS["executor"]
                    q.create_worker(db, f"Executor of {room['name']}",
                                    preset["systemPromptPrefix"], role="executor",
                                    room_id=room_id,
                                    cycle_gap_ms=preset["cycleGapMs"],
                                    max_turns=preset["maxTurns"])
                    q.log_room_activity(db, room_id, "worker",
                                        "Auto-created executor worker",
                                        actor_id=worker_id)

            cycle_id = q.create_worker_cycle(db, worker_id, room_id,
                                             model=worker.get("model")
                                             or room["worker_model"])
            q.set_worker_state(db, worker_id, "thinking")
            prompt = self._build_cycle_prompt(db, room, worker, is_queen, state)
            session = q.get_agent_session(db, worker_id)

        log_buffer = CycleLogBuffer(self.ldb, cycle_id, bus=self.bus,
                                    room_id=room_id)
        self.bus.emit(f"room:{room_id}", "cycle_started",
                      {"cycle_id": cycle_id, "worker_id": worker_id})

        # session continuity: rotate after N cycles, compress long histories
        messages = None
        turn_count = 0
        model = worker.get("model") or room["worker_model"]
        if session and session["model"] == model:
            turn_count = session["turn_count"] or 0
            if turn_count < SESSION_ROTATE_CYCLES and session["messages_json"]:
                try:
                    messages = json.loads(session["messages_json"])
                except (ValueError, TypeError):
                    messages = None
                if messages and len(messages) >= SESSION_COMPRESS_AT_MSGS:
                    messages = compress_session(messages, model=model)
            elif turn_count >= SESSION_ROTATE_CYCLES:
                turn_count = 0  # rotation: fresh session

        tools = agent_tools.tools_for_role("queen" if is_queen else worker.get("role"))

        execution_tools_used: set[str] = set()

        def tool_executor(call: ToolCall) -> str:
            if is_queen and call.name in QUEEN_EXECUTION_TOOLS:
                execution_tools_used.add(call.name)
            with self.ldb as db:
                return agent_tools.execute_agent_tool(db, room_id, worker_id, call,
                                                      embed_fn=self.embed_fn,
                                                      memory=self.memory)

        options = AgentExecutionOptions(
            prompt=prompt, model=model,
            system_prompt=worker["system_prompt"],
            max_turns=max_turns or (room["queen_max_turns"] if is_queen
                                    else worker.get("max_turns") or 10),
            messages=messages, tools=tools, tool_executor=tool_executor,
            worker_id=worker_id, room_id=room_id,
            on_log=log_buffer.append,
        )
        if max_new_tokens is not None:
            options.max_new_tokens = max_new_tokens
        if os.environ.get("ROOMAMD_CYCLE_PROF") == "1":
            import sys as _sys
            print(f"[cycleprof] w{worker_id} prep {1000*(time.time()-start):.0f}"
                  f" ms before execute (t={time.time()%100:.3f})",
                  file=_sys.stderr)
        result = await asyncio.to_thread(execute_agent, options)

        # context-overflow retry with fresh session (:773-782)
        if not result.success and result.error and "context" in result.error.lower():
            options.messages = None
            result = await asyncio.to_thread(execute_agent, options)

        rl = detect_rate_limit(result.error or ("" if result.success else result.text))
        duration_ms = int((time.time() - start) * 1000)

        with self.ldb as db:
            if rl.detected:
                q.complete_worker_cycle(db, cycle_id, "failed",
                                        error_message="rate limited",
                                        duration_ms=duration_ms)
                log_buffer.flush()
                raise RateLimitError(result.error or "rate limited", rl.wait_ms)

            status = "completed" if result.success else "failed"
            q.complete_worker_cycle(db, cycle_id, status,
                                    error_message=result.error,
                                    duration_ms=duration_ms,
                                    input_tokens=result.input_tokens,
                                    output_tokens=result.output_tokens)
            q.set_worker_state(db, worker_id, "idle")
            summary = (result.text or result.error or "")[:200]
            q.log_room_activity(db, room_id, "cycle",
                                f"{worker['name']}: {summary}",
                                actor_id=worker_id, is_public=False)
            # Model-B queen policy deviation (agent-loop.ts:707-728): log it
            # and pin the control-plane hint into WIP
            if is_queen and execution_tools_used:
                used = ", ".join(sorted(execution_tools_used))
                q.log_room_activity(
                    db, room_id, "system",
                    f"Queen policy deviation: execution tool use detected "
                    f"({used}).", actor_id=worker_id)
                fresh = q.get_worker(db, worker_id)
                existing = ((fresh or {}).get("wip") or "").strip()
                if QUEEN_POLICY_WIP_HINT not in existing:
                    nxt = (f"{existing}\n\n{QUEEN_POLICY_WIP_HINT}"
                           if existing else QUEEN_POLICY_WIP_HINT)
                    q.set_worker_wip(db, worker_id, nxt[:2000])
            # auto-WIP fallback: if the agent didn't save WIP and produced text,
            # keep a trace so the next cycle continues (:854-863)
            w = q.get_worker(db, worker_id)
            if result.success and result.text and not (w and w.get("wip")) \
                    and result.tool_calls_executed == 0:
                q.set_worker_wip(db, worker_id, f"[auto] {result.text[:500]}")
            # persist session; trim at MESSAGE granularity so the stored
            # blob is always valid JSON (a char-level cut would make the next
            # cycle's json.loads fail and silently reset the session)
            msgs = result.messages or []
            blob = json.dumps(msgs)
            while len(blob) > 200_000 and len(msgs) > 2:
                msgs = msgs[2:]            # drop the oldest exchange
                blob = json.dumps(msgs)
            q.save_agent_session(
                db, worker_id, session_id=result.session_id,
                messages_json=blob, model=model, turn_count=turn_count + 1)
            q.prune_old_cycles(db, room_id)

        # stuck detector: same summary 3 cycles in a row (:605-617)
        if state is not None:
            state.last_summaries.append(summary)
            state.last_summaries = state.last_summaries[-3:]

        log_buffer.flush()
        self.bus.emit(f"room:{room_id}", "cycle_finished",
                      {"cycle_id": cycle_id, "worker_id": worker_id,
                       "status": "completed" if result.success else "failed",
                       "duration_ms": duration_ms})
        return {"cycle_id": cycle_id, "result": result, "duration_ms": duration_ms}

    def is_stuck(self, state: LoopState) -> bool:
        return (len(state.last_summaries) == 3
                and len(set(state.last_summaries)) == 1
                and bool(state.last_summaries[0]))

    # ------------------------------------------------------------- prompt

    def _build_cycle_prompt(self, db, room: dict, worker: dict, is_queen: bool,
                            state: LoopState | None) -> str:
        """Context parts 1-8 (reference agent-loop.ts:534-685)."""
        parts: list[str] = []
        role = "Queen (control plane — coordinate, do not execute)" if is_queen \
            else f"{worker.get('role') or 'worker'}"
        parts.append(f"You are {worker['name']} ({role}) in room "
                     f"'{room['name']}' (#{room['id']}). Worker id: {worker['id']}.")

        if worker.get("wip"):
            parts.append("## CONTINUE FORWARD\nYour saved work-in-progress:\n"
                         f"{worker['wip']}\nContinue from here; update or clear it "
                         "with room_save_wip.")

        goals = q.list_room_goals(db, room["id"])
        active = [g for g in goals if g["status"] in ("active", "in_progress")]
        lines = [f"OBJECTIVE: {room.get('goal') or '(none set)'}"]
        subgoals = [g for g in active if g["parent_goal_id"] is not None]
        if not subgoals and is_queen:
            lines.append("No subgoals yet — decompose the objective with room_set_goal.")
        for g in active[:15]:
            assignee = f" → worker #{g['assigned_worker_id']}" if g["assigned_worker_id"] else ""
            lines.append(f"- goal #{g['id']} [{g['status']}, "
                         f"{int((g['progress'] or 0) * 100)}%]: {g['description']}{assignee}")
        mine = [g for g in active if g["assigned_worker_id"] == worker["id"]]
        if mine:
            lines.append("Assigned to YOU: " + ", ".join(f"#{g['id']}" for g in mine))
        parts.append("## Objective & goals\n" + "\n".join(lines))

        # top-5 room memory via hybrid search on the objective context
        query = f"{room.get('goal') or room['name']}"
        vec = self.embed_fn(query) if self.embed_fn else None
        semantic = (self.memory.store.search(vec, k=20)
                    if (self.memory is not None and vec is not None) else None)
        hits = q.hybrid_search(db, query, vec, limit=MEMORY_TOP_K,
                               room_id=room["id"], semantic_hits=semantic)
        if hits:
            mem_lines = [f"- {h['name']}: {'; '.join(h['observations'][:2])}"
                         for h in hits]
            parts.append("## Relevant room memory\n" + "\n".join(mem_lines))

        skills_block, _ = load_skills_for_agent(db, room["id"],
                                                f"{query} {worker.get('role') or ''}")
        if skills_block:
            parts.append(skills_block)

        if state is not None and self.is_stuck(state):
            parts.append("## WARNING\nYour last 3 cycles produced identical "
                         "output. Change approach or escalate to the keeper.")

        # housekeeping: announced decisions / keeper answers / roster / messages
        housekeeping = []
        announced = q.list_room_decisions(db, room["id"], status="announced")
        for d in announced[:5]:
            housekeeping.append(f"- Announced decision #{d['id']}: {d['proposal']}"
                                f" (object with room_object before {d['effective_at']})")
        for e in q.get_pending_keeper_answers(db, room["id"])[:5]:
            housekeeping.append(f"- Keeper answered: Q: {e['question']} A: {e['answer']}")
            q.mark_escalation_consumed(db, e["id"])
        if is_queen:
            roster = q.list_room_workers(db, room["id"])
            housekeeping.append("Workers: " + ", ".join(
                f"#{w['id']} {w['name']} ({w.get('role')}, {w['agent_state']})"
                for w in roster))
            for esc in q.list_escalations(db, room["id"], status="pending")[:5]:
                housekeeping.append(f"- Pending escalation #{esc['id']}: {esc['question']}")
        if housekeeping:
            parts.append("## Housekeeping\n" + "\n".join(housekeeping))

        unread = q.get_unread_room_messages(db, room["id"])
        if unread:
            parts.append("## Unread inter-room messages\n" + "\n".join(
                f"- from {m['from_room_id']}: {m['subject']}: {m['body'][:200]}"
                for m in unread[:5]))
            for m in unread:
                q.mark_room_message_read(db, m["id"])

        parts.append("## Instructions\nTake the most useful next action(s) via "
                     "tool calls. Save WIP before finishing if work remains.")
        return "\n\n".join(parts)
