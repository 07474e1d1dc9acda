"""Server runtime loops (reference: src/server/runtime.ts).

asyncio tasks: cron-triggered task execution (15s refresh), due-once tasks,
maintenance (stale-run cleanup, 60s), queen inbox poll (2.5s) that wakes a
queen when pending escalations appear, and background embedding indexing.
"""
from __future__ import annotations

import asyncio
from datetime import datetime

from ..core.cron import CronExpression
from ..core.tasks import TaskRunner
from ..db import LockedDb
from ..db import queries as q

CRON_REFRESH_S = float(__import__("os").environ.get("ROOMAMD_CRON_REFRESH_S", "15"))
MAINTENANCE_S = 60.0
INBOX_POLL_S = 2.5
INDEXER_S = 10.0
WATCHER_S = 5.0


class ServerRuntime:
    def __init__(self, ldb: LockedDb, runner: TaskRunner, loop_mgr=None,
                 memory=None, bus=None, commentary_model: str | None = None):
        self.ldb = ldb
        self.runner = runner
        self.loop_mgr = loop_mgr
        self.memory = memory
        self.bus = bus
        self.commentary = None
        if bus is not None and commentary_model:
            from ..core.clerk import CommentaryEngine
            self.commentary = CommentaryEngine(ldb, bus, model=commentary_model)
        self._tasks: list[asyncio.Task] = []
        self._stop = asyncio.Event()
        self._last_cron_minute: dict[int, str] = {}
        # fire-and-forget executions need a strong ref: asyncio holds only
        # weak refs to tasks, so an untracked create_task can be GC'd mid-run
        self._bg: set[asyncio.Task] = set()

    def _spawn(self, coro) -> None:
        t = asyncio.create_task(coro)
        self._bg.add(t)
        t.add_done_callback(self._bg.discard)

    async def start(self) -> None:
        with self.ldb as db:
            n = q.cleanup_stale_cycles(db)
            n += q.cleanup_all_running_runs(db)
        self._tasks = [
            asyncio.create_task(self._cron_loop()),
            asyncio.create_task(self._maintenance_loop()),
            asyncio.create_task(self._inbox_loop()),
            asyncio.create_task(self._indexer_loop()),
        ]
        if self.commentary is not None:
            self._tasks.append(asyncio.create_task(self._commentary_loop()))
        # cloud activity push (reference cloud.ts): inert without
        # ROOMAMD_CLOUD_API, 1 push/s/room rate limit, fail-silent
        if self.bus is not None:
            from ..core.cloud_sync import ActivityPusher
            self._activity_pusher = ActivityPusher(self.bus, self.ldb)
        self._tasks.append(asyncio.create_task(self._alert_relay_loop()))
        self._tasks.append(asyncio.create_task(self._watcher_loop()))
        self._watch_mtimes: dict[int, float] = {}

    async def stop(self) -> None:
        self._stop.set()
        for t in self._tasks:
            t.cancel()
        for t in self._tasks:
            try:
                await t
            except (asyncio.CancelledError, Exception):
                pass
        self._tasks = []
        if getattr(self, "_activity_pusher", None) is not None:
            self._activity_pusher.stop()
            self._activity_pusher = None
        # reap any still-running managed children (browser trees etc.) —
        # reference index.ts:974-1005 graceful shutdown → process supervisor
        from ..core.browser import close_all_sessions
        from ..core.process_supervisor import terminate_managed_processes
        try:
            close_all_sessions()
        except Exception:
            pass
        terminate_managed_processes()

    async def _sleep(self, s: float) -> bool:
        try:
            await asyncio.wait_for(self._stop.wait(), timeout=s)
            return True
        except asyncio.TimeoutError:
            return False

    # --- cron + due-once tasks
    async def _cron_loop(self) -> None:
        while not self._stop.is_set():
            try:
                now = datetime.now()
                minute_key = now.strftime("%Y-%m-%d %H:%M")
                with self.ldb as db:
                    crons = [t for t in q.list_tasks(db, status="active")
                             if t["trigger_type"] == "cron" and t["cron_expression"]]
                    due_once = q.get_due_once_tasks(db)
                for t in crons:
                    if self._last_cron_minute.get(t["id"]) == minute_key:
                        continue
                    try:
                        if CronExpression(t["cron_expression"]).matches(now):
                            self._last_cron_minute[t["id"]] = minute_key
                            self._spawn(self.runner.execute_task(t["id"]))
                    except ValueError:
                        pass
                for t in due_once:
                    if not self.runner.is_task_running(t["id"]):
                        self._spawn(self.runner.execute_task(t["id"]))
            except Exception:
                pass
            if await self._sleep(CRON_REFRESH_S):
                return

    # --- maintenance
    async def _maintenance_loop(self) -> None:
        from ..core.update_checker import UpdateChecker
        self.update_checker = UpdateChecker("0.1.0")
        self.update_checker.boot_health_check()
        while not self._stop.is_set():
            try:
                with self.ldb as db:
                    q.cleanup_stale_runs(db)
                # release poll on the updateChecker.ts 4h cadence (offline →
                # state recorded, never raises)
                await asyncio.to_thread(self.update_checker.maybe_check)
                # project docs → clerk memory (hash-gated, 60s min interval)
                from ..core.clerk import sync_project_docs
                await asyncio.to_thread(sync_project_docs, self.ldb)
            except Exception:
                pass
            if await self._sleep(MAINTENANCE_S):
                return

    # --- queen inbox poll: wake queens with pending escalations/messages
    async def _inbox_loop(self) -> None:
        while not self._stop.is_set():
            try:
                if self.loop_mgr is not None:
                    with self.ldb as db:
                        rooms = q.list_rooms(db)
                        for room in rooms:
                            if room["status"] != "active" or not room["queen_worker_id"]:
                                continue
                            pend = q.list_escalations(db, room["id"], status="answered")
                            unread = q.get_unread_room_messages(db, room["id"])
                            if pend or unread:
                                self.loop_mgr.trigger_agent(room["queen_worker_id"])
            except Exception:
                pass
            if await self._sleep(INBOX_POLL_S):
                return

    # --- clerk alert relay: pending keeper-facing escalations ride the
    # notification outbox (email/telegram when configured) on a 15 s poll
    # (reference runtime.ts clerk alert relay + clerk-notifications.ts)
    async def _alert_relay_loop(self) -> None:
        from ..core.notifications import notify_keeper
        self._alerted: set[int] = getattr(self, "_alerted", set())
        while not self._stop.is_set():
            try:
                with self.ldb as db:
                    rooms = q.list_rooms(db)
                    pend = []
                    for room in rooms:
                        for e in q.list_escalations(db, room["id"],
                                                    status="pending"):
                            if e["to_agent_id"] is None and \
                                    e["id"] not in self._alerted:
                                pend.append((room, e))
                for room, e in pend:
                    self._alerted.add(e["id"])
                    await asyncio.to_thread(
                        notify_keeper,
                        f"[{room['name']}] escalation #{e['id']}",
                        e["question"], room_id=room["id"], bus=self.bus)
            except Exception:
                pass
            if await self._sleep(15.0):
                return

    # --- clerk commentary narration
    async def _commentary_loop(self) -> None:
        while not self._stop.is_set():
            try:
                await asyncio.to_thread(self.commentary.tick)
            except Exception:
                pass
            if await self._sleep(self.commentary.pace_s):
                return

    # --- file watchers: mtime changes trigger a queen escalation (reference
    # watcher tool semantics: watched path + action prompt)
    async def _watcher_loop(self) -> None:
        import os as _os
        self._watch_mtimes = getattr(self, "_watch_mtimes", {})
        while not self._stop.is_set():
            try:
                with self.ldb as db:
                    watches = [w for w in q.list_watches(db)
                               if w["status"] == "active"]
                for w in watches:
                    try:
                        mtime = _os.path.getmtime(_os.path.expanduser(w["path"]))
                    except OSError:
                        continue
                    prev = self._watch_mtimes.get(w["id"])
                    self._watch_mtimes[w["id"]] = mtime
                    if prev is not None and mtime > prev and w["room_id"]:
                        with self.ldb as db:
                            q.create_escalation(
                                db, w["room_id"],
                                (w["action_prompt"] or "Watched path changed")
                                + f" (path: {w['path']})")
                            db.execute(
                                "UPDATE watches SET last_triggered = ?,"
                                " trigger_count = trigger_count + 1 WHERE id = ?",
                                (q.now_iso(), w["id"]))
                        if self.loop_mgr is not None:
                            with self.ldb as db:
                                room = q.get_room(db, w["room_id"])
                            if room and room.get("queen_worker_id"):
                                self.loop_mgr.trigger_agent(room["queen_worker_id"])
            except Exception:
                pass
            if await self._sleep(WATCHER_S):
                return

    # --- background embedding indexer
    async def _indexer_loop(self) -> None:
        while not self._stop.is_set():
            try:
                if self.memory is not None:
                    await asyncio.to_thread(self.memory.index_pending)
            except Exception:
                pass
            if await self._sleep(INDEXER_S):
                return
