"""Managed child-process registry with process-tree termination.

Reference: src/shared/process-supervisor.ts:1-117 — every spawned child is
registered; on shutdown (or room stop) the supervisor kills the whole
process TREE of each registered child (children may have forked their own
helpers, e.g. Chromium's renderer/zygote processes), SIGTERM first, then a
grace period, then SIGKILL for survivors.

Linux-only here (the deploy target): the tree is discovered by walking
/proc/<pid>/task/<tid>/children (complete and race-free enough for
teardown; falls back to a full /proc scan of PPids when the children file
is unavailable).
"""
from __future__ import annotations

import os
import signal
import threading
import time

_lock = threading.Lock()
_managed: dict[int, str] = {}   # pid -> label


def register_managed_process(pid: int, label: str = "") -> None:
    """Track a spawned child so shutdown can reap its whole tree."""
    with _lock:
        _managed[pid] = label


def unregister_managed_process(pid: int) -> None:
    with _lock:
        _managed.pop(pid, None)


def managed_pids() -> dict[int, str]:
    with _lock:
        return dict(_managed)


def _children_of(pid: int) -> list[int]:
    """Direct children via /proc/<pid>/task/*/children."""
    out: list[int] = []
    task_dir = f"/proc/{pid}/task"
    try:
        for tid in os.listdir(task_dir):
            try:
                with open(f"{task_dir}/{tid}/children") as f:
                    out.extend(int(c) for c in f.read().split())
            except (OSError, ValueError):
                continue
    except OSError:
        # /proc children file unavailable: scan PPid fields
        try:
            for entry in os.listdir("/proc"):
                if not entry.isdigit():
                    continue
                try:
                    with open(f"/proc/{entry}/status") as f:
                        for line in f:
                            if line.startswith("PPid:"):
                                if int(line.split()[1]) == pid:
                                    out.append(int(entry))
                                break
                except (OSError, ValueError):
                    continue
        except OSError:
            pass
    return out


def process_tree(pid: int) -> list[int]:
    """pid + all descendants, parents before children."""
    seen, order, stack = set(), [], [pid]
    while stack:
        p = stack.pop(0)
        if p in seen:
            continue
        seen.add(p)
        order.append(p)
        stack.extend(_children_of(p))
    return order


def _alive(pid: int) -> bool:
    try:
        os.kill(pid, 0)
        return True
    except ProcessLookupError:
        return False
    except PermissionError:
        return True


def terminate_tree(pid: int, grace: float = 3.0) -> list[int]:
    """SIGTERM the whole tree at once, wait up to `grace` seconds, then
    SIGKILL survivors (reference semantics: signal the full list, then
    force-kill what remains). Returns pids that needed SIGKILL."""
    tree = process_tree(pid)
    for p in tree:
        try:
            os.kill(p, signal.SIGTERM)
        except OSError:
            pass
    deadline = time.time() + grace
    while time.time() < deadline:
        if not any(_alive(p) for p in tree):
            break
        time.sleep(0.05)
    killed = []
    for p in tree:
        if _alive(p):
            try:
                os.kill(p, signal.SIGKILL)
                killed.append(p)
            except OSError:
                pass
    return killed


def terminate_managed_processes(grace: float = 3.0) -> dict[int, list[int]]:
    """Terminate every registered child's tree; returns {pid: force-killed}.
    Reference: process-supervisor.ts:105-117 (shutdown path)."""
    result = {}
    for pid in list(managed_pids()):
        result[pid] = terminate_tree(pid, grace=grace)
        unregister_managed_process(pid)
    return result
