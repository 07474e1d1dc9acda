"""Process supervisor: registry + process-tree termination.

Reference semantics: src/shared/process-supervisor.ts (registry of spawned
children, kill whole trees SIGTERM→grace→SIGKILL on shutdown).
"""
import os
import signal
import subprocess
import time

import pytest

from room_amd.core import process_supervisor as ps


def _spawn_tree():
    """Parent sh that spawns a child sleep; returns (proc, wait_for_child)."""
    proc = subprocess.Popen(["sh", "-c", "sleep 30 & wait"])
    deadline = time.time() + 5
    while time.time() < deadline:
        if len(ps.process_tree(proc.pid)) >= 2:
            break
        time.sleep(0.05)
    return proc


def _alive(pid):
    try:
        os.kill(pid, 0)
        return True
    except ProcessLookupError:
        return False


def test_register_unregister():
    ps.register_managed_process(12345, "x")
    assert 12345 in ps.managed_pids()
    ps.unregister_managed_process(12345)
    assert 12345 not in ps.managed_pids()


def test_process_tree_includes_children():
    proc = _spawn_tree()
    try:
        tree = ps.process_tree(proc.pid)
        assert proc.pid in tree
        assert len(tree) >= 2, tree  # sh + sleep
        assert tree[0] == proc.pid   # parent first
    finally:
        ps.terminate_tree(proc.pid, grace=1.0)
        proc.wait(timeout=5)


def test_terminate_tree_kills_descendants():
    proc = _spawn_tree()
    tree = ps.process_tree(proc.pid)
    assert len(tree) >= 2
    ps.terminate_tree(proc.pid, grace=2.0)
    proc.wait(timeout=5)
    time.sleep(0.1)
    for pid in tree:
        assert not _alive(pid), f"pid {pid} survived"


def test_terminate_tree_sigkills_term_ignorers():
    # child that traps/ignores SIGTERM must still die via SIGKILL
    proc = subprocess.Popen(["sh", "-c", "trap '' TERM; sleep 30"])
    time.sleep(0.2)
    killed = ps.terminate_tree(proc.pid, grace=0.5)
    proc.wait(timeout=5)
    assert not _alive(proc.pid)
    assert proc.pid in killed  # needed the SIGKILL path


def test_terminate_managed_processes_drains_registry():
    proc = _spawn_tree()
    ps.register_managed_process(proc.pid, "t")
    out = ps.terminate_managed_processes(grace=2.0)
    proc.wait(timeout=5)
    assert proc.pid in out
    assert proc.pid not in ps.managed_pids()
    assert not _alive(proc.pid)


def test_terminate_tree_on_dead_pid_is_noop():
    proc = subprocess.Popen(["true"])
    proc.wait()
    assert ps.terminate_tree(proc.pid, grace=0.1) == []


def test_runtime_stop_reaps_managed_children(tmp_path):
    """ServerRuntime.stop() must terminate registered child trees
    (reference index.ts:974-1005 graceful shutdown → process supervisor)."""
    import asyncio

    from room_amd.core.agent_loop import AgentLoopManager
    from room_amd.core.tasks import TaskRunner
    from room_amd.db import LockedDb, connect
    from room_amd.server.runtime import ServerRuntime

    ldb = LockedDb(connect(str(tmp_path / "t.db")))
    proc = subprocess.Popen(["sleep", "30"])
    ps.register_managed_process(proc.pid, "test-child")
    rt = ServerRuntime(ldb, TaskRunner(ldb), loop_mgr=AgentLoopManager(ldb))

    async def go():
        await rt.start()
        await asyncio.sleep(0.1)
        await rt.stop()

    asyncio.run(go())
    proc.wait(timeout=5)
    assert not _alive(proc.pid)
    assert proc.pid not in ps.managed_pids()
