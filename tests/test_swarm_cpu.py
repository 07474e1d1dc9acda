"""Multi-process swarm-collective tests on CPU (gloo, world_size 2) — the
distributed path the driver scales to 8 GPUs must be correct by construction."""
import os

import pytest
import torch
import torch.multiprocessing as mp

from room_amd.parallel.swarm import SwarmContext


def _find_free_port() -> int:
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker(rank: int, world: int, port: int, fail_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        os.environ["LOCAL_RANK"] = str(rank)
        ctx = SwarmContext.from_env(device=torch.device("cpu"))
        assert ctx.is_distributed

        # quorum all-gather: rank 0 votes [1, 1], rank 1 votes [-1, 0]
        votes = torch.tensor([1, 1] if rank == 0 else [-1, 0], dtype=torch.int8)
        tally = ctx.quorum_tally(votes)
        assert tally == {"yes": 2, "no": 1, "abstain": 1, "total": 4}, tally

        # goal broadcast from queen rank
        obj = {"goal": "ship it", "skills": ["deploy"]} if rank == 0 else None
        got = ctx.broadcast_blob(obj, src=0)
        assert got == {"goal": "ship it", "skills": ["deploy"]}, got

        # memory top-k merge with globally-offset ids
        local_v = torch.tensor([0.9, 0.5] if rank == 0 else [0.8, 0.7])
        local_i = torch.tensor([0, 1] if rank == 0 else [100, 101])
        v, i = ctx.topk_merge(local_v, local_i, k=3)
        assert i.tolist() == [0, 100, 101], i.tolist()

        ctx.barrier()
        torch.distributed.destroy_process_group()
    except Exception as e:  # propagate to parent
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


def test_swarm_collectives_world2():
    port = _find_free_port()
    fail_q = mp.get_context("spawn").SimpleQueue()
    mp.spawn(_worker, args=(2, port, fail_q), nprocs=2, join=True)
    assert fail_q.empty()


def test_single_rank_fallbacks():
    ctx = SwarmContext(0, 1, torch.device("cpu"))
    votes = torch.tensor([1, -1, 0], dtype=torch.int8)
    t = ctx.quorum_tally(votes)
    assert t == {"yes": 1, "no": 1, "abstain": 1, "total": 3}
    assert ctx.broadcast_blob({"a": 1}) == {"a": 1}
    v, i = ctx.topk_merge(torch.tensor([0.1, 0.9]), torch.tensor([5, 6]), k=1)
    assert i.tolist() == [6]


def test_bench_torchrun_world2_cpu():
    """The driver's exact multi-GPU launch shape (torch.distributed.run,
    nproc 2, 127.0.0.1) against bench.py on CPU/gloo + StubEngine: the rank-0
    JSON contract line must come out well-formed with n_gpus=2."""
    import json
    import subprocess
    import sys
    from pathlib import Path

    repo = Path(__file__).resolve().parent.parent
    proc = None
    for attempt in range(3):  # master-port races with concurrent tests
        port = _find_free_port()
        proc = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
             "--master-port", str(port), str(repo / "bench.py"),
             "--gpus", "2", "--steps", "2", "--warmup", "1"],
            capture_output=True, text=True, timeout=240, cwd=repo)
        if proc.returncode == 0:
            break
    assert proc.returncode == 0, proc.stderr[-2000:]
    line = next(l for l in proc.stdout.splitlines()
                if l.startswith("{") and '"metric"' in l)
    out = json.loads(line)
    assert out["metric"] == "agent-cycles/sec"
    assert out["n_gpus"] == 2
    assert out["steps"] == 2
    assert out["value"] > 0
    assert out["scaling"] == "weak"
    assert out["config"]["agents_per_gpu"] == 5


def test_bench_torchrun_world8_cpu():
    """8-rank shape the driver uses on a full node (VERDICT r01 #2 done
    criterion: torchrun world-8 CPU bench completes with the integrated
    quorum/sync/recall collectives in the step path)."""
    import json
    import subprocess
    import sys
    from pathlib import Path

    repo = Path(__file__).resolve().parent.parent
    proc = None
    for attempt in range(3):
        port = _find_free_port()
        proc = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
             "--master-port", str(port), str(repo / "bench.py"),
             "--gpus", "8", "--steps", "1", "--warmup", "0",
             "--agents-per-gpu", "2"],
            capture_output=True, text=True, timeout=420, cwd=repo)
        if proc.returncode == 0:
            break
    assert proc.returncode == 0, proc.stderr[-2000:]
    line = next(l for l in proc.stdout.splitlines()
                if l.startswith("{") and '"metric"' in l)
    out = json.loads(line)
    assert out["n_gpus"] == 8
    assert out["value"] > 0


def _worker4(rank: int, world: int, port: int, fail_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        os.environ["LOCAL_RANK"] = str(rank)
        ctx = SwarmContext.from_env(device=torch.device("cpu"))
        votes = torch.tensor([1 if rank % 2 == 0 else -1] * 5,
                             dtype=torch.int8)
        tally = ctx.quorum_tally(votes)
        assert tally == {"yes": 10, "no": 10, "abstain": 0, "total": 20}, tally
        got = ctx.broadcast_blob({"g": rank} if rank == 0 else None, src=0)
        assert got == {"g": 0}
        v, i = ctx.topk_merge(torch.tensor([float(rank), float(rank) - 10.0]),
                              torch.tensor([rank * 2, rank * 2 + 1]), k=4)
        assert i.tolist() == [6, 4, 2, 0], i.tolist()  # ranks desc
        ctx.barrier()
        torch.distributed.destroy_process_group()
    except Exception as e:
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


def test_swarm_collectives_world4():
    """Driver-shape world=4: quorum tally, broadcast, top-k merge."""
    port = _find_free_port()
    fail_q = mp.get_context("spawn").SimpleQueue()
    mp.spawn(_worker4, args=(4, port, fail_q), nprocs=4, join=True)
    assert fail_q.empty()
