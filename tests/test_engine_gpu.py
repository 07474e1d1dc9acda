"""End-to-end engine tests on GPU (tiny Qwen3-MoE config: same architecture,
4 layers) — generation, session KV reuse, concurrent batching, agent cycle."""
import threading
import time

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)

from room_amd.engine import tokenizer as tok
from room_amd.engine.llm import LocalEngine
from room_amd.models.qwen3_moe import Qwen3MoEConfig


@pytest.fixture(scope="module")
def engine():
    eng = LocalEngine(cfg=Qwen3MoEConfig.tiny(), kv_gb=2.0, max_seqs=16)
    yield eng
    eng.shutdown()


def test_generate_basic(engine):
    prompt = tok.encode("the quick brown fox " * 20)
    req = engine.generate(prompt, max_new_tokens=16)
    assert len(req.out_tokens) == 16
    assert all(0 <= t < engine.cfg.vocab_size for t in req.out_tokens)
    assert req.prefill_tokens_run == len(prompt)


def test_session_prefix_reuse(engine):
    base = tok.encode("session test alpha beta gamma " * 10)
    r1 = engine.generate(base, max_new_tokens=4, session_key="sess-a")
    assert r1.prefill_tokens_run == len(base)
    ext = base + r1.out_tokens[:-1] + tok.encode("continue now")
    r2 = engine.generate(ext, max_new_tokens=4, session_key="sess-a")
    # only the suffix should have been prefilled (KV prefix reused)
    assert r2.prefill_tokens_run < len(ext) / 2
    engine.release_session("sess-a")


def test_concurrent_chat_batching(engine):
    """Multiple agent threads → the scheduler batches their decode steps."""
    results = {}

    def run(i):
        prompt = tok.encode(f"agent {i} observes the room and acts " * 8)
        req = engine.generate(prompt, max_new_tokens=12, session_key=f"c{i}")
        results[i] = req.out_tokens

    threads = [threading.Thread(target=run, args=(i,)) for i in range(6)]
    before = engine.stats["decode_steps"]
    for t in threads:
        t.start()
    for t in threads:
        t.join(60)
    assert len(results) == 6
    assert all(len(v) == 12 for v in results.values())
    # batching: 6 agents × 12 tokens in far fewer than 72 decode steps
    steps = engine.stats["decode_steps"] - before
    assert steps < 50, f"no batching? {steps} steps for 72 tokens"
    for i in range(6):
        engine.release_session(f"c{i}")


def test_logits_finite(engine):
    prompt = tok.encode("numerics check " * 4)
    req = engine.generate(prompt, max_new_tokens=2, temperature=0.0)
    assert len(req.out_tokens) == 2


def test_agent_cycle_on_gpu_engine(engine):
    """Full observe→prompt→decode→persist cycle through the GPU engine."""
    import asyncio

    from room_amd.core import room as room_mod
    from room_amd.core.agent_loop import AgentLoopManager
    from room_amd.db import LockedDb, init_test_db
    from room_amd.engine.providers import register_engine

    register_engine("tiny-gpu", engine)
    ldb = LockedDb(init_test_db())
    with ldb as db:
        r = room_mod.create_room(db, "gpu-room", goal="Test the GPU cycle",
                                 worker_model="tiny-gpu")
    mgr = AgentLoopManager(ldb)
    out = asyncio.run(mgr.run_cycle(r["id"], r["queen_worker_id"], max_turns=1))
    assert out["result"].success, out["result"].error
    assert out["result"].output_tokens > 0
    with ldb as db:
        from room_amd.db import queries as q
        cycles = q.list_room_cycles(db, r["id"])
        assert cycles[0]["status"] == "completed"


def test_session_lru_eviction():
    """More sessions than KV slots → LRU sessions evicted, not errors."""
    from room_amd.engine.llm import LocalEngine
    from room_amd.models.qwen3_moe import Qwen3MoEConfig
    eng = LocalEngine(cfg=Qwen3MoEConfig.tiny(), kv_gb=1.0, max_seqs=4)
    try:
        # 3 usable slots (one is the graph pad slot); run 6 sessions
        for i in range(6):
            p = tok.encode(f"session {i} content " * 6)
            r = eng.generate(p, max_new_tokens=2, session_key=f"lru-{i}")
            assert len(r.out_tokens) == 2
        assert len(eng.sessions) <= 3
        # oldest sessions were evicted; the newest still has its slot
        assert "lru-5" in eng.sessions
        assert "lru-0" not in eng.sessions
    finally:
        eng.shutdown()


def test_concurrency_storm_with_eviction_pressure():
    """32 threads × repeated generates against 8 KV slots: every request
    completes, no slot/block leak afterwards (the round-1 leak class)."""
    import queue as _q

    from room_amd.engine.llm import LocalEngine
    from room_amd.models.qwen3_moe import Qwen3MoEConfig
    eng = LocalEngine(cfg=Qwen3MoEConfig.tiny(), kv_gb=1.0, max_seqs=8)
    errors: _q.Queue = _q.Queue()

    def run(i):
        try:
            for rep in range(3):
                p = tok.encode(f"storm agent {i} rep {rep} " * 6)
                skey = f"storm-{i}" if i % 2 == 0 else None  # half session-less
                r = eng.generate(p, max_new_tokens=4, session_key=skey,
                                 timeout=120)
                assert len(r.out_tokens) == 4
        except Exception as e:  # pragma: no cover - surfaced below
            errors.put(f"{i}: {e}")

    threads = [threading.Thread(target=run, args=(i,)) for i in range(32)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(240)
    try:
        assert errors.empty(), errors.get()
        # drain: no active requests; all session-less slots returned
        time.sleep(0.3)
        with eng._lock:
            assert not eng.admitter.active_slots
            used = 1 + len(eng.sessions)          # pad slot + live sessions
            free = len(eng.cache.free_slots)
            assert used + free == eng.cache.max_seqs, (used, free)
    finally:
        eng.shutdown()
