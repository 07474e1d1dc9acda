"""CPU unit tests for the engine's slot/session/eviction bookkeeping
(room_amd/engine/admission.py) — the logic whose GPU-only coverage let the
round-1 prefill regression ship. Runs against a real PagedKVCache on the CPU
device (the admitter touches no GPU-specific state).

Session semantics mirror the reference's agent_sessions continuity
(src/shared/agent-loop.ts:462-532): prefix reuse, divergence reset, durable
re-prefill after eviction."""
import pytest
import torch

from room_amd.engine.admission import SessionAdmitter
from room_amd.engine.kv_cache import BLOCK_SIZE, PagedKVCache


class Req:
    """Duck-typed stand-in for GenRequest (admitter only reads/writes these)."""

    def __init__(self, prompt, session_key=None, max_new_tokens=8):
        self.prompt_tokens = list(prompt)
        self.session_key = session_key
        self.max_new_tokens = max_new_tokens
        self.out_tokens = []
        self.slot = -1
        self.pos = 0
        self.pending_prefill = []
        self.prefill_tokens_run = 0


def make(num_blocks=64, max_seqs=4, max_position=4096):
    cache = PagedKVCache(num_layers=1, num_kv_heads=1, head_dim=8,
                         num_blocks=num_blocks, max_seqs=max_seqs,
                         max_blocks_per_seq=max_position // BLOCK_SIZE,
                         device=torch.device("cpu"))
    return cache, SessionAdmitter(cache, max_position)


def run_to_completion(adm, req, n_out=3):
    """Simulate the scheduler finishing a request normally."""
    req.out_tokens = list(range(n_out))
    adm.finish(req, req.prompt_tokens + req.out_tokens[:-1])


def test_fresh_admit_prefills_whole_prompt():
    cache, adm = make()
    req = Req(list(range(80)), session_key="a")
    adm.admit(req)
    assert req.slot >= 0
    assert req.pos == 0
    assert req.pending_prefill == req.prompt_tokens
    assert req.prefill_tokens_run == 80          # the round-1 regression: was 0
    assert "a" in adm.sessions
    assert adm.sessions["a"].slot == req.slot
    assert req.slot in adm.active_slots


def test_fresh_admit_without_session_frees_slot_on_finish():
    cache, adm = make()
    free0 = len(cache.free_slots)
    for _ in range(20):                           # round-1 leak: exhausted at ~63
        req = Req(list(range(30)))
        adm.admit(req)
        cache.ensure_capacity(req.slot, 32)
        run_to_completion(adm, req)
    assert len(cache.free_slots) == free0
    assert cache.blocks_free() == cache.num_blocks - 1  # block 0 reserved


def test_session_prefix_reuse():
    cache, adm = make()
    base = list(range(100, 160))
    r1 = Req(base, session_key="s")
    adm.admit(r1)
    cache.ensure_capacity(r1.slot, len(base) + 4)
    r1.out_tokens = [7, 8, 9]
    adm.finish(r1, base + [7, 8])
    ext = base + [7, 8] + [500, 501]
    r2 = Req(ext, session_key="s")
    adm.admit(r2)
    assert r2.slot == r1.slot                    # same slot retained
    assert r2.pos == len(base) + 2               # rolls forward to divergence
    assert r2.pending_prefill == [500, 501]
    assert r2.prefill_tokens_run == 2


def test_identical_prompt_reruns_last_token():
    cache, adm = make()
    base = list(range(40))
    r1 = Req(base, session_key="s")
    adm.admit(r1)
    run_to_completion(adm, r1)
    sess_tokens = adm.sessions["s"].tokens
    r2 = Req(sess_tokens, session_key="s")       # exactly what's cached
    adm.admit(r2)
    # the common-prefix limit always leaves the last token to re-run (its
    # logits seed the first sampled token)
    assert r2.prefill_tokens_run == 1
    assert r2.pending_prefill == [sess_tokens[-1]]
    assert r2.pos == len(sess_tokens) - 1


def test_divergent_prompt_resets_session():
    cache, adm = make()
    r1 = Req(list(range(40)), session_key="s")
    adm.admit(r1)
    slot1 = r1.slot
    run_to_completion(adm, r1)
    r2 = Req(list(range(900, 940)), session_key="s")   # no common prefix
    adm.admit(r2)
    assert r2.pos == 0
    assert r2.pending_prefill == r2.prompt_tokens
    assert adm.sessions["s"].slot == r2.slot
    # old slot went back to the pool (maybe re-alloc'd as r2.slot)
    assert slot1 == r2.slot or slot1 in cache.free_slots


def test_lru_eviction_under_slot_pressure():
    cache, adm = make(max_seqs=3)
    for i in range(6):
        req = Req(list(range(20)), session_key=f"k{i}")
        adm.admit(req)
        run_to_completion(adm, req)
    assert len(adm.sessions) <= 3
    assert "k5" in adm.sessions
    assert "k0" not in adm.sessions


def test_lru_eviction_under_block_pressure():
    # 12 usable blocks; each session wants ~6 → admitting the 3rd evicts k0
    cache, adm = make(num_blocks=13, max_seqs=8)
    for i in range(3):
        req = Req(list(range(i * 1000, i * 1000 + 60)), session_key=f"k{i}",
                  max_new_tokens=4)
        adm.admit(req)
        cache.ensure_capacity(req.slot, req.pos + len(req.pending_prefill) + 4)
        run_to_completion(adm, req)
    assert "k0" not in adm.sessions
    assert "k2" in adm.sessions


def test_eviction_never_evicts_own_or_active_slot():
    cache, adm = make(num_blocks=9, max_seqs=8)
    r1 = Req(list(range(60)), session_key="mine", max_new_tokens=4)
    adm.admit(r1)   # needs ~6 of 8 usable blocks; only own session exists
    # block-pressure loop must not evict "mine" (exclude_slot) → no crash,
    # session intact
    assert "mine" in adm.sessions
    assert adm.sessions["mine"].slot == r1.slot


def test_cancelled_finish_drops_session_and_frees_slot():
    cache, adm = make()
    req = Req(list(range(30)), session_key="c")
    adm.admit(req)
    cache.ensure_capacity(req.slot, 32)
    adm.finish(req, None)                        # cancelled / errored
    assert "c" not in adm.sessions
    assert req.slot in cache.free_slots
    assert cache.blocks_free() == cache.num_blocks - 1


def test_release_while_active_defers_slot_free():
    cache, adm = make()
    req = Req(list(range(30)), session_key="r")
    adm.admit(req)
    adm.release("r")                             # e.g. worker deleted mid-gen
    assert "r" not in adm.sessions
    assert req.slot not in cache.free_slots      # still decoding on it
    adm.finish(req, [1, 2, 3])                   # session gone → slot freed
    assert req.slot in cache.free_slots


def test_context_window_truncation_keeps_head_and_tail():
    cache, adm = make(num_blocks=300, max_seqs=2, max_position=512)
    prompt = list(range(2000))
    req = Req(prompt, max_new_tokens=64)
    adm.admit(req)
    budget = 512 - 64 - 8
    assert len(req.prompt_tokens) == budget
    assert req.prompt_tokens[0] == 0             # head kept (system prompt)
    assert req.prompt_tokens[-1] == 1999         # recent tail kept
    assert req.prefill_tokens_run == budget


def test_slot_exhaustion_by_active_requests_raises_cachefull():
    """All slots held by ACTIVE requests → CacheFull (the scheduler defers
    the admission rather than failing the caller)."""
    from room_amd.engine.admission import CacheFull
    cache, adm = make(max_seqs=2)
    r1, r2 = Req([1, 2, 3]), Req([4, 5, 6])
    adm.admit(r1)
    adm.admit(r2)                    # both slots now active
    r3 = Req([7, 8, 9])
    with pytest.raises(CacheFull):
        adm.admit(r3)
    # a completion frees a slot and the deferred admit succeeds
    adm.finish(r1, None)
    adm.admit(r3)
    assert r3.slot >= 0
