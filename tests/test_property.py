"""Property-based tests (hypothesis) for the codec-critical paths: the
synthetic tokenizer's id round-trip (KV-prefix validity depends on it) and
the embedding blob codec (on-disk format, reference embeddings.ts:116-122)."""
import math

from hypothesis import given, settings
from hypothesis import strategies as st

from room_amd.db.queries import blob_to_vector, vector_to_blob
from room_amd.engine import tokenizer as tok


@settings(max_examples=200, deadline=None)
@given(st.lists(st.integers(min_value=0, max_value=tok.VOCAB_SIZE - 1),
                min_size=0, max_size=64))
def test_token_ids_survive_decode_reencode(ids):
    """Generated ids must round-trip through decode→encode exactly — the
    engine reuses KV prefixes by comparing re-encoded session text, so any
    drift would silently invalidate cached prefixes."""
    text = tok.decode(ids)
    again = tok.encode(text)
    assert again == ids


@settings(max_examples=200, deadline=None)
@given(st.text(alphabet=st.characters(blacklist_categories=("Cs",)),
               max_size=200))
def test_encode_always_in_vocab(text):
    ids = tok.encode(text)
    assert all(0 <= i < tok.VOCAB_SIZE for i in ids)


@settings(max_examples=200, deadline=None)
@given(st.lists(st.floats(min_value=-1e6, max_value=1e6, allow_nan=False,
                          width=32), min_size=0, max_size=384))
def test_vector_blob_roundtrip(vec):
    out = blob_to_vector(vector_to_blob(vec))
    assert len(out) == len(vec)
    for a, b in zip(out, vec):
        assert math.isclose(a, b, rel_tol=1e-6, abs_tol=1e-6)


@settings(max_examples=100, deadline=None)
@given(st.lists(st.sampled_from(
    ["system", "user", "assistant"]), min_size=1, max_size=6).flatmap(
        lambda roles: st.tuples(st.just(roles), st.lists(
            st.text(max_size=40), min_size=len(roles), max_size=len(roles)))))
def test_chat_encoding_monotone(roles_contents):
    """encode_chat output grows with the transcript and stays in-vocab."""
    roles, contents = roles_contents
    msgs = [{"role": r, "content": c} for r, c in zip(roles, contents)]
    ids = tok.encode_chat(msgs)
    assert all(0 <= i < tok.VOCAB_SIZE for i in ids)
    longer = tok.encode_chat(msgs + [{"role": "user", "content": "more text"}])
    assert len(longer) > len(ids)


@settings(max_examples=150, deadline=None)
@given(st.integers(min_value=0, max_value=59),
       st.integers(min_value=0, max_value=23),
       st.integers(min_value=1, max_value=28),
       st.integers(min_value=1, max_value=12))
def test_cron_next_after_always_matches(minute, hour, dom, month):
    """next_after of a fully-pinned cron lands exactly on a matching time,
    strictly in the future."""
    from datetime import datetime

    from room_amd.core.cron import CronExpression as Cron
    c = Cron(f"{minute} {hour} {dom} {month} *")
    base = datetime(2026, 1, 1, 0, 0)
    nxt = c.next_after(base)
    assert nxt is not None and nxt > base
    assert (nxt.minute, nxt.hour, nxt.day, nxt.month) == (minute, hour, dom,
                                                          month)
    assert c.matches(nxt)


@settings(max_examples=150, deadline=None)
@given(st.integers(min_value=2, max_value=30))
def test_cron_step_interval_spacing(step):
    """*/N minute crons fire at multiples of N, and consecutive fires are
    exactly the step (or the hour rollover remainder) apart."""
    from datetime import datetime

    from room_amd.core.cron import CronExpression as Cron
    c = Cron(f"*/{step} * * * *")
    t = datetime(2026, 3, 5, 10, 1)
    a = c.next_after(t)
    b = c.next_after(a)
    assert a.minute % step == 0 and b.minute % step == 0
    assert c.matches(a) and c.matches(b)
    assert (b - a).total_seconds() <= step * 60 + 3600  # step or rollover


# ---------------------------------------------------------------- admission
# Stateful property test: arbitrary admit/finish/release/evict interleavings
# must preserve the slot/block accounting invariants (the class of bug that
# broke round 1 shipped because only example-based GPU tests covered this).

from hypothesis.stateful import (Bundle, RuleBasedStateMachine, invariant,
                                 rule)


class AdmissionMachine(RuleBasedStateMachine):
    def __init__(self):
        super().__init__()
        import torch

        from room_amd.engine.admission import CacheFull, SessionAdmitter
        from room_amd.engine.kv_cache import PagedKVCache
        self.CacheFull = CacheFull
        self.cache = PagedKVCache(num_layers=1, num_kv_heads=1, head_dim=8,
                                  num_blocks=40, max_seqs=5,
                                  max_blocks_per_seq=64,
                                  device=torch.device("cpu"))
        self.adm = SessionAdmitter(self.cache, max_position=512)
        self.active: list = []
        self.counter = 0

    reqs = Bundle("reqs")

    @rule(target=reqs, sess=st.booleans(), plen=st.integers(1, 60))
    def admit(self, sess, plen):
        class R:
            pass
        r = R()
        self.counter += 1
        r.prompt_tokens = list(range(self.counter, self.counter + plen))
        r.session_key = f"s{self.counter % 3}" if sess else None
        r.max_new_tokens = 8
        r.out_tokens = []
        r.slot, r.pos, r.pending_prefill, r.prefill_tokens_run = -1, 0, [], 0
        try:
            self.adm.admit(r)
        except self.CacheFull:
            r.slot = -1
            return r
        assert r.pending_prefill, "admitted request must have prefill work"
        self.cache.ensure_capacity(r.slot, r.pos + len(r.pending_prefill) + 2)
        self.active.append(r)
        return r

    @rule(r=reqs, ok=st.booleans())
    def finish(self, r, ok):
        if r not in self.active:
            return
        self.active.remove(r)
        r.out_tokens = [1, 2, 3]
        self.adm.finish(r, (r.prompt_tokens + r.out_tokens[:-1]) if ok else None)

    @rule(key=st.sampled_from(["s0", "s1", "s2", "nope"]))
    def release(self, key):
        self.adm.release(key)

    @invariant()
    def accounting_holds(self):
        c = self.cache
        # every slot is exactly one of: free, or allocated (tracked in seq_len)
        assert len(c.free_slots) + len(c.seq_len) == c.max_seqs
        assert set(c.free_slots).isdisjoint(c.seq_len.keys())
        # block conservation (block 0 reserved)
        used = sum(c.seq_nblocks.values())
        assert used + c.blocks_free() == c.num_blocks - 1
        # sessions and active requests only reference live slots
        for s in self.adm.sessions.values():
            assert s.slot in c.seq_len
        for slot in self.adm.active_slots:
            assert slot in c.seq_len


TestAdmissionMachine = AdmissionMachine.TestCase


class KVCacheMachine(RuleBasedStateMachine):
    """Block-pool accounting invariants under arbitrary alloc/extend/free
    interleavings (engine/kv_cache.py, CPU tensors): no block owned twice,
    conservation of blocks, capacity math, slot reuse safety."""

    def __init__(self):
        super().__init__()
        import torch

        from room_amd.engine.kv_cache import BLOCK_SIZE, PagedKVCache
        self.BS = BLOCK_SIZE
        self.c = PagedKVCache(num_layers=1, num_kv_heads=1, head_dim=8,
                              num_blocks=64, max_seqs=8, max_blocks_per_seq=16,
                              device=torch.device("cpu"))
        self.live: set[int] = set()

    @rule()
    def alloc(self):
        if self.c.free_slots:
            slot = self.c.alloc_seq()
            assert slot not in self.live
            self.live.add(slot)
        else:
            import pytest as _pytest
            with _pytest.raises(RuntimeError):
                self.c.alloc_seq()

    @rule(tokens=st.integers(min_value=1, max_value=260))
    def extend(self, tokens):
        if not self.live:
            return
        slot = sorted(self.live)[0]
        need = (tokens + self.BS - 1) // self.BS
        if need > self.c.max_blocks_per_seq:
            import pytest as _pytest
            with _pytest.raises(RuntimeError):
                self.c.ensure_capacity(slot, tokens)
        elif need - self.c.seq_nblocks[slot] > len(self.c.free_blocks):
            import pytest as _pytest
            with _pytest.raises(RuntimeError):
                self.c.ensure_capacity(slot, tokens)
        else:
            self.c.ensure_capacity(slot, tokens)
            assert self.c.seq_nblocks[slot] >= need

    @rule()
    def free(self):
        if self.live:
            slot = sorted(self.live)[-1]
            self.c.free_seq(slot)
            self.live.discard(slot)

    @invariant()
    def no_block_shared_and_conserved(self):
        owned = []
        for slot in self.live:
            n = self.c.seq_nblocks.get(slot, 0)
            owned.extend(self.c._bt_host[slot][:n])
        assert len(owned) == len(set(owned)), "block owned twice"
        assert not (set(owned) & set(self.c.free_blocks)), "owned AND free"
        assert 0 not in owned, "reserved scrap block handed out"
        # conservation: free + owned == all blocks minus the reserved one
        assert len(self.c.free_blocks) + len(owned) == self.c.num_blocks - 1

    @invariant()
    def slots_conserved(self):
        assert len(self.c.free_slots) + len(self.live) == self.c.max_seqs


KVCacheTest = KVCacheMachine.TestCase
KVCacheTest.settings = settings(max_examples=60, stateful_step_count=40,
                                deadline=None)
