"""Device-free admission / session / eviction bookkeeping for LocalEngine.

Factored out of the scheduler thread so the slot/session/block accounting is
unit-testable on CPU (round-1 shipped a fatal admission regression precisely
because this logic could only run on a GPU box). The class owns the
session-as-KV-cache map — the MI355X replacement for the reference's
agent_sessions continuity (src/shared/agent-loop.ts:462-532): a session keeps
its sequence slot and paged KV blocks across agent cycles, and the durable
agent_sessions SQLite row lets an evicted session re-prefill next cycle.

It talks to the KV cache only through alloc_seq/free_seq/free_slots/
blocks_free, so tests drive it with a real PagedKVCache on the CPU device.
"""
from __future__ import annotations

import time
from typing import Callable, Iterable, Optional

from .kv_cache import BLOCK_SIZE, PagedKVCache


class CacheFull(RuntimeError):
    """Every KV slot is held by an ACTIVE request (nothing evictable).
    Retryable: the scheduler defers the admission until a request
    completes, instead of failing the caller."""


class Session:
    __slots__ = ("slot", "tokens", "last_used")

    def __init__(self, slot: int, tokens: list[int]):
        self.slot = slot
        self.tokens = tokens
        self.last_used = time.time()


class SessionAdmitter:
    """Owns sessions + slot lifetime. All calls happen under the engine lock.

    Invariants it maintains:
      * every admitted request has a valid slot and a non-empty
        pending_prefill (fresh requests prefill their WHOLE prompt);
      * a request with a session_key always has a registered session whose
        slot matches the request's slot;
      * a request with no session_key gets a throwaway slot that is freed on
        completion/cancellation (no slot leak);
      * sessions of active requests are never evicted.
    """

    def __init__(self, cache: PagedKVCache, max_position: int):
        self.cache = cache
        self.max_position = max_position
        self.sessions: dict[str, Session] = {}
        self.active_slots: set[int] = set()

    # ------------------------------------------------------------ admission

    def admit(self, req) -> None:
        """Assign a slot + prefill plan to `req` (a GenRequest-like object).

        Mutates req.prompt_tokens (context-window truncation), req.slot,
        req.pos, req.pending_prefill, req.prefill_tokens_run.
        """
        cache = self.cache
        # context-window guard: middle-truncate prompts that cannot fit (keep
        # the head [system prompt] and the recent tail), mirroring the
        # reference's context-overflow recovery (agent-loop.ts:773-782)
        budget = self.max_position - req.max_new_tokens - 8
        if len(req.prompt_tokens) > budget:
            head = budget // 4
            tail = budget - head
            req.prompt_tokens = (req.prompt_tokens[:head]
                                 + req.prompt_tokens[-tail:])

        sess = self.sessions.get(req.session_key) if req.session_key else None
        if sess is not None and sess.slot in self.active_slots:
            # another request is mid-flight on this session's slot (e.g. a
            # timed-out caller retrying before its cancelled request reached
            # a scheduler boundary). Never share or free a live slot: drop
            # the session binding and admit fresh — the in-flight finish
            # sees the missing session and frees the old slot itself.
            self.sessions.pop(req.session_key, None)
            sess = None
        if sess is not None:
            # reuse the longest common token prefix: the session keeps its
            # slot and blocks; position rolls back to the divergence point.
            cached = sess.tokens
            common = 0
            limit = min(len(cached), len(req.prompt_tokens) - 1)
            while common < limit and cached[common] == req.prompt_tokens[common]:
                common += 1
            if common > 0:
                req.slot = sess.slot
                req.pos = common
                req.pending_prefill = req.prompt_tokens[common:]
                sess.last_used = time.time()
            else:
                cache.free_seq(sess.slot)
                self.sessions.pop(req.session_key, None)
                sess = None

        if req.slot < 0:  # fresh admission (no reusable session prefix)
            if not cache.free_slots:
                self.evict_lru()
            req.slot = cache.alloc_seq()
            req.pos = 0
            req.pending_prefill = list(req.prompt_tokens)
            if req.session_key:
                self.sessions[req.session_key] = Session(req.slot, [])

        # block-pressure relief: if the pool cannot hold this request's
        # prompt + generation budget, evict idle LRU sessions until it can
        # (their durable agent_sessions rows re-prefill on the next cycle)
        need = ((len(req.prompt_tokens) + req.max_new_tokens) // BLOCK_SIZE
                + 2 - cache.seq_nblocks.get(req.slot, 0))
        while cache.blocks_free() < need:
            try:
                self.evict_lru(exclude_slot=req.slot)
            except CacheFull:
                break  # nothing evictable — ensure_capacity will raise

        req.prefill_tokens_run = len(req.pending_prefill)
        if not req.pending_prefill:
            # prompt identical to cache (rare): re-run last token for logits
            req.pos = max(0, req.pos - 1)
            req.pending_prefill = [req.prompt_tokens[-1]]
        self.active_slots.add(req.slot)

    # ------------------------------------------------------------ completion

    def finish(self, req, kv_tokens: Optional[list[int]]) -> None:
        """Request left the active set. `kv_tokens` = the token stream whose
        KV now resides in the slot (prompt + out minus the unwritten last
        sample), or None on cancellation/error (session state unknown →
        drop the session so the next cycle re-prefills from the DB row)."""
        self.active_slots.discard(req.slot)
        key = req.session_key
        sess = self.sessions.get(key) if key else None
        if sess is not None and sess.slot == req.slot:
            if kv_tokens is None:
                self.sessions.pop(key, None)
                self.cache.free_seq(req.slot)
            else:
                sess.tokens = kv_tokens
                sess.last_used = time.time()
        else:
            # session-less request, or session released/evicted mid-flight
            self.cache.free_seq(req.slot)

    def release(self, session_key: str) -> None:
        """Explicit session teardown (worker deleted / room stopped)."""
        s = self.sessions.pop(session_key, None)
        if s is not None and s.slot not in self.active_slots:
            self.cache.free_seq(s.slot)
        # if active: finish() sees the missing session and frees the slot

    # ------------------------------------------------------------ eviction

    def evict_lru(self, exclude_slot: int = -1) -> None:
        """Slot/block pressure: drop the least-recently-used idle session
        (its KV blocks free; the durable agent_sessions row lets the next
        cycle re-prefill). Sessions of currently-active requests — and the
        request being admitted (exclude_slot) — are never evicted."""
        blocked = self.active_slots | {exclude_slot}
        candidates = [(s.last_used, key) for key, s in self.sessions.items()
                      if s.slot not in blocked]
        if not candidates:
            raise CacheFull("KV cache: no evictable sessions")
        _, key = min(candidates)
        s = self.sessions.pop(key)
        self.cache.free_seq(s.slot)
