"""In-process GPU inference engine with continuous batching.

Replaces the reference's Ollama sidecar + HTTP hop (src/shared/local-model.ts,
agent-executor.ts:327-338). Agents call chat() from their own threads; a
single scheduler thread owns the GPU and fuses concurrent agents' work into
batched prefill/decode kernel launches — the control inversion SURVEY §7
calls out: per-agent cycle semantics unchanged, per-GPU kernels batched.

Sessions: chat() reuses the sequence slot (and its paged KV blocks) when the
new prompt extends the previous token stream — session-as-KV-cache, with the
agent_sessions SQLite row as the durable fallback.
"""
from __future__ import annotations

import os
import queue
import threading
import time
from dataclasses import dataclass, field
from typing import Optional

import torch

from .. import ops
from ..models.qwen3_moe import Qwen3MoEConfig, Qwen3MoEModel
from . import tokenizer as tok
from .admission import SessionAdmitter
from .kv_cache import BLOCK_SIZE, PagedKVCache
from .types import AgentExecutionOptions, ToolDef

PREFILL_CHUNK = 4096          # max tokens per prefill forward
DEFAULT_MAX_SEQS = 64
DECODE_BUCKETS = (1, 2, 3, 4, 5, 6, 7, 8, 16, 32, 64)  # hipGraph capture sizes
# ≤8 buckets are exact (padding inflates MoE pair traffic ~linearly)


@dataclass
class GenRequest:
    prompt_tokens: list[int]
    max_new_tokens: int = 128
    temperature: float = 0.7
    top_p: float = 0.95
    top_k: int = 40
    session_key: Optional[str] = None
    done: threading.Event = field(default_factory=threading.Event)
    out_tokens: list[int] = field(default_factory=list)
    error: Optional[str] = None
    prefill_tokens_run: int = 0   # actually-prefilled (after cache reuse)
    cancelled: bool = False       # set by the caller on timeout; scheduler drops
    # scheduler state
    slot: int = -1
    pos: int = 0                  # next position to write
    pending_prefill: list[int] = field(default_factory=list)
    last_token: int = -1


class _DecodeGraph:
    """One captured hipGraph per (batch bucket, sampling params), replayed K
    steps per host sync (multi-step decode).

    Inside the capture: forward → fused sampler (on-device RNG state) →
    token-history append (device step counter) → sampled tokens copied into
    the next step's input → positions += 1. A block of K replays therefore
    runs K full decode steps with ZERO host work; the host reads the K×B
    token history once at the end. Padded lanes use a reserved pad slot whose
    KV writes land in scrap block 0."""

    KMAX = 32

    def __init__(self, engine: "LocalEngine", bucket: int,
                 temperature: float, top_p: float, top_k: int):
        self.bucket = bucket
        self.pad_slot = engine.pad_slot
        dev = engine.device
        self.tok_in = torch.zeros(bucket, dtype=torch.int64, device=dev)
        self.seq_in = torch.full((bucket,), engine.pad_slot, dtype=torch.int32,
                                 device=dev)
        self.pos_in = torch.zeros(bucket, dtype=torch.int32, device=dev)
        self.seeds = torch.randint(1, 2**62, (bucket,), dtype=torch.int64,
                                   device=dev)
        self.hist = torch.zeros(self.KMAX * bucket, dtype=torch.int32, device=dev)
        self.ctr = torch.zeros(1, dtype=torch.int32, device=dev)
        self.hist_host = torch.zeros(self.KMAX * bucket, dtype=torch.int32,
                                     pin_memory=True)
        # pinned host staging for the block entry
        self.h_tok = torch.zeros(bucket, dtype=torch.int64, pin_memory=True)
        self.h_seq = torch.full((bucket,), engine.pad_slot, dtype=torch.int32,
                                pin_memory=True)
        self.h_pos = torch.zeros(bucket, dtype=torch.int32, pin_memory=True)
        model, cache = engine.model, engine.cache
        from .. import ops as _ops
        C = _ops._require()

        def run_step():
            logits = model.forward(self.tok_in, self.seq_in, self.pos_in,
                                   cache.block_table, cache.kcaches,
                                   cache.vcaches)
            toks = torch.empty(bucket, dtype=torch.int32, device=dev)
            C.sample_tokens_v4(toks, logits, self.seeds, top_k, temperature,
                               top_p)
            C.hist_append(self.hist, self.ctr, toks, self.KMAX)
            self.tok_in.copy_(toks)      # feed the next replay
            self.pos_in.add_(1)

        # warmup on a side stream (required before capture)
        strm = torch.cuda.Stream(dev)
        strm.wait_stream(torch.cuda.current_stream(dev))
        with torch.cuda.stream(strm):
            for _ in range(2):
                run_step()
        torch.cuda.current_stream(dev).wait_stream(strm)
        torch.cuda.synchronize(dev)
        self.graph = torch.cuda.CUDAGraph()
        # thread_local: agents' encoder/embedding work on other threads may
        # call the allocator mid-capture; global mode corrupts those calls
        with torch.cuda.graph(self.graph, capture_error_mode="thread_local"):
            run_step()

    def run_block(self, tokens: list[int], slots: list[int],
                  positions: list[int], k: int) -> list[int]:
        """Run k decode steps; returns the flat K×B token history (host)."""
        n, b = len(tokens), self.bucket
        k = min(k, self.KMAX)
        for i in range(n):
            self.h_tok[i] = tokens[i]
            self.h_seq[i] = slots[i]
            self.h_pos[i] = positions[i]
        for i in range(n, b):
            self.h_tok[i] = 0
            self.h_seq[i] = self.pad_slot
            self.h_pos[i] = 0
        self.tok_in.copy_(self.h_tok, non_blocking=True)
        self.seq_in.copy_(self.h_seq, non_blocking=True)
        self.pos_in.copy_(self.h_pos, non_blocking=True)
        self.ctr.zero_()
        for _ in range(k):
            self.graph.replay()
        self.hist_host[:k * b].copy_(self.hist[:k * b])  # syncs
        return self.hist_host[:k * b].tolist()


PREFILL_BUCKETS = (512, 1024, 2048, 4096)   # graph-captured chunk shapes
PREFILL_MAX_ROWS = 8                         # fixed logits-row slots


class _PrefillGraph:
    """One captured hipGraph per prefill bucket size.

    The prefill forward was built host-sync-free (device-side MoE tile
    descriptors, tensor-driven rope/attention), but launching its ~700
    python-level ops costs ~30 ms per chunk — more than the GPU compute it
    enqueues, so prefill ran launch-bound. Chunks are padded to a few fixed
    bucket shapes (pad rows target the engine's scrap pad slot; zero-count
    q-tiles no-op in flash_prefill) and each bucket's whole forward replays
    as one graph. logits are always computed for PREFILL_MAX_ROWS fixed row
    slots; the scheduler slices the real ones."""

    def __init__(self, engine: "LocalEngine", tb: int):
        import sys as _sys
        _t0 = time.time()
        print(f"[room_amd] capturing prefill graph tb={tb} ...",
              file=_sys.stderr, flush=True)
        dev = engine.device
        self.tb = tb
        self.pad_slot = engine.pad_slot
        qt = 32  # flash_prefill q-tile rows (FP_QTOK)
        # worst case: 8 request segments + 1 pad segment, each wasting <1 tile
        self.gmax = tb // qt + PREFILL_MAX_ROWS + 2
        self.tok = torch.zeros(tb, dtype=torch.int64, device=dev)
        self.seq = torch.full((tb,), engine.pad_slot, dtype=torch.int32,
                              device=dev)
        self.pos = torch.zeros(tb, dtype=torch.int32, device=dev)
        self.rows = torch.zeros(PREFILL_MAX_ROWS, dtype=torch.int64, device=dev)
        self.qtiles = torch.zeros(self.gmax, 2, dtype=torch.int32, device=dev)
        # pinned staging
        self.h_tok = torch.zeros(tb, dtype=torch.int64, pin_memory=True)
        self.h_seq = torch.full((tb,), engine.pad_slot, dtype=torch.int32,
                                pin_memory=True)
        self.h_pos = torch.zeros(tb, dtype=torch.int32, pin_memory=True)
        self.h_rows = torch.zeros(PREFILL_MAX_ROWS, dtype=torch.int64,
                                  pin_memory=True)
        self.h_qtiles = torch.zeros(self.gmax, 2, dtype=torch.int32,
                                    pin_memory=True)
        # guards pinned-staging reuse: back-to-back same-bucket chunks (long
        # prompts, nothing sampled between) would otherwise overwrite h_* on
        # the host while the previous run's async H2D copies are in flight
        self.copied = torch.cuda.Event()
        model, cache = engine.model, engine.cache
        model.capture_gemm = True   # hipBLASLt is not capture-safe

        def run_fwd():
            return model.forward(self.tok, self.seq, self.pos,
                                 cache.block_table, cache.kcaches,
                                 cache.vcaches, logits_rows=self.rows,
                                 qtile_desc=self.qtiles)

        # warmup on the CURRENT stream, not a side stream: the canonical
        # side-stream warmup HANGS on the 30b config at tb=512 (bisected in
        # scripts/gpu_pg_bisect.py — warmup-only on a side stream deadlocks;
        # the identical forward on the default stream warms up, captures and
        # replays green). torch.cuda.graph still captures on its own stream.
        for _ in range(2):
            run_fwd()
        torch.cuda.synchronize(dev)
        self.graph = torch.cuda.CUDAGraph()
        try:
            with torch.cuda.graph(self.graph,
                                  capture_error_mode="thread_local"):
                self.logits = run_fwd()        # [PREFILL_MAX_ROWS, vocab] f32
        finally:
            model.capture_gemm = False
        print(f"[room_amd] prefill graph tb={tb} captured "
              f"in {time.time() - _t0:.1f}s", file=_sys.stderr, flush=True)

    def run(self, tokens: list[int], seq_ids: list[int], q_pos: list[int],
            last_rows: list[int], qtiles_host: list) -> torch.Tensor:
        n = len(tokens)
        tb = self.tb
        self.copied.synchronize()   # previous call's H2D copies done
        self.h_tok[:n] = torch.tensor(tokens, dtype=torch.int64)
        self.h_tok[n:] = 0
        self.h_seq[:n] = torch.tensor(seq_ids, dtype=torch.int32)
        self.h_seq[n:] = self.pad_slot
        self.h_pos[:n] = torch.tensor(q_pos, dtype=torch.int32)
        self.h_pos[n:] = 0
        k = len(last_rows)
        self.h_rows[:k] = torch.tensor(last_rows, dtype=torch.int64)
        self.h_rows[k:] = 0
        g = len(qtiles_host)
        self.h_qtiles[:g] = torch.tensor(qtiles_host, dtype=torch.int32)
        self.h_qtiles[g:] = 0                  # zero-count tiles: no-op
        self.tok.copy_(self.h_tok, non_blocking=True)
        self.seq.copy_(self.h_seq, non_blocking=True)
        self.pos.copy_(self.h_pos, non_blocking=True)
        self.rows.copy_(self.h_rows, non_blocking=True)
        self.qtiles.copy_(self.h_qtiles, non_blocking=True)
        self.copied.record()
        self.graph.replay()
        return self.logits


class LocalEngine:
    """ChatEngine implementation running qwen3-coder-30b on this GPU."""

    def __init__(self, cfg: Qwen3MoEConfig | None = None,
                 device: str = "cuda", seed: int = 1234,
                 kv_gb: float | None = None, max_seqs: int = DEFAULT_MAX_SEQS):
        if not torch.cuda.is_available():
            raise RuntimeError("LocalEngine requires a GPU (MI355X)")
        if cfg is None:
            cfg = (Qwen3MoEConfig.tiny()
                   if os.environ.get("ROOMAMD_MODEL_CONFIG") == "tiny"
                   else Qwen3MoEConfig.qwen3_coder_30b())
        self.cfg = cfg
        self.device = torch.device(device)
        if self.device.type == "cuda" and self.device.index is None:
            # resolve NOW, in the constructing thread: the CUDA current
            # device is per-thread, so the scheduler thread would otherwise
            # see device 0 on every rank of a one-process-per-GPU job and
            # launch kernels on rank-0's streams against rank-k's memory
            self.device = torch.device("cuda", torch.cuda.current_device())
        t0 = time.time()
        self.model = Qwen3MoEModel(cfg, self.device, seed=seed)
        self.load_seconds = time.time() - t0

        bpb = PagedKVCache.bytes_per_block(cfg.num_layers, cfg.num_kv_heads,
                                           cfg.head_dim)
        if kv_gb is None:
            free_b, _total = torch.cuda.mem_get_info(self.device)
            kv_bytes = max(int(free_b - 8e9), 2 << 30)  # leave 8 GB headroom
        else:
            kv_bytes = int(kv_gb * (1 << 30))
        max_blocks_per_seq = cfg.max_position // BLOCK_SIZE
        num_blocks = min(kv_bytes // bpb, max_seqs * max_blocks_per_seq + 1)
        self.cache = PagedKVCache(cfg.num_layers, cfg.num_kv_heads, cfg.head_dim,
                                  int(num_blocks), max_seqs, max_blocks_per_seq,
                                  self.device)
        # pad slot for hipGraph batch padding (one scrap block absorbs the
        # dummy lanes' KV writes)
        self.pad_slot = self.cache.alloc_seq()
        self.cache.ensure_capacity(self.pad_slot, 1)
        self.graphs_enabled = (os.environ.get("ROOMAMD_NO_GRAPHS") != "1")
        # hipGraph-captured prefill is OPT-IN: capture-safe on the tiny
        # config (multi-bucket, concurrent sessions, prefix reuse all pass —
        # scripts/gpu_prefill_graph_probe*.py) and cuts enqueue 0.53→0.26 s,
        # but on the 30b config the tb=512 bucket's warmup/capture hits an
        # HSAIL memory fault not yet isolated (profiles/PERF_NOTES.md round-2
        # "prefill graphs"); eager prefill is the default until it is.
        self.prefill_graphs_enabled = (
            os.environ.get("ROOMAMD_PREFILL_GRAPHS") == "1")
        self._graphs: dict[tuple, _DecodeGraph] = {}
        self._prefill_graphs: dict[int, _PrefillGraph] = {}
        self._graphs_broken = False
        self._prefill_graphs_broken = False
        self.admitter = SessionAdmitter(self.cache, cfg.max_position)
        self._queue: "queue.Queue[GenRequest]" = queue.Queue()
        self._active: list[GenRequest] = []
        self._waiting: list[GenRequest] = []   # deferred: all slots active
        self._lock = threading.Lock()
        self._stop = False
        self.stats = {"decode_steps": 0, "decode_tokens": 0, "prefill_tokens": 0,
                      "decode_time": 0.0, "prefill_time": 0.0,
                      "prefill_prep_time": 0.0, "prefill_enq_time": 0.0,
                      "graphs_captured": 0}
        # capture prefill-bucket graphs up front, before any agent/encoder
        # thread can put concurrent work on the device mid-capture (captures
        # are thread_local-mode too, but boot-time capture removes the window
        # entirely); decode graphs stay lazy (keyed by sampling params)
        if (self.graphs_enabled and self.prefill_graphs_enabled
                and os.environ.get("ROOMAMD_NO_PRECAPTURE") != "1"):
            try:
                for tb in PREFILL_BUCKETS:
                    self._prefill_graphs[tb] = _PrefillGraph(self, tb)
            except Exception as e:
                import sys as _sys
                print(f"[room_amd] prefill graph pre-capture failed: {e}",
                      file=_sys.stderr)
                self._prefill_graphs.clear()
                self._prefill_graphs_broken = True
                torch.cuda.synchronize(self.device)
        self._thread = threading.Thread(target=self._scheduler_loop, daemon=True,
                                        name="room-amd-engine")
        self._thread.start()

    @property
    def sessions(self):
        return self.admitter.sessions

    # ------------------------------------------------------------ public API

    def generate(self, prompt_tokens: list[int], max_new_tokens: int = 128,
                 temperature: float = 0.7, top_p: float = 0.95, top_k: int = 40,
                 session_key: str | None = None,
                 timeout: float | None = None) -> GenRequest:
        if timeout is None:
            timeout = float(os.environ.get("ROOMAMD_GEN_TIMEOUT", "600"))
        req = GenRequest(prompt_tokens=list(prompt_tokens),
                         max_new_tokens=max_new_tokens, temperature=temperature,
                         top_p=top_p, top_k=top_k, session_key=session_key)
        if os.environ.get("ROOMAMD_CYCLE_PROF") == "1":
            import sys as _sys
            print(f"[cycleprof] enqueue {session_key} {len(prompt_tokens)}tok "
                  f"(t={time.time()%100:.3f})", file=_sys.stderr)
        self._queue.put(req)
        if not req.done.wait(timeout):
            # mark cancelled: the scheduler drops it (admit or next step
            # boundary) and frees its slot — it must NOT keep decoding to
            # max_new_tokens holding KV blocks and GPU time
            req.cancelled = True
            raise TimeoutError("generation timed out")
        if req.error:
            raise RuntimeError(req.error)
        return req

    def chat(self, messages: list[dict], tools: list[ToolDef],
             options: AgentExecutionOptions) -> tuple[str, int, int]:
        """ChatEngine protocol (engine/providers.py)."""
        prompt = tok.encode_chat(messages)
        skey = f"w{options.worker_id}" if options.worker_id is not None else None
        req = self.generate(prompt, max_new_tokens=options.max_new_tokens,
                            temperature=options.temperature, top_p=options.top_p,
                            top_k=options.top_k, session_key=skey)
        text = tok.decode(req.out_tokens)
        return text, len(prompt), len(req.out_tokens)

    def release_session(self, session_key: str) -> None:
        with self._lock:
            self.admitter.release(session_key)

    def shutdown(self) -> None:
        self._stop = True
        self._thread.join(timeout=5)

    # ------------------------------------------------------------ scheduler

    def _admit(self, req: GenRequest) -> None:
        # map token ids into this model's vocab (the synthetic tokenizer emits
        # Qwen-range ids; reduced test configs have a smaller embedding table —
        # out-of-range ids would be an out-of-bounds gather on the GPU)
        V = self.cfg.vocab_size
        req.prompt_tokens = [t % V for t in req.prompt_tokens]
        # slot/session/block bookkeeping lives in the device-free admitter
        # (admission.py) so it has CPU unit-test coverage
        self.admitter.admit(req)

    def _complete(self, req: GenRequest, ok: bool) -> None:
        """Take `req` out of the active set and resolve its future. ok=False
        (cancelled/error) drops the session so the slot/blocks free."""
        kv_tokens = (req.prompt_tokens + req.out_tokens[:-1]) if ok else None
        with self._lock:
            self.admitter.finish(req, kv_tokens)
        self._active.remove(req)
        req.done.set()

    def _scheduler_loop(self) -> None:
        torch.cuda.set_device(self.device.index)   # resolved in __init__
        while not self._stop:
            try:
                self._scheduler_iteration()
            except Exception as e:  # engine errors resolve all active futures
                import sys
                import traceback
                traceback.print_exc(file=sys.stderr)
                for r in list(self._active):
                    r.error = f"engine error: {e}"
                    self._complete(r, ok=False)

    def _try_admit(self, req: GenRequest) -> bool:
        """True = admitted; False = deferred (all slots held by ACTIVE
        requests — retry when one completes). Other failures resolve the
        future with an error (the caller blocks on req.done)."""
        from .admission import CacheFull
        try:
            with self._lock:
                self._admit(req)
            self._active.append(req)
            return True
        except CacheFull:
            self._waiting.append(req)
            return False
        except Exception as e:
            req.error = f"admit failed: {e}"
            req.done.set()
            return True  # resolved, don't retry

    def _scheduler_iteration(self) -> None:
        # re-try deferred admissions first (slots may have freed)
        waiting, self._waiting = self._waiting, []
        for req in waiting:
            if req.cancelled:
                req.done.set()
                continue
            self._try_admit(req)
        # admit new requests
        block = not (self._active or self._waiting)
        try:
            while True:
                req = self._queue.get(timeout=0.05 if block else 0)
                block = False
                if req.cancelled:  # caller gave up before admission
                    req.done.set()
                    continue
                self._try_admit(req)
        except queue.Empty:
            pass
        # drop requests whose caller timed out (a timed-out
        # request must not keep decoding and holding a KV slot)
        for r in [r for r in self._active if r.cancelled]:
            r.error = "cancelled"
            self._complete(r, ok=False)
        if not self._active:
            return

        # phase 1: prefill pending prompts (batched across requests, chunked)
        pre = [r for r in self._active if r.pending_prefill]
        if pre:
            self._prefill_step(pre)
            return  # re-check queue between chunks

        # phase 2: one decode step for all active sequences
        self._decode_step([r for r in self._active if not r.pending_prefill])

    def _prefill_step(self, reqs: list[GenRequest]) -> None:
        t0 = time.time()
        host_t0 = t0
        budget = PREFILL_CHUNK
        tokens, seq_ids, q_pos, last_rows, sampled_reqs = [], [], [], [], []
        segments = []  # (row0, count) per request — host-side q-tile info
        for r in reqs:
            if budget <= 0:
                break
            take = min(budget, len(r.pending_prefill))
            chunk = r.pending_prefill[:take]
            r.pending_prefill = r.pending_prefill[take:]
            self.cache.ensure_capacity(r.slot, r.pos + take)
            segments.append((len(tokens), take))
            tokens.extend(chunk)
            seq_ids.extend([r.slot] * take)
            q_pos.extend(range(r.pos, r.pos + take))
            r.pos += take
            budget -= take
            if not r.pending_prefill:  # prompt complete → sample first token
                last_rows.append(len(tokens) - 1)
                sampled_reqs.append(r)
        dev = self.device
        n = len(tokens)
        graph_ok = (self.graphs_enabled and self.prefill_graphs_enabled
                    and not self._graphs_broken
                    and not self._prefill_graphs_broken
                    and n > 8 and len(sampled_reqs) <= PREFILL_MAX_ROWS)
        if graph_ok:
            # graph path: pad the chunk to a fixed bucket shape and replay
            # the captured forward (the eager forward is python-launch-bound
            # at ~30 ms/chunk — more than the GPU work it enqueues)
            tb = next(b for b in PREFILL_BUCKETS if b >= n)
            # pad rows live in their own 1-token-context segment — in a COPY:
            # the eager fallback below reuses `segments` against n-row tensors,
            # so a leaked pad segment would read q rows past the tensor end
            seg_graph = segments + ([(n, tb - n)] if n < tb else [])
            qtiles_host = self._qtile_list(seg_graph)
            self.stats["prefill_prep_time"] += time.time() - host_t0
            try:
                g = self._prefill_graphs.get(tb)
                if g is None:
                    g = _PrefillGraph(self, tb)
                    self._prefill_graphs[tb] = g
                logits = g.run(tokens, seq_ids, q_pos, last_rows, qtiles_host)
                self.stats["prefill_enq_time"] += time.time() - host_t0
                self.stats["prefill_tokens"] += n
                if sampled_reqs:
                    self._sample_and_append(sampled_reqs,
                                            logits[:len(sampled_reqs)])
                self.stats["prefill_time"] += time.time() - t0
                return
            except Exception as e:
                import sys
                print(f"[room_amd] hipGraph prefill disabled: {e}",
                      file=sys.stderr)
                self._prefill_graphs_broken = True
                # a capture that aborted mid-graph can leave stale stream
                # state; fence the device before falling back to eager
                try:
                    torch.cuda.synchronize(dev)
                except Exception:
                    pass
        tokens_t = torch.tensor(tokens, dtype=torch.int64, device=dev)
        seq_t = torch.tensor(seq_ids, dtype=torch.int32, device=dev)
        pos_t = torch.tensor(q_pos, dtype=torch.int32, device=dev)
        rows_t = torch.tensor(last_rows, dtype=torch.int64, device=dev)
        qtiles = (ops.build_qtile_desc(segments, dev)
                  if len(tokens) > 8 else None)
        self.stats["prefill_prep_time"] += time.time() - host_t0
        logits = self.model.forward(tokens_t, seq_t, pos_t, self.cache.block_table,
                                    self.cache.kcaches, self.cache.vcaches,
                                    logits_rows=rows_t, qtile_desc=qtiles)
        self.stats["prefill_enq_time"] += time.time() - host_t0
        self.stats["prefill_tokens"] += len(tokens)
        if sampled_reqs:
            self._sample_and_append(sampled_reqs, logits)
        self.stats["prefill_time"] += time.time() - t0

    @staticmethod
    def _qtile_list(segments: list) -> list:
        """(row0, count) segments → ≤32-row q-tile descriptors (host list)."""
        out = []
        for row0, count in segments:
            r = row0
            while r < row0 + count:
                take = min(32, row0 + count - r)
                out.append((r, take))
                r += take
        return out

    def _decode_step(self, reqs: list[GenRequest]) -> None:
        if not reqs:
            return
        t0 = time.time()
        dev = self.device
        tokens = [r.last_token for r in reqs]
        slots = [r.slot for r in reqs]
        positions = [r.pos for r in reqs]

        # multi-step graph block: K bounded by the smallest remaining budget so
        # every request stops exactly at max_new (admission latency is K steps)
        steps = 1
        graph_ok = (self.graphs_enabled and not self._graphs_broken
                    and len(reqs) <= 8)
        r0 = reqs[0]
        uniform = all((r.temperature, r.top_p, r.top_k)
                      == (r0.temperature, r0.top_p, r0.top_k) for r in reqs)
        if graph_ok and uniform:
            steps = max(1, min(min(r.max_new_tokens - len(r.out_tokens)
                                   for r in reqs), _DecodeGraph.KMAX))
            for r in reqs:
                self.cache.ensure_capacity(r.slot, r.pos + steps)
            bucket = next(b for b in DECODE_BUCKETS if b >= len(reqs))
            # context band: 64 attention splits once any sequence's KV span
            # passes 8k (shorter per-WG serial chunk chains); banded graphs
            # keep each capture's grid fixed
            band = 64 if max(positions) + steps >= 16384 else 32
            self.model.attn_splits = band
            key = (bucket, band, round(r0.temperature, 3), round(r0.top_p, 3),
                   r0.top_k)
            try:
                g = self._graphs.get(key)
                if g is None:
                    g = _DecodeGraph(self, bucket, r0.temperature, r0.top_p,
                                     r0.top_k)
                    self._graphs[key] = g
                    self.stats["graphs_captured"] = len(self._graphs)
                hist = g.run_block(tokens, slots, positions, steps)
                finished = []
                for si in range(steps):
                    row = hist[si * g.bucket: si * g.bucket + len(reqs)]
                    for r, t in zip(reqs, row):
                        if r in finished:
                            continue
                        r.last_token = int(t)
                        r.out_tokens.append(int(t))
                        r.pos += 1
                        if (len(r.out_tokens) >= r.max_new_tokens
                                or t in (tok.EOS, tok.IM_END)):
                            finished.append(r)
                for r in finished:
                    self._complete(r, ok=True)
                self.stats["decode_steps"] += steps
                self.stats["decode_tokens"] += steps * len(reqs)
                self.stats["decode_time"] += time.time() - t0
                return
            except Exception as e:
                import sys
                print(f"[room_amd] hipGraph decode disabled: {e}",
                      file=sys.stderr)
                self._graphs_broken = True

        # eager fallback: one step
        self.model.attn_splits = 64 if max(positions) >= 16384 else 32
        for r in reqs:
            self.cache.ensure_capacity(r.slot, r.pos + 1)
        for r in reqs:
            r.pos += 1
        tokens_t = torch.tensor(tokens, dtype=torch.int64, device=dev)
        seq_t = torch.tensor(slots, dtype=torch.int32, device=dev)
        pos_t = torch.tensor(positions, dtype=torch.int32, device=dev)
        logits = self.model.forward(tokens_t, seq_t, pos_t,
                                    self.cache.block_table,
                                    self.cache.kcaches, self.cache.vcaches)
        self._sample_and_append(reqs, logits)
        self.stats["decode_steps"] += 1
        self.stats["decode_tokens"] += len(reqs)
        self.stats["decode_time"] += time.time() - t0

    def _sample_and_append(self, reqs: list[GenRequest], logits: torch.Tensor) -> None:
        # sample per sampling-param group (concurrent requests
        # with different params must not inherit the first request's settings)
        dev = self.device
        groups: dict[tuple, list[int]] = {}
        for i, r in enumerate(reqs):
            groups.setdefault((r.temperature, r.top_p, r.top_k), []).append(i)
        toks_host = [0] * len(reqs)
        for (temperature, top_p, top_k), idxs in groups.items():
            rows = (logits if len(idxs) == len(reqs)
                    else logits[torch.tensor(idxs, device=dev)])
            seeds = torch.randint(1, 2**62, (rows.size(0),), dtype=torch.int64,
                                  device=dev)
            toks = ops.sample_tokens(rows, seeds, top_k=top_k,
                                     temperature=temperature, top_p=top_p)
            for i, t in zip(idxs, toks.tolist()):
                toks_host[i] = int(t)
        self._finish_host_tokens(reqs, toks_host)

    def _finish_host_tokens(self, reqs: list[GenRequest],
                            toks_host: list[int]) -> None:
        finished = []
        for r, t in zip(reqs, toks_host):
            r.last_token = int(t)
            r.out_tokens.append(int(t))
            if (len(r.out_tokens) >= r.max_new_tokens
                    or t in (tok.EOS, tok.IM_END)):
                finished.append(r)
        for r in finished:
            # the sampled token at r.pos is NOT yet in KV; it will be written
            # if the session continues (prompt extension re-runs it)
            self._complete(r, ok=True)


# ------------------------------------------------------------ singleton

_engine: LocalEngine | None = None
_engine_lock = threading.Lock()


def get_local_engine(model: str = "qwen3-coder-30b") -> LocalEngine:
    global _engine
    with _engine_lock:
        if _engine is None:
            _engine = LocalEngine()
        return _engine


def set_local_engine(engine: LocalEngine) -> None:
    global _engine
    _engine = engine
