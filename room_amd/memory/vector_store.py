"""HBM-resident semantic vector store (BASELINE config 4).

The reference scans embedding BLOBs in SQLite with sqlite-vec's
vec_distance_cosine (src/shared/db-queries.ts:995-1010). Here the hot index
is a [capacity, 384] bf16 matrix resident in GPU HBM (10M rows = 7.4 GB of
the 288 GB HBM3E), queried by the vs_topk HIP kernel (wave-per-row dot +
per-block top-k + merge). SQLite's embeddings table remains the durable copy;
this store is rebuilt from it on boot.

On CPU-only environments the store falls back to a torch CPU matmul so the
engine semantics (and tests) run anywhere; the GPU path is exercised by
tests/test_memory_gpu.py.
"""
from __future__ import annotations

import threading

import torch

from ..core.constants import EMBEDDING_DIM


class GpuVectorStore:
    def __init__(self, capacity: int = 1_000_000, device: str | None = None):
        self.device = torch.device(device or
                                   ("cuda" if torch.cuda.is_available() else "cpu"))
        self.capacity = capacity
        dtype = torch.bfloat16 if self.device.type == "cuda" else torch.float32
        self.mat = torch.zeros(capacity, EMBEDDING_DIM, dtype=dtype,
                               device=self.device)
        self.ids = torch.zeros(capacity, dtype=torch.int64)  # host-side id map
        self.size = 0
        self._lock = threading.Lock()
        self._id_to_row: dict[int, int] = {}

    def upsert(self, entity_id: int, vector: list[float] | torch.Tensor) -> int:
        v = torch.as_tensor(vector, dtype=torch.float32)
        v = torch.nn.functional.normalize(v, dim=-1)
        with self._lock:
            row = self._id_to_row.get(entity_id)
            if row is None:
                if self.size >= self.capacity:
                    raise RuntimeError("vector store full")
                row = self.size
                self.size += 1
                self._id_to_row[entity_id] = row
                self.ids[row] = entity_id
            self.mat[row] = v.to(self.mat.dtype).to(self.device)
        return row

    def upsert_batch(self, entity_ids: list[int], vectors: torch.Tensor) -> None:
        v = torch.nn.functional.normalize(vectors.float(), dim=-1)
        with self._lock:
            for i, eid in enumerate(entity_ids):
                row = self._id_to_row.get(eid)
                if row is None:
                    if self.size >= self.capacity:
                        raise RuntimeError("vector store full")
                    row = self.size
                    self.size += 1
                    self._id_to_row[eid] = row
                    self.ids[row] = eid
                self.mat[row] = v[i].to(self.mat.dtype).to(self.device)

    def remove(self, entity_id: int) -> None:
        with self._lock:
            row = self._id_to_row.pop(entity_id, None)
            if row is None:
                return
            last = self.size - 1
            if row != last:  # swap-delete
                self.mat[row] = self.mat[last]
                last_id = int(self.ids[last])
                self.ids[row] = last_id
                self._id_to_row[last_id] = row
            self.mat[last].zero_()
            self.size = last

    def search(self, query: list[float] | torch.Tensor, k: int = 5
               ) -> list[tuple[int, float]]:
        """Returns [(entity_id, cosine)] sorted desc."""
        if self.size == 0:
            return []
        q = torch.as_tensor(query, dtype=torch.float32)
        q = torch.nn.functional.normalize(q, dim=-1).to(self.device)
        k = min(k, self.size)
        if self.device.type == "cuda":
            from .. import ops
            v, i = ops.vs_topk(self.mat[:self.size].contiguous(), q, k)
            rows = i.cpu().tolist()
            vals = v.cpu().tolist()
        else:
            scores = self.mat[:self.size] @ q
            vals_t, rows_t = scores.topk(k)
            rows, vals = rows_t.tolist(), vals_t.tolist()
        out = []
        for r, s in zip(rows, vals):
            if r < 0:
                continue
            out.append((int(self.ids[r]), float(s)))
        return out

    def rebuild_from_db(self, db) -> int:
        """Load all durable embeddings from SQLite into HBM (boot path)."""
        from ..db import queries as q
        rows = q.all_embeddings(db)
        count = 0
        for r in rows:
            vec = q.blob_to_vector(r["vector"])
            if len(vec) == EMBEDDING_DIM:
                self.upsert(r["entity_id"], vec)
                count += 1
        return count


class MemoryService:
    """Binds the durable SQLite memory to the GPU store + embedder: the
    remember/recall surface the agent tools and API use. Hybrid fusion stays
    host-side and identical to the reference (FTS RRF×0.4 + cosine×0.6)."""

    def __init__(self, ldb, store: GpuVectorStore | None = None,
                 capacity: int = 1_000_000):
        from . import embedder
        self.ldb = ldb
        self.embedder = embedder
        self.store = store or GpuVectorStore(capacity=capacity)

    def embed(self, text: str) -> list[float]:
        return self.embedder.embed(text)

    def remember(self, room_id: int | None, name: str, content: str,
                 category: str | None = None, source: str = "agent") -> int:
        from ..db import queries as q
        with self.ldb as db:
            ent = q.get_entity_by_name(db, name, room_id)
            if ent is None:
                ent = q.create_entity(db, name, category=category, room_id=room_id)
            q.add_observation(db, ent["id"], content, source=source)
            vec = self.embedder.embed(f"{name} {content}")
            q.upsert_embedding(db, ent["id"], vec,
                               self.embedder.text_hash(content))
        self.store.upsert(ent["id"], vec)
        return ent["id"]

    def recall(self, room_id: int | None, query: str, limit: int = 5) -> list[dict]:
        """Hybrid recall. In the multi-GPU swarm each rank owns a memory
        shard: every shard fuses its local FTS+cosine hits, then the fused
        top-k rides one RCCL all-gather and the global list is re-ranked by
        score — the topk-merge semantics of SURVEY §2c, integrated here so
        every agent recall is swarm-wide."""
        from ..db import queries as q
        qvec = self.embedder.embed(query)
        semantic = self.store.search(qvec, k=20)
        with self.ldb as db:
            hits = q.hybrid_search(db, query, qvec, limit=limit, room_id=room_id,
                                   semantic_hits=semantic)
        from ..parallel.swarm import get_swarm_context
        ctx = get_swarm_context()
        if ctx is not None and ctx.collective_safe:
            payload = [{"name": h.get("name"), "category": h.get("category"),
                        "score": h.get("score"),
                        "observations": h.get("observations", [])[:5],
                        "shard": ctx.rank} for h in hits]
            merged = [h for shard in ctx.allgather_obj(payload) for h in shard]
            merged.sort(key=lambda h: -(h.get("score") or 0.0))
            # dedupe by name (same entity remembered on several shards)
            seen, out = set(), []
            for h in merged:
                if h["name"] in seen:
                    continue
                seen.add(h["name"])
                out.append(h)
                if len(out) >= limit:
                    break
            return out
        return hits

    def index_pending(self, batch: int = 64) -> int:
        """Background indexing of unembedded entities (reference:
        embedding-indexer.ts — name + first 5 observations, 2000-char cap)."""
        from ..db import queries as q
        with self.ldb as db:
            pending = q.get_unembedded_entities(db, limit=batch)
            done = 0
            for ent in pending:
                obs = q.get_observations(db, ent["id"])[:5]
                text = (ent["name"] + " " +
                        " ".join(o["content"] for o in obs))[:2000]
                vec = self.embedder.embed(text)
                q.upsert_embedding(db, ent["id"], vec,
                                   self.embedder.text_hash(text))
                try:
                    self.store.upsert(ent["id"], vec)
                except RuntimeError:
                    # GPU index at capacity: SQLite row is still embedded
                    # (durable + FTS-searchable); hot-index misses degrade
                    # recall to keyword-only for the overflow, not an error
                    pass
                done += 1
        return done
