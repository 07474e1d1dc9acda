"""Memory subsystem tests: embedder invariants (mirrors the reference's
embeddings.test.ts golden tests), CPU vector store, MemoryService."""
import pytest
import torch

from room_amd.core import room
from room_amd.db import LockedDb
from room_amd.memory import embedder
from room_amd.memory.vector_store import GpuVectorStore, MemoryService


def test_embedder_invariants():
    v = embedder.embed("hello world")
    assert len(v) == 384
    norm = sum(x * x for x in v) ** 0.5
    assert norm == pytest.approx(1.0, abs=1e-6)
    assert embedder.embed("hello world") == v  # deterministic
    # similar text → higher cosine than unrelated
    sim = embedder.cosine_similarity(v, embedder.embed("hello there world"))
    dis = embedder.cosine_similarity(v, embedder.embed("quantum flux capacitor"))
    assert sim > dis


def test_text_hash_shape():
    h = embedder.text_hash("abc")
    assert len(h) == 16 and all(c in "0123456789abcdef" for c in h)


def test_vector_store_cpu_search_and_remove():
    vs = GpuVectorStore(capacity=100, device="cpu")
    vs.upsert(10, embedder.embed("deploy the server"))
    vs.upsert(20, embedder.embed("pricing research notes"))
    vs.upsert(30, embedder.embed("team meeting agenda"))
    hits = vs.search(embedder.embed("server deployment"), k=2)
    assert hits[0][0] == 10
    vs.remove(10)
    hits = vs.search(embedder.embed("server deployment"), k=2)
    assert all(h[0] != 10 for h in hits)
    assert vs.size == 2


def test_vector_store_upsert_overwrites():
    vs = GpuVectorStore(capacity=10, device="cpu")
    vs.upsert(1, embedder.embed("alpha"))
    vs.upsert(1, embedder.embed("totally different text now"))
    assert vs.size == 1
    hits = vs.search(embedder.embed("totally different text now"), k=1)
    assert hits[0][1] > 0.9


def test_memory_service_remember_recall(db):
    r = room.create_room(db, "memsvc", worker_model="stub")
    svc = MemoryService(LockedDb(db), store=GpuVectorStore(capacity=100,
                                                           device="cpu"))
    svc.remember(r["id"], "api pricing", "competitor charges $99 per month")
    svc.remember(r["id"], "deploy steps", "use blue-green deployment on k8s")
    hits = svc.recall(r["id"], "how much does the competitor charge", limit=2)
    assert hits and hits[0]["name"] == "api pricing"


def test_memory_service_rebuild(db):
    r = room.create_room(db, "memrb", worker_model="stub")
    ldb = LockedDb(db)
    svc = MemoryService(ldb, store=GpuVectorStore(capacity=100, device="cpu"))
    svc.remember(r["id"], "fact one", "the sky is blue today")
    # fresh store rebuilt from the durable SQLite copy
    vs2 = GpuVectorStore(capacity=100, device="cpu")
    with ldb as conn:
        n = vs2.rebuild_from_db(conn)
    assert n == 1
    hits = vs2.search(embedder.embed("fact one the sky is blue today"), k=1)
    assert hits[0][1] > 0.9


def test_index_pending(db):
    from room_amd.db import queries as q
    r = room.create_room(db, "memidx", worker_model="stub")
    q.create_entity(db, "unindexed note", room_id=r["id"],
                    observations=["remember this content"])
    ldb = LockedDb(db)
    svc = MemoryService(ldb, store=GpuVectorStore(capacity=100, device="cpu"))
    n = svc.index_pending()
    assert n == 1
    assert svc.store.size == 1
    with ldb as conn:
        assert q.get_unembedded_entities(conn) == []


def test_hybrid_fusion_weights(db):
    """Fusion contract (db-queries.ts:1021-1059): FTS reciprocal-rank
    (k=60) × 0.4 + cosine × 0.6 — a strong semantic hit outranks a
    rank-1 FTS hit, and scores combine additively for entities in both."""
    from room_amd.core import room as room_mod
    from room_amd.db import queries as q

    r = room_mod.create_room(db, "fuse", worker_model="stub")
    kw = q.create_entity(db, "keyword hit", room_id=r["id"],
                         observations=["alpha beta gamma"])
    sem = q.create_entity(db, "semantic hit", room_id=r["id"],
                          observations=["unrelated words entirely"])
    both = q.create_entity(db, "both hit", room_id=r["id"],
                           observations=["alpha beta overlap"])

    semantic = [(sem["id"], 0.95), (both["id"], 0.5)]
    hits = q.hybrid_search(db, "alpha beta", None, limit=3, room_id=r["id"],
                           semantic_hits=semantic)
    by_id = {h["id"]: h["score"] for h in hits}
    # rank-1 FTS alone ≈ 0.4/61 = 0.0066; cosine 0.95 × 0.6 = 0.57
    assert by_id[sem["id"]] > by_id[kw["id"]]
    # "both" accumulates fts + semantic, beating its semantic part alone
    assert by_id[both["id"]] > 0.5 * 0.6
    # exact weight spot check for the pure-semantic entity
    assert abs(by_id[sem["id"]] - 0.95 * 0.6) < 1e-9
