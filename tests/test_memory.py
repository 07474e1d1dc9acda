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
