"""384-dim text embedder.

The reference runs Xenova/all-MiniLM-L6-v2 via ONNX on CPU (mean-pool +
L2-normalize, src/shared/embeddings.ts:48-69). This environment has no
network to fetch those weights, so the default embedder is a deterministic
hashed n-gram projection (feature hashing) producing the same 384-dim fp32
L2-normalized output format and the same blob codec — semantically useful
(token overlap → cosine similarity), fully self-contained, and identical
across processes. A learned encoder can be dropped in behind `embed()`
without touching callers (the GPU store only sees 384-dim vectors).
"""
from __future__ import annotations

import hashlib
import math
import re

from ..core.constants import EMBEDDING_DIM

_TOKEN_RE = re.compile(r"[a-z0-9]+")


def _features(text: str) -> list[str]:
    toks = _TOKEN_RE.findall(text.lower())
    feats = list(toks)
    feats += [f"{a}_{b}" for a, b in zip(toks, toks[1:])]  # bigrams
    return feats


def _mode() -> str:
    """Embedder selection: 'encoder' (MiniLM-shaped transformer, encoder.py)
    or 'hash' (n-gram projection). Default: encoder on GPU hosts (the GPU
    runs the forward; SURVEY §2b calls for a GPU encoder), hash on CPU-only
    hosts (the 6-layer fp32 forward is too slow per-call on one core).
    Override with ROOMAMD_EMBEDDER=encoder|hash."""
    import os
    m = os.environ.get("ROOMAMD_EMBEDDER")
    if m in ("encoder", "hash"):
        return m
    try:
        import torch
        return "encoder" if torch.cuda.is_available() else "hash"
    except Exception:
        return "hash"


_EMBED_CACHE: dict = {}
_EMBED_CACHE_MAX = 4096


def embed(text: str) -> list[float]:
    """384-dim L2-normalized embedding (deterministic per host mode).

    LRU-cached: agent cycles re-embed near-identical recall queries (the
    room objective, the bench's fixed probe) every cycle, and the encoder
    forward costs ~40 ms of event-loop time per call — cache hits remove
    the prompt-build stagger that delays agents joining the decode batch."""
    mode = _mode()
    key = (mode, text)
    hit = _EMBED_CACHE.get(key)
    if hit is not None:
        return list(hit)
    if mode == "encoder":
        from . import encoder
        vec = encoder.encode_texts([text])[0]
    else:
        vec = embed_hash(text)
    if len(_EMBED_CACHE) >= _EMBED_CACHE_MAX:
        _EMBED_CACHE.pop(next(iter(_EMBED_CACHE)))
    _EMBED_CACHE[key] = tuple(vec)
    return vec


def embed_hash(text: str) -> list[float]:
    """Deterministic hashed n-gram projection (CPU fallback / baseline)."""
    vec = [0.0] * EMBEDDING_DIM
    for f in _features(text):
        h = hashlib.blake2s(f.encode(), digest_size=8).digest()
        idx = int.from_bytes(h[:4], "little") % EMBEDDING_DIM
        sign = 1.0 if h[4] & 1 else -1.0
        vec[idx] += sign
    norm = math.sqrt(sum(x * x for x in vec))
    if norm == 0:
        vec[0] = 1.0
        return vec
    return [x / norm for x in vec]


def embed_batch(texts: list[str]) -> list[list[float]]:
    if _mode() == "encoder":
        from . import encoder
        return encoder.encode_texts(texts)
    return [embed_hash(t) for t in texts]


def cosine_similarity(a: list[float], b: list[float]) -> float:
    dot = sum(x * y for x, y in zip(a, b))
    na = math.sqrt(sum(x * x for x in a)) or 1.0
    nb = math.sqrt(sum(x * x for x in b)) or 1.0
    return dot / (na * nb)


def text_hash(text: str) -> str:
    """16-hex content hash (same shape as the reference's, embeddings.ts:124-126)."""
    return hashlib.sha256(text.encode()).hexdigest()[:16]
