"""Learned-encoder embedder: a MiniLM-shaped transformer on the GPU.

The reference embeds memory text with Xenova/all-MiniLM-L6-v2 (384-dim fp32,
mean-pool + L2-normalize, src/shared/embeddings.ts:48-69). This build has no
network, so pretrained MiniLM weights cannot be fetched; the encoder here is
the same *shape* (6 layers, 384 dim, 12 heads, mean-pool + normalize) with:

  * a self-contained subword tokenizer (suffix-stripping word pieces), so
    morphological variants share pieces ("deploying" → "deploy" + "##ing");
  * piece embeddings from the deterministic hash projection (embedder.py) —
    the residual stream therefore carries a lexically meaningful base that
    the randomly-initialized contextual layers perturb only mildly (init
    std 0.02), keeping retrieval quality ≥ the hash baseline while the
    architecture is the real one;
  * order-invariant masked mean pooling (the hash baseline's bigram
    features are word-order sensitive; pooling is not);
  * a loader for real pretrained weights (safetensors state dict via
    ROOMAMD_ENCODER_WEIGHTS) that replaces the random init when available.

Forward runs in fp32 on whatever device is available (the model is tiny:
~7M params); on ROCm the Linear layers ride hipBLASLt. Output format (384-d
fp32, L2-normalized) and the SQLite blob codec are unchanged, so the HIP
vector store and hybrid fusion are untouched.
"""
from __future__ import annotations

import hashlib
import math
import os
import re
from typing import Optional

import torch

from ..core.constants import EMBEDDING_DIM

_TOKEN_RE = re.compile(r"[a-z0-9]+")
# common English suffixes, longest-first (deterministic subword stemming)
_SUFFIXES = ["ations", "ation", "ities", "iness", "ingly", "ments", "ness",
             "ingly", "able", "ible", "ment", "tion", "sion", "ing", "est",
             "ers", "ies", "ed", "er", "ly", "es", "s"]


def tokenize(text: str, max_pieces: int = 128) -> list[str]:
    """Lowercased word pieces: stem + ##suffix when a known suffix applies
    and the stem stays ≥3 chars; long words additionally split every 8
    chars. Deterministic and self-contained (no vocab file)."""
    pieces: list[str] = []
    for w in _TOKEN_RE.findall(text.lower()):
        stem, suffix = w, None
        for suf in _SUFFIXES:
            if w.endswith(suf) and len(w) - len(suf) >= 3:
                stem, suffix = w[: len(w) - len(suf)], "##" + suf
                break
        while len(stem) > 8:
            pieces.append(stem[:8] + "~")
            stem = stem[8:]
        pieces.append(stem)
        if suffix:
            pieces.append(suffix)
        if len(pieces) >= max_pieces:
            break
    return pieces[:max_pieces]


def _piece_vector(piece: str) -> torch.Tensor:
    """Deterministic 384-dim hash projection of one piece (the lexical base
    the contextual layers refine)."""
    v = torch.zeros(EMBEDDING_DIM)
    for salt in range(4):  # 4 sparse ±1 features per piece
        h = hashlib.blake2s(f"{salt}:{piece}".encode(), digest_size=8).digest()
        idx = int.from_bytes(h[:4], "little") % EMBEDDING_DIM
        v[idx] += 1.0 if h[4] & 1 else -1.0
    return v / 2.0  # unit-ish norm (4 ±1 entries)


class MiniEncoder:
    """6-layer pre-norm transformer encoder, 384 dim, 12 heads, FFN 1536.
    Mean-pool + L2 normalize (embeddings.ts:56-62 semantics)."""

    LAYERS = 6
    HEADS = 12
    FFN = 1536

    def __init__(self, device: Optional[str] = None, seed: int = 7,
                 weights_path: Optional[str] = None):
        self.device = torch.device(device or (
            "cuda" if torch.cuda.is_available() else "cpu"))
        d = EMBEDDING_DIM
        gen = torch.Generator().manual_seed(seed)

        def rnd(*shape):
            return torch.empty(*shape).normal_(0, 0.02, generator=gen)

        self.layers = []
        for _ in range(self.LAYERS):
            self.layers.append({
                "norm1": torch.ones(d), "wq": rnd(d, d), "wk": rnd(d, d),
                "wv": rnd(d, d), "wo": rnd(d, d),
                "norm2": torch.ones(d), "w1": rnd(self.FFN, d),
                "w2": rnd(d, self.FFN),
            })
        self.final_norm = torch.ones(d)
        # sinusoidal positions (deterministic, scaled small so pooling stays
        # order-tolerant)
        pos = torch.arange(256).unsqueeze(1)
        div = torch.exp(torch.arange(0, d, 2) * (-math.log(10000.0) / d))
        pe = torch.zeros(256, d)
        pe[:, 0::2] = torch.sin(pos * div)
        pe[:, 1::2] = torch.cos(pos * div)
        self.pos = 0.02 * pe
        weights_path = weights_path or os.environ.get("ROOMAMD_ENCODER_WEIGHTS")
        if weights_path and os.path.exists(weights_path):
            self._load_weights(weights_path)
        self._to_device()
        self._piece_cache: dict[str, torch.Tensor] = {}
        # anisotropy correction: transformer outputs share a large common
        # direction (GELU's positive mean drifts every text the same way),
        # which flattens cosine contrast. Estimate the common mean from
        # deterministic probe texts once and center before normalizing
        # ("all-but-the-top" post-processing).
        self.mu = torch.zeros(d, device=self.device)
        probes = [" ".join(
            hashlib.blake2s(f"{i}:{j}".encode()).hexdigest()[:6]
            for j in range(12)) for i in range(24)]
        self.mu = self._pool(probes).mean(0).to(self.device)

    def _to_device(self) -> None:
        for layer in self.layers:
            for k in layer:
                layer[k] = layer[k].to(self.device)
        self.final_norm = self.final_norm.to(self.device)
        self.pos = self.pos.to(self.device)

    def _load_weights(self, path: str) -> None:
        """Replace random init with a user-provided safetensors state dict
        (keys: layers.<i>.{norm1,wq,wk,wv,wo,norm2,w1,w2}, final_norm)."""
        from safetensors.torch import load_file
        sd = load_file(path)
        for i, layer in enumerate(self.layers):
            for k in layer:
                key = f"layers.{i}.{k}"
                if key in sd:
                    layer[k] = sd[key].float()
        if "final_norm" in sd:
            self.final_norm = sd["final_norm"].float()

    def _rmsnorm(self, x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
        return x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-6) * w

    @torch.inference_mode()
    def encode(self, texts: list[str]) -> torch.Tensor:
        """[B, 384] fp32, centered + L2-normalized."""
        pooled = self._pool(texts).to(self.device) - self.mu
        return torch.nn.functional.normalize(pooled, dim=-1).cpu()

    @torch.inference_mode()
    def _pool(self, texts: list[str]) -> torch.Tensor:
        """Raw mean-pooled encoder output (pre-centering)."""
        d = EMBEDDING_DIM
        piece_lists = [tokenize(t) or ["<empty>"] for t in texts]
        T = max(len(p) for p in piece_lists)
        B = len(texts)
        x = torch.zeros(B, T, d)
        mask = torch.zeros(B, T)
        for b, pieces in enumerate(piece_lists):
            vecs = []
            for p in pieces:
                vec = self._piece_cache.get(p)
                if vec is None:
                    vec = _piece_vector(p)
                    if len(self._piece_cache) < 100_000:
                        self._piece_cache[p] = vec
                vecs.append(vec)
            # one stack+copy per text (per-piece __setitem__ was ~10 ms/text)
            x[b, : len(pieces)] = torch.stack(vecs)
            mask[b, : len(pieces)] = 1.0
        x = x.to(self.device) + self.pos[:T]
        mask = mask.to(self.device)
        attn_bias = (1.0 - mask).unsqueeze(1).unsqueeze(1) * -1e9  # [B,1,1,T]
        hd = d // self.HEADS
        for layer in self.layers:
            h = self._rmsnorm(x, layer["norm1"])
            q = (h @ layer["wq"].T).view(B, T, self.HEADS, hd).transpose(1, 2)
            k = (h @ layer["wk"].T).view(B, T, self.HEADS, hd).transpose(1, 2)
            v = (h @ layer["wv"].T).view(B, T, self.HEADS, hd).transpose(1, 2)
            s = (q @ k.transpose(-1, -2)) * (hd ** -0.5) + attn_bias
            a = torch.softmax(s, dim=-1) @ v
            a = a.transpose(1, 2).reshape(B, T, d)
            x = x + a @ layer["wo"].T
            h = self._rmsnorm(x, layer["norm2"])
            x = x + torch.nn.functional.gelu(h @ layer["w1"].T) @ layer["w2"].T
        x = self._rmsnorm(x, self.final_norm)
        # masked mean pool (embeddings.ts:56-62); centering+normalize in encode()
        return ((x * mask.unsqueeze(-1)).sum(1)
                / mask.sum(1, keepdim=True)).cpu()


_encoder: MiniEncoder | None = None


def get_encoder() -> MiniEncoder:
    global _encoder
    if _encoder is None:
        _encoder = MiniEncoder()
    return _encoder


def encode_texts(texts: list[str]) -> list[list[float]]:
    return get_encoder().encode(texts).tolist()
