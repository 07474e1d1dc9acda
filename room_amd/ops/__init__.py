"""CDNA4 kernel library bindings.

On a GPU box the extension MUST load — there is no eager/PyTorch fallback for
the compute path (a silent fallback would fake GPU test results). CPU-only
environments (CI here) can import this module; calling any op raises.
"""
from __future__ import annotations

import torch

_C = None
_import_error: Exception | None = None
try:
    from room_amd import _C  # type: ignore
except Exception as e:  # pragma: no cover
    _import_error = e

if _C is None and torch.cuda.is_available():
    raise ImportError(
        "room_amd._C HIP extension missing on a GPU machine — build it with "
        "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace`. "
        f"Import error: {_import_error}")


def _require():
    if _C is None:
        raise RuntimeError(
            f"room_amd._C not available (CPU-only environment?): {_import_error}")
    return _C


def rmsnorm(out: torch.Tensor, x: torch.Tensor, weight: torch.Tensor,
            eps: float = 1e-6) -> torch.Tensor:
    _require().rmsnorm(out, x, weight, eps)
    return out


def fused_add_rmsnorm(out: torch.Tensor, residual: torch.Tensor, x: torch.Tensor,
                      weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    _require().fused_add_rmsnorm(out, residual, x, weight, eps)
    return out


def qk_norm_rope(q: torch.Tensor, k: torch.Tensor, q_w: torch.Tensor,
                 k_w: torch.Tensor, cos_t: torch.Tensor, sin_t: torch.Tensor,
                 positions: torch.Tensor, n_qheads: int, n_kvheads: int,
                 head_dim: int = 128, eps: float = 1e-6) -> None:
    _require().qk_norm_rope(q, k, q_w, k_w, cos_t, sin_t, positions,
                            n_qheads, n_kvheads, head_dim, eps)


def qk_rope_write_kv(qkv: torch.Tensor, n_qheads: int,
                     kcache: torch.Tensor, vcache: torch.Tensor,
                     q_w: torch.Tensor, k_w: torch.Tensor,
                     cos_t: torch.Tensor, sin_t: torch.Tensor,
                     block_table: torch.Tensor, seq_ids: torch.Tensor,
                     positions: torch.Tensor, eps: float = 1e-6) -> None:
    """Fused per-head QK RMSNorm + RoPE + paged KV scatter, operating directly
    on the packed [T, (Hq+2Hk)*D] projection output (no q/k/v copies)."""
    _require().qk_rope_write_kv(qkv, kcache, vcache, q_w, k_w, cos_t,
                                sin_t, block_table, seq_ids, positions,
                                n_qheads, eps)


def silu_mul(out: torch.Tensor, gateup: torch.Tensor) -> torch.Tensor:
    _require().silu_mul(out, gateup)
    return out


def paged_attention(out: torch.Tensor, q: torch.Tensor, kcache: torch.Tensor,
                    vcache: torch.Tensor, block_table: torch.Tensor,
                    seq_ids: torch.Tensor, q_pos: torch.Tensor,
                    scale: float) -> torch.Tensor:
    _require().paged_attention(out, q, kcache, vcache, block_table, seq_ids,
                               q_pos, scale)
    return out


def write_kv(kcache: torch.Tensor, vcache: torch.Tensor, k: torch.Tensor,
             v: torch.Tensor, block_table: torch.Tensor, seq_ids: torch.Tensor,
             q_pos: torch.Tensor) -> None:
    _require().write_kv(kcache, vcache, k, v, block_table, seq_ids, q_pos)


def attn_nsplits() -> int:
    """Decode split-KV NSPLITS (compiled into the kernel)."""
    if _C is not None and hasattr(_C, "attn_nsplits"):
        return int(_C.attn_nsplits())
    return 32


def paged_attention_split(out: torch.Tensor, q: torch.Tensor, kcache: torch.Tensor,
                          vcache: torch.Tensor, block_table: torch.Tensor,
                          seq_ids: torch.Tensor, q_pos: torch.Tensor,
                          part: torch.Tensor, part_ml: torch.Tensor,
                          scale: float, splits: int = 32) -> torch.Tensor:
    """Split-KV flash-decode: part [T, Hq, NSPLITS_MAX, 128] f32, part_ml
    [.., 2]; `splits` (32 short / 64 long contexts) sets the launch grid —
    shorter per-WG chunk chains win once the serial span passes ~8k."""
    _require().paged_attention_split(out, q, kcache, vcache, block_table,
                                     seq_ids, q_pos, part, part_ml, scale,
                                     splits)
    return out


def gemv(y: torch.Tensor, x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """y[B,N] = x[B,H] @ w[N,H]^T for B<=8 (decode projections)."""
    _require().gemv(y, x, w)
    return y


def paged_attention_splitk(part: torch.Tensor, part_ml: torch.Tensor,
                           q: torch.Tensor, kcache: torch.Tensor,
                           vcache: torch.Tensor, block_table: torch.Tensor,
                           seq_ids: torch.Tensor, q_pos: torch.Tensor,
                           scale: float, o_zero: torch.Tensor,
                           splits: int = 32) -> None:
    """Split-KV decode attention, partials only (no bf16 merge output); zeros
    `o_zero` (the attn_merge_o f32 accumulator) as a side job."""
    _require().paged_attention_splitk(part, part_ml, q, kcache, vcache,
                                      block_table, seq_ids, q_pos, scale,
                                      o_zero, splits)


def attn_merge_o(o_accum: torch.Tensor, part: torch.Tensor,
                 part_ml: torch.Tensor, wo: torch.Tensor,
                 splits: int = 32) -> torch.Tensor:
    """Fused split-partial merge + O-projection: o_accum[B, H] (f32,
    pre-zeroed) += Wo @ merged-attention. Replaces the standalone merge
    kernel + O GEMV on the decode path."""
    _require().attn_merge_o(o_accum, part, part_ml, wo, splits)
    return o_accum


def router_addnorm(y: torch.Tensor, x: torch.Tensor, delta: torch.Tensor,
                   x_out: torch.Tensor, xn_out: torch.Tensor,
                   gamma: torch.Tensor, w: torch.Tensor,
                   eps: float = 1e-6) -> torch.Tensor:
    """x_out = x + delta (f32); xn_out = rmsnorm(x_out)·γ; y = xn @ w^T
    (f32 router logits). One kernel replaces fused_add_rmsnorm + the H-split
    router partial GEMV on the decode path. x_out must not alias x."""
    _require().router_addnorm(y, x, delta, x_out, xn_out, gamma, w, eps)
    return y


def flash_prefill(out: torch.Tensor, q: torch.Tensor, kcache: torch.Tensor,
                  vcache: torch.Tensor, block_table: torch.Tensor,
                  seq_ids: torch.Tensor, q_pos: torch.Tensor,
                  tile_desc: torch.Tensor, scale: float) -> torch.Tensor:
    """MFMA flash prefill; tile_desc [G,2] = (row0, n_tokens≤16), one sequence
    per tile (build with build_qtile_desc)."""
    _require().flash_prefill(out, q, kcache, vcache, block_table, seq_ids,
                             q_pos, tile_desc, scale)
    return out


def build_qtile_desc(segments: list, device) -> torch.Tensor:
    """Q-tile descriptors from host-known prefill segments [(row0, count), ...]
    (one sequence per segment) — no device sync. Tile rows match the
    flash_prefill kernel's FP_QTOK."""
    qt = int(_C.flash_prefill_qtile()) if (
        _C is not None and hasattr(_C, "flash_prefill_qtile")) else 32
    desc = []
    for row0, count in segments:
        r = row0
        while r < row0 + count:
            n = min(qt, row0 + count - r)
            desc.append((r, n))
            r += n
    if not desc:
        return torch.zeros(0, 2, dtype=torch.int32, device=device)
    return torch.tensor(desc, dtype=torch.int32, device=device)


def gemv_addnorm(y: torch.Tensor, x: torch.Tensor, delta: torch.Tensor,
                 x_out: torch.Tensor, gamma: torch.Tensor, w: torch.Tensor,
                 eps: float = 1e-6) -> torch.Tensor:
    """y = rmsnorm(x + delta)·γ @ w^T; x_out = x + delta (pass delta with
    numel 0 for a plain norm+gemv). Decode-path fusion (B ≤ 8)."""
    _require().gemv_addnorm(y, x, delta, x_out, gamma, w, eps)
    return y


def moe_router(logits: torch.Tensor, k: int) -> tuple[torch.Tensor, torch.Tensor]:
    T = logits.size(0)
    ids = torch.empty(T, k, dtype=torch.int32, device=logits.device)
    w = torch.empty(T, k, dtype=torch.float32, device=logits.device)
    _require().moe_router(ids, w, logits, k)
    return ids, w


def router_gemv_topk(x: torch.Tensor, wr: torch.Tensor,
                     k: int) -> tuple[torch.Tensor, torch.Tensor]:
    """Decode router: 4-way H-split partial GEMV (wide grid) + top-k kernel
    that sums the partials on load. Replaces gemv + moe_router."""
    T = x.size(0)
    ids = torch.empty(T, k, dtype=torch.int32, device=x.device)
    w = torch.empty(T, k, dtype=torch.float32, device=x.device)
    _require().router_gemv_topk(ids, w, x, wr, k)
    return ids, w


def router_topk(x: torch.Tensor, wr: torch.Tensor,
                k: int) -> tuple[torch.Tensor, torch.Tensor]:
    """Fused decode router: logits = x @ wr.T → softmax → top-k → renorm."""
    T = x.size(0)
    ids = torch.empty(T, k, dtype=torch.int32, device=x.device)
    w = torch.empty(T, k, dtype=torch.float32, device=x.device)
    _require().router_topk(ids, w, x, wr, k)
    return ids, w


def moe_gemv_h(h: torch.Tensor, x: torch.Tensor, w13: torch.Tensor,
               pair_token: torch.Tensor, pair_expert: torch.Tensor,
               out_zero: torch.Tensor | None = None) -> torch.Tensor:
    """Gate/up GEMV + silu-mul per (token, expert) pair. out_zero: optional
    f32 tensor zero-filled as a side job (the down kernel's accumulator)."""
    if out_zero is None:
        out_zero = torch.empty(0, dtype=torch.float32, device=h.device)
    _require().moe_gemv_h(h, x, w13, pair_token, pair_expert, out_zero)
    return h


def moe_gemv_down(out: torch.Tensor, h: torch.Tensor, w2: torch.Tensor,
                  pair_w: torch.Tensor, pair_token: torch.Tensor,
                  pair_expert: torch.Tensor) -> torch.Tensor:
    _require().moe_gemv_down(out, h, w2, pair_w, pair_token, pair_expert)
    return out


def moe_grouped_gemm(out: torch.Tensor, x: torch.Tensor, w: torch.Tensor,
                     pair_token: torch.Tensor, tile_desc: torch.Tensor) -> torch.Tensor:
    _require().moe_grouped_gemm(out, x, w, pair_token, tile_desc)
    return out


def moe_grouped_gemm128(out: torch.Tensor, x: torch.Tensor, w: torch.Tensor,
                        pair_token: torch.Tensor, tile_desc: torch.Tensor,
                        bm: int = 128) -> torch.Tensor:
    """Grouped MFMA GEMM: expert weight panels read once per BM-row m-tile
    (bm must match the tile_desc's build bm; 256 halves W traffic when
    experts average ≥128 pairs)."""
    _require().moe_grouped_gemm128(out, x, w, pair_token, tile_desc, bm)
    return out


def moe_combine(out: torch.Tensor, z: torch.Tensor, pair_w: torch.Tensor,
                pair_token: torch.Tensor) -> torch.Tensor:
    _require().moe_combine(out, z, pair_w, pair_token)
    return out


def moe_gemv_dedup(out: torch.Tensor, x: torch.Tensor, w13: torch.Tensor,
                   w2: torch.Tensor, topk_ids: torch.Tensor,
                   topk_w: torch.Tensor, scratch: dict) -> torch.Tensor:
    """Expert-deduped decode MoE: weight rows read once per active expert.
    scratch holds persistent buffers keyed by (E, I): counts/tok_list/w_list/h."""
    E = w13.size(0)
    I = w2.size(2)
    key = (E, I)
    if key not in scratch:
        dev = x.device
        scratch[key] = {
            "counts": torch.zeros(E, dtype=torch.int32, device=dev),
            "tok_list": torch.zeros(E, 64, dtype=torch.int32, device=dev),
            "w_list": torch.zeros(E, 64, dtype=torch.float32, device=dev),
            "h": torch.zeros(E * 64, I, dtype=torch.bfloat16, device=dev),
            "active": torch.full((512,), -1, dtype=torch.int32, device=dev),
        }
    sc = scratch[key]
    _require().moe_gemv_dedup(out, x, w13, w2, topk_ids.contiguous(),
                              topk_w.contiguous(), sc["counts"],
                              sc["tok_list"], sc["w_list"], sc["h"],
                              sc["active"])
    return out


def moe_build_desc_device(pair_expert_sorted: torch.Tensor, num_experts: int,
                          bm: int = 128) -> torch.Tensor:
    """Sync-free [GMAX,3] (expert,row0,msize) descriptors; GMAX computed from
    the host-known pair count, empty tiles zero-sized."""
    P = pair_expert_sorted.numel()
    gmax = num_experts + (P + bm - 1) // bm
    # NOT torch.bincount: its nbins probe (.max().item()) host-syncs, which
    # aborts hipGraph capture — scatter_add_ is sync-free
    counts = torch.zeros(num_experts, dtype=torch.int64,
                         device=pair_expert_sorted.device)
    counts.scatter_add_(0, pair_expert_sorted.long(),
                        torch.ones(P, dtype=torch.int64,
                                   device=pair_expert_sorted.device))
    desc = torch.empty(gmax, 3, dtype=torch.int32,
                       device=pair_expert_sorted.device)
    _require().moe_build_desc(desc, counts, bm)
    return desc


def dense_grouped_gemm(y: torch.Tensor, x: torch.Tensor, w: torch.Tensor,
                       desc: torch.Tensor, pair_token: torch.Tensor
                       ) -> torch.Tensor:
    """Capture-safe dense GEMM y[T,N] = x[T,H] @ w[N,H]^T via the grouped
    MFMA kernel with a single expert. hipBLASLt is faster eager but its
    internal calls abort hipGraph capture; inside captured prefill forwards
    this path runs instead (desc/pair_token prebuilt per T by the model)."""
    _require().moe_grouped_gemm128(y, x, w.unsqueeze(0), pair_token, desc, 128)
    return y


def moe_combine_gather(out: torch.Tensor, z: torch.Tensor, topk_w: torch.Tensor,
                       inv_order: torch.Tensor) -> torch.Tensor:
    _require().moe_combine_gather(out, z, topk_w, inv_order)
    return out


def sample_tokens(logits: torch.Tensor, seeds: torch.Tensor, top_k: int = 40,
                  temperature: float = 0.7, top_p: float = 0.95) -> torch.Tensor:
    out = torch.empty(logits.size(0), dtype=torch.int32, device=logits.device)
    # v3: register-resident top-8 scan + tournament (4× the v2 LDS variant)
    _require().sample_tokens_v4(out, logits, seeds, top_k, temperature, top_p)
    return out


def vs_topk(mat: torch.Tensor, query: torch.Tensor, k: int,
            nblocks: int = 512) -> tuple[torch.Tensor, torch.Tensor]:
    """Cosine top-k over an HBM-resident [N, 384] bf16 matrix (rows L2-normed)."""
    dev = mat.device
    nblocks = min(nblocks, max(1, (mat.size(0) + 255) // 256))
    cand_v = torch.empty(nblocks, k, dtype=torch.float32, device=dev)
    cand_i = torch.empty(nblocks, k, dtype=torch.int32, device=dev)
    out_v = torch.empty(k, dtype=torch.float32, device=dev)
    out_i = torch.empty(k, dtype=torch.int64, device=dev)
    _require().vs_topk(out_v, out_i, cand_v, cand_i, mat, query, k)
    return out_v, out_i


def build_moe_tile_desc(pair_expert_sorted: torch.Tensor, n_tiles_n: int,
                        num_experts: int, bm: int = 16) -> torch.Tensor:
    """Tile descriptors for the grouped MFMA GEMM: one row
    (expert, row_start, m_size, n_tile) per (M-tile × N-tile). Fully
    vectorized on the device — no host sync (runs once per layer per step)."""
    dev = pair_expert_sorted.device
    counts = torch.zeros(num_experts, dtype=torch.int64, device=dev)
    counts.scatter_add_(0, pair_expert_sorted.long(),
                        torch.ones(pair_expert_sorted.numel(),
                                   dtype=torch.int64, device=dev))
    mtiles = (counts + bm - 1) // bm                      # [E]
    e_ids = torch.repeat_interleave(
        torch.arange(num_experts, device=dev), mtiles)    # [G_m]
    starts = torch.cumsum(counts, 0) - counts             # row offset per expert
    mt_off = torch.cumsum(mtiles, 0) - mtiles             # first tile idx per expert
    g = e_ids.numel()
    tile_in_e = torch.arange(g, device=dev) - mt_off[e_ids]
    row_start = starts[e_ids] + bm * tile_in_e
    m_size = torch.minimum(counts[e_ids] - bm * tile_in_e,
                           torch.full_like(row_start, bm))
    desc_m = torch.stack([e_ids, row_start, m_size], dim=1)  # [G_m, 3]
    # cross-product with the N tiles
    desc = desc_m.repeat_interleave(n_tiles_n, dim=0)
    nt = torch.arange(n_tiles_n, device=dev).repeat(g).unsqueeze(1)
    return torch.cat([desc, nt], dim=1).int().contiguous()
