"""Pure-PyTorch fp32 reference implementations of every HIP kernel.

Used ONLY by numerics tests (tests/test_kernels_gpu.py compares each CDNA4
kernel against these on random inputs) and never on the GPU compute path.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F


def rmsnorm_ref(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    xf = x.float()
    scale = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * scale * w.float()).to(x.dtype)


def fused_add_rmsnorm_ref(residual: torch.Tensor, x: torch.Tensor, w: torch.Tensor,
                          eps: float = 1e-6) -> tuple[torch.Tensor, torch.Tensor]:
    new_res = (residual.float() + x.float()).to(x.dtype)
    return rmsnorm_ref(new_res, w, eps), new_res


def rope_tables(max_pos: int, head_dim: int, theta: float) -> tuple[torch.Tensor, torch.Tensor]:
    half = head_dim // 2
    inv_freq = 1.0 / (theta ** (torch.arange(0, half, dtype=torch.float64) / half))
    pos = torch.arange(max_pos, dtype=torch.float64)
    ang = torch.outer(pos, inv_freq)
    return ang.cos().float(), ang.sin().float()


def qk_norm_rope_ref(q: torch.Tensor, k: torch.Tensor, q_w: torch.Tensor,
                     k_w: torch.Tensor, cos_t: torch.Tensor, sin_t: torch.Tensor,
                     positions: torch.Tensor, eps: float = 1e-6
                     ) -> tuple[torch.Tensor, torch.Tensor]:
    """q: [T, Hq, D], k: [T, Hk, D]; per-head RMSNorm then rotate-half RoPE."""

    def one(x, w):
        xf = x.float()
        scale = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
        xn = xf * scale * w.float()
        half = x.size(-1) // 2
        c = cos_t[positions.long()].unsqueeze(1)  # [T,1,half]
        s = sin_t[positions.long()].unsqueeze(1)
        x1, x2 = xn[..., :half], xn[..., half:]
        out = torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)
        return out.to(x.dtype)

    return one(q, q_w), one(k, k_w)


def silu_mul_ref(gateup: torch.Tensor) -> torch.Tensor:
    inter = gateup.size(-1) // 2
    g, u = gateup[..., :inter].float(), gateup[..., inter:].float()
    return (F.silu(g) * u).to(gateup.dtype)


def paged_attention_ref(q: torch.Tensor, kcache: torch.Tensor, vcache: torch.Tensor,
                        block_table: torch.Tensor, seq_ids: torch.Tensor,
                        q_pos: torch.Tensor, scale: float) -> torch.Tensor:
    """q: [T, Hq, D]; caches [NB, Hk, BS, D]. Gathers each token's KV prefix
    and does fp32 SDPA per token."""
    T, Hq, D = q.shape
    Hk, BS = kcache.size(1), kcache.size(2)
    group = Hq // Hk
    out = torch.zeros_like(q, dtype=torch.float32)
    for t in range(T):
        seq = int(seq_ids[t])
        bound = int(q_pos[t]) + 1
        nblocks = (bound + BS - 1) // BS
        blocks = block_table[seq, :nblocks].long()
        k = kcache[blocks].transpose(0, 1).reshape(Hk, nblocks * BS, D)[:, :bound]
        v = vcache[blocks].transpose(0, 1).reshape(Hk, nblocks * BS, D)[:, :bound]
        for hq in range(Hq):
            hk = hq // group
            scores = (q[t, hq].float() @ k[hk].float().T) * scale
            p = torch.softmax(scores, dim=-1)
            out[t, hq] = p @ v[hk].float()
    return out.to(q.dtype)


def moe_router_ref(logits: torch.Tensor, k: int) -> tuple[torch.Tensor, torch.Tensor]:
    probs = torch.softmax(logits.float(), dim=-1)
    topw, topi = probs.topk(k, dim=-1)
    topw = topw / topw.sum(-1, keepdim=True)
    return topi.int(), topw.float()


def moe_ref(x: torch.Tensor, w13: torch.Tensor, w2: torch.Tensor,
            topk_ids: torch.Tensor, topk_w: torch.Tensor) -> torch.Tensor:
    """x: [T,H]; w13: [E,2I,H]; w2: [E,H,I]; returns [T,H] fp32."""
    T, H = x.shape
    I = w2.size(2)
    out = torch.zeros(T, H, dtype=torch.float32, device=x.device)
    for t in range(T):
        for j in range(topk_ids.size(1)):
            e = int(topk_ids[t, j])
            gu = x[t].float() @ w13[e].float().T  # [2I]
            h = F.silu(gu[:I]) * gu[I:]
            # the kernels round h to bf16 between stages; mirror that
            h = h.to(torch.bfloat16).float()
            z = h @ w2[e].float().T  # [H]
            out[t] += float(topk_w[t, j]) * z
    return out


def vs_topk_ref(mat: torch.Tensor, query: torch.Tensor, k: int
                ) -> tuple[torch.Tensor, torch.Tensor]:
    scores = mat.float() @ query.float()
    v, i = scores.topk(k)
    return v, i
