"""Qwen3-MoE (qwen3-coder-30b = Qwen3-30B-A3B shape) — MI355X-native forward.

The reference pinned `qwen3-coder:30b` served by an external Ollama sidecar
(src/shared/local-model.ts:3-5); here the model runs in-process: weights are
bf16 tensors resident in HBM3E, and the hot ops are the CDNA4 HIP kernels in
room_amd/ops (RMSNorm, fused QK-norm+RoPE+KV-scatter, split-KV paged
attention, MFMA flash prefill, H-split router + top-k, MoE pair GEMV /
grouped MFMA GEMM, BN-specialized dense GEMV, two-stage sampling). At decode
every projection (QKV/O/router/lm_head) runs on the hand-written GEMVs;
prefill's large GEMMs go through hipBLASLt via torch.nn.functional.linear.

Inference-only (no autograd); a training path is out of scope for the
reference's semantics (it never trains).
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn.functional as F

from .. import ops
from ..ops.reference import rope_tables


@dataclass
class Qwen3MoEConfig:
    vocab_size: int = 151936
    hidden_size: int = 2048
    num_layers: int = 48
    num_q_heads: int = 32
    num_kv_heads: int = 4
    head_dim: int = 128
    num_experts: int = 128
    num_experts_per_tok: int = 8
    moe_intermediate_size: int = 768
    rms_eps: float = 1e-6
    rope_theta: float = 10_000_000.0
    max_position: int = 16384
    # GEMV→grouped-GEMM switchover: below this many tokens the decode GEMV
    # path wins (per-pair weight streaming); above it, grouped MFMA GEMM.
    moe_grouped_threshold: int = 16

    @staticmethod
    def qwen3_coder_30b() -> "Qwen3MoEConfig":
        return Qwen3MoEConfig()

    @staticmethod
    def tiny(vocab: int = 4096) -> "Qwen3MoEConfig":
        """Small config for smoke tests / CPU-free validation on one GPU call."""
        return Qwen3MoEConfig(vocab_size=vocab, hidden_size=512, num_layers=4,
                              num_q_heads=8, num_kv_heads=1, head_dim=128,
                              num_experts=16, num_experts_per_tok=4,
                              moe_intermediate_size=256, max_position=4096)


class Qwen3MoELayer:
    def __init__(self, cfg: Qwen3MoEConfig, device: torch.device, gen: torch.Generator):
        h, d = cfg.hidden_size, cfg.head_dim
        qdim = cfg.num_q_heads * d
        kvdim = cfg.num_kv_heads * d
        std = 0.02

        def rnd(*shape):
            return torch.empty(*shape, dtype=torch.bfloat16, device=device).normal_(
                0, std, generator=gen)

        self.input_norm_w = torch.ones(h, dtype=torch.bfloat16, device=device)
        self.post_attn_norm_w = torch.ones(h, dtype=torch.bfloat16, device=device)
        self.q_norm_w = torch.ones(d, dtype=torch.bfloat16, device=device)
        self.k_norm_w = torch.ones(d, dtype=torch.bfloat16, device=device)
        self.wqkv = rnd(qdim + 2 * kvdim, h)      # fused QKV projection
        self.wo = rnd(h, qdim)
        self.router_w = rnd(cfg.num_experts, h)
        self.w13 = rnd(cfg.num_experts, 2 * cfg.moe_intermediate_size, h)
        self.w2 = rnd(cfg.num_experts, h, cfg.moe_intermediate_size)


class Qwen3MoEModel:
    """Flat-tensor model (no nn.Module overhead on the decode path)."""

    def __init__(self, cfg: Qwen3MoEConfig, device: str | torch.device = "cuda",
                 seed: int = 1234):
        self.cfg = cfg
        self.device = torch.device(device)
        gen = torch.Generator(device=self.device)
        gen.manual_seed(seed)
        h = cfg.hidden_size
        self.embed = torch.empty(cfg.vocab_size, h, dtype=torch.bfloat16,
                                 device=self.device).normal_(0, 0.02, generator=gen)
        self.layers = [Qwen3MoELayer(cfg, self.device, gen)
                       for _ in range(cfg.num_layers)]
        self.final_norm_w = torch.ones(h, dtype=torch.bfloat16, device=self.device)
        self.lm_head = torch.empty(cfg.vocab_size, h, dtype=torch.bfloat16,
                                   device=self.device).normal_(0, 0.02, generator=gen)
        import os as _os
        self._moe_scratch: dict = {}   # dedup-GEMV persistent buffers
        self.moe_dedup = _os.environ.get('ROOMAMD_MOE_DEDUP') == '1'
        # fused decode kernels (attn_merge_o / router_addnorm): measured
        # SLOWER than the two-kernel paths at B=5 (profiles/PERF_NOTES.md
        # round-2 negative results — attn_merge_o re-reads the split partials
        # once per output tile, 16× redundant HBM traffic, 31.8 µs vs the
        # 19.4 µs merge+GEMV pair; router_addnorm 10.9 vs 9.75 on a
        # latency-starved 32-WG grid). Kept behind env for the record.
        self.fuse_o = _os.environ.get('ROOMAMD_FUSE_O') == '1'
        self.fuse_router = _os.environ.get('ROOMAMD_FUSE_ROUTER') == '1'
        # grouped-GEMM m-tile: 256 halves expert-panel re-reads when experts
        # average >=128 pairs (prefill chunks >=2k tokens)
        self.moe_bm = 256 if _os.environ.get('ROOMAMD_MOE_BM') == '256' else 128
        # decode split-KV band (32 short / 64 long contexts); the engine sets
        # this per decode step (and per captured graph band)
        self.attn_splits = 32
        # capture-safe GEMMs: hipBLASLt aborts hipGraph capture, so captured
        # prefill forwards route dense projections through the grouped MFMA
        # kernel (single-expert). The engine flips this around capture.
        self.capture_gemm = False
        self._dense_desc: dict = {}   # T -> (desc, pair_token) for E=1 GEMMs
        self._dense_wpad: dict = {}   # id(w) -> N-padded weight (N % 64 != 0)
        cos_t, sin_t = rope_tables(cfg.max_position, cfg.head_dim, cfg.rope_theta)
        self.cos_t = cos_t.to(self.device)
        self.sin_t = sin_t.to(self.device)
        self.scale = cfg.head_dim ** -0.5

    def num_params(self) -> int:
        cfg = self.cfg
        per_layer = (self.layers[0].wqkv.numel() + self.layers[0].wo.numel()
                     + self.layers[0].router_w.numel() + self.layers[0].w13.numel()
                     + self.layers[0].w2.numel())
        return self.embed.numel() + self.lm_head.numel() + cfg.num_layers * per_layer

    def _dense(self, x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
        """Dense projection: hipBLASLt eager; grouped MFMA when capturing."""
        if not self.capture_gemm:
            return F.linear(x, w)
        T = x.size(0)
        dp = self._dense_desc.get(T)
        if dp is None:
            pe = torch.zeros(T, dtype=torch.int32, device=x.device)
            desc = ops.moe_build_desc_device(pe, 1, bm=128)
            pt = torch.arange(T, dtype=torch.int32, device=x.device)
            dp = self._dense_desc[T] = (desc, pt)
        N = w.size(0)
        if N % 64:  # e.g. tiny-config router (16 experts): pad N, slice out
            wp = self._dense_wpad.get(id(w))
            if wp is None:
                wp = torch.zeros((N + 63) // 64 * 64, w.size(1),
                                 dtype=w.dtype, device=w.device)
                wp[:N] = w
                self._dense_wpad[id(w)] = wp
            w = wp
        y = torch.empty(T, w.size(0), dtype=torch.bfloat16, device=x.device)
        ops.dense_grouped_gemm(y, x, w, dp[0], dp[1])
        return y[:, :N] if N != w.size(0) else y

    @torch.inference_mode()
    def forward(self, tokens: torch.Tensor, seq_ids: torch.Tensor,
                q_pos: torch.Tensor, block_table: torch.Tensor,
                kcaches: list[torch.Tensor], vcaches: list[torch.Tensor],
                logits_rows: torch.Tensor | None = None,
                qtile_desc: torch.Tensor | None = None) -> torch.Tensor:
        """tokens/seq_ids/q_pos: [T] on device. kcaches/vcaches: one pair per
        layer. logits_rows: row indices to compute logits for (default: all).
        Returns fp32 logits [R, vocab]."""
        cfg = self.cfg
        T = tokens.numel()
        dev = self.device
        # decode-mode (T ≤ 8): dense projections use the hand-written GEMV
        # (hipBLASLt ~1.1 TB/s on M≤8 skinny shapes) and attention uses the
        # split-KV flash-decode kernels so the tiny grid still fills the chip.
        decode = T <= 8
        qdim = cfg.num_q_heads * cfg.head_dim
        kvdim = cfg.num_kv_heads * cfg.head_dim
        x = self.embed[tokens.long()]              # [T, H] bf16 (residual stream)
        hbuf = torch.empty_like(x)                 # normed activations
        moe_out = None                             # pending delta for fused add
        # layer-invariant MoE pair/token indices (was re-built 48×/step: an
        # arange+repeat_interleave kernel pair per layer in the decode graph)
        K = cfg.num_experts_per_tok
        self._pair_token_flat = torch.arange(
            T, device=dev, dtype=torch.int32).repeat_interleave(K)
        self._arangeP = torch.arange(T * K, device=dev, dtype=torch.int32)
        if decode:
            qkv = torch.empty(T, qdim + 2 * kvdim, dtype=torch.bfloat16, device=dev)
            x_alt = torch.empty_like(x)     # residual ping-pong for fused norms
            empty_delta = torch.empty(0, dtype=torch.float32, device=dev)
            nsp = ops.attn_nsplits()
            part = torch.empty(T, cfg.num_q_heads, nsp, cfg.head_dim,
                               dtype=torch.float32, device=dev)
            part_ml = torch.empty(T, cfg.num_q_heads, nsp, 2, dtype=torch.float32,
                                  device=dev)
            # fused decode path (attn_merge_o / router_addnorm): collapses
            # merge+O-GEMV and add-norm+router-partial into single kernels
            fuse_o, fuse_router = self.fuse_o, self.fuse_router
            if fuse_o:
                o_accum = torch.empty(T, cfg.hidden_size, dtype=torch.float32,
                                      device=dev)
            else:
                obuf = torch.empty(T, cfg.hidden_size, dtype=torch.bfloat16,
                                   device=dev)
            if fuse_router:
                rlogits = torch.empty(T, cfg.num_experts, dtype=torch.float32,
                                      device=dev)

        for li, layer in enumerate(self.layers):
            # --- attention block
            if decode and T <= 2:
                # fused add(moe_prev)+RMSNorm+QKV-GEMV: one kernel replaces
                # fused_add_rmsnorm + gemv. Wins at B≤2 where the removed
                # launch/drain bubble dominates (5.7 → 5.2 ms/step at B=1);
                # at B≥3 the cooperative norm passes cost more than the bubble
                # (measured 9.1 vs 8.8 at B=5) so the unfused path is used.
                if moe_out is None:
                    ops.gemv_addnorm(qkv, x, empty_delta, x, layer.input_norm_w,
                                     layer.wqkv, cfg.rms_eps)
                else:
                    ops.gemv_addnorm(qkv, x, moe_out, x_alt, layer.input_norm_w,
                                     layer.wqkv, cfg.rms_eps)
                    x, x_alt = x_alt, x
            elif decode:
                if moe_out is None:
                    ops.rmsnorm(hbuf, x, layer.input_norm_w, cfg.rms_eps)
                else:
                    ops.fused_add_rmsnorm(hbuf, x, moe_out, layer.input_norm_w,
                                          cfg.rms_eps)
                ops.gemv(qkv, hbuf, layer.wqkv)
            else:
                if moe_out is None:
                    ops.rmsnorm(hbuf, x, layer.input_norm_w, cfg.rms_eps)
                else:
                    ops.fused_add_rmsnorm(hbuf, x, moe_out, layer.input_norm_w,
                                          cfg.rms_eps)
                qkv = self._dense(hbuf, layer.wqkv)
            # q stays a strided view into qkv (the attention kernels take a
            # row stride; .contiguous() copies profiled at 0.67 ms/step)
            q = qkv[:, :qdim].view(T, cfg.num_q_heads, cfg.head_dim)
            ops.qk_rope_write_kv(qkv, cfg.num_q_heads, kcaches[li], vcaches[li],
                                 layer.q_norm_w, layer.k_norm_w, self.cos_t,
                                 self.sin_t, block_table, seq_ids, q_pos,
                                 cfg.rms_eps)
            if decode:
                if fuse_o:
                    # split partials (+zero the O accumulator as a side job),
                    # then one kernel merges per-head and projects through Wo
                    # with f32 atomics — no bf16 attn round-trip
                    ops.paged_attention_splitk(part, part_ml, q, kcaches[li],
                                               vcaches[li], block_table,
                                               seq_ids, q_pos, self.scale,
                                               o_accum,
                                               splits=self.attn_splits)
                    ops.attn_merge_o(o_accum, part, part_ml, layer.wo,
                                     splits=self.attn_splits)
                    o = o_accum                   # f32 delta
                else:
                    attn = torch.empty(T, cfg.num_q_heads, cfg.head_dim,
                                       dtype=torch.bfloat16, device=dev)
                    ops.paged_attention_split(attn, q, kcaches[li],
                                              vcaches[li], block_table,
                                              seq_ids, q_pos, part, part_ml,
                                              self.scale,
                                              splits=self.attn_splits)
                    ops.gemv(obuf, attn.reshape(T, qdim), layer.wo)
                    o = obuf
                # --- MoE block
                if fuse_router:
                    # add+norm+router logits in one kernel
                    ops.router_addnorm(rlogits, x, o, x_alt, hbuf,
                                       layer.post_attn_norm_w, layer.router_w,
                                       cfg.rms_eps)
                    x, x_alt = x_alt, x
                    topk_ids, topk_w = ops.moe_router(rlogits,
                                                      cfg.num_experts_per_tok)
                else:
                    ops.fused_add_rmsnorm(hbuf, x, o, layer.post_attn_norm_w,
                                          cfg.rms_eps)
                    topk_ids, topk_w = ops.router_gemv_topk(
                        hbuf, layer.router_w, cfg.num_experts_per_tok)
                moe_out = self._moe(hbuf, layer, topk_ids, topk_w)
                continue
            attn = torch.empty(T, cfg.num_q_heads, cfg.head_dim,
                               dtype=torch.bfloat16, device=dev)
            if qtile_desc is not None:
                ops.flash_prefill(attn, q, kcaches[li], vcaches[li], block_table,
                                  seq_ids, q_pos, qtile_desc, self.scale)
                o = self._dense(attn.reshape(T, qdim), layer.wo)
            else:
                ops.paged_attention(attn, q, kcaches[li], vcaches[li], block_table,
                                    seq_ids, q_pos, self.scale)
                o = F.linear(attn.reshape(T, qdim), layer.wo)

            # --- MoE block (prefill; decode handled above)
            ops.fused_add_rmsnorm(hbuf, x, o, layer.post_attn_norm_w, cfg.rms_eps)
            router_logits = self._dense(hbuf, layer.router_w).float()
            topk_ids, topk_w = ops.moe_router(router_logits,
                                              cfg.num_experts_per_tok)
            # f32 accumulator feeds fused_add_rmsnorm directly (templated
            # input dtype — skips a cast kernel per layer)
            moe_out = self._moe(hbuf, layer, topk_ids, topk_w)

        # final residual add + norm; skip lm_head entirely for logits-free
        # prefill chunks (an empty logits_rows means "no sampling this chunk")
        if logits_rows is not None and logits_rows.numel() == 0:
            return torch.empty(0, cfg.vocab_size, dtype=torch.float32,
                               device=x.device)
        ops.fused_add_rmsnorm(hbuf, x, moe_out, self.final_norm_w, cfg.rms_eps)
        sel = hbuf if logits_rows is None else hbuf[logits_rows.long()]
        if sel.size(0) <= 8:
            # B-specialized GEMV (the earlier 2.5 TB/s vs hipBLASLt 3 TB/s
            # comparison predates the exact-batch template)
            logits = torch.empty(sel.size(0), cfg.vocab_size,
                                 dtype=torch.float32, device=x.device)
            ops.gemv(logits, sel.contiguous(), self.lm_head)
        else:
            logits = F.linear(sel, self.lm_head).float()
        return logits

    def _moe(self, hbuf: torch.Tensor, layer: Qwen3MoELayer,
             topk_ids: torch.Tensor, topk_w: torch.Tensor) -> torch.Tensor:
        cfg = self.cfg
        T = hbuf.size(0)
        K = cfg.num_experts_per_tok
        H, I = cfg.hidden_size, cfg.moe_intermediate_size

        if T < cfg.moe_grouped_threshold:
            out = torch.empty(T, H, dtype=torch.float32, device=hbuf.device)
            if self.moe_dedup:
                out.zero_()
                # expert-deduped GEMV (weight rows once per active expert);
                # measured slower than the pair path at B<=8 on random routing
                # -- kept behind ROOMAMD_MOE_DEDUP=1 pending a win
                ops.moe_gemv_dedup(out, hbuf, layer.w13, layer.w2, topk_ids,
                                   topk_w, self._moe_scratch)
                return out
            # decode path: GEMV per (token, expert) pair
            pair_token = self._pair_token_flat
            pair_expert = topk_ids.flatten().contiguous()
            pair_w = topk_w.flatten().contiguous()
            P = pair_token.numel()
            h = torch.empty(P, I, dtype=torch.bfloat16, device=hbuf.device)
            ops.moe_gemv_h(h, hbuf, layer.w13, pair_token, pair_expert,
                           out_zero=out)
            ops.moe_gemv_down(out, h, layer.w2, pair_w, pair_token, pair_expert)
            return out

        # prefill path: sort pairs by expert, grouped MFMA GEMMs
        flat_expert = topk_ids.flatten()
        order = torch.argsort(flat_expert)
        pair_expert = flat_expert[order].int().contiguous()
        pair_token = self._pair_token_flat[order].contiguous()
        P = pair_token.numel()
        inv_order = torch.empty_like(order)
        inv_order[order] = torch.arange(P, device=hbuf.device)
        inv_order = inv_order.int().contiguous()
        bm = self.moe_bm
        desc = ops.moe_build_desc_device(pair_expert, cfg.num_experts, bm=bm)
        gateup = torch.empty(P, 2 * I, dtype=torch.bfloat16, device=hbuf.device)
        ops.moe_grouped_gemm128(gateup, hbuf, layer.w13, pair_token, desc, bm=bm)
        h = torch.empty(P, I, dtype=torch.bfloat16, device=hbuf.device)
        ops.silu_mul(h, gateup)
        z = torch.empty(P, H, dtype=torch.bfloat16, device=hbuf.device)
        ops.moe_grouped_gemm128(z, h, layer.w2, self._arangeP, desc, bm=bm)
        out_bf = torch.empty(T, H, dtype=torch.bfloat16, device=hbuf.device)
        ops.moe_combine_gather(out_bf, z, topk_w.contiguous(), inv_order)
        return out_bf
