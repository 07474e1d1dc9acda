"""Numerics tests: every CDNA4 HIP kernel vs a plain PyTorch fp32 reference
(room_amd/ops/reference.py) on random inputs. GPU-only."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from room_amd import ops
    from room_amd.ops import reference as ref
else:
    pytest.skip("no GPU", allow_module_level=True)

DEV = "cuda"


def bf16_close(a, b, atol=2e-2, rtol=2e-2):
    return torch.allclose(a.float(), b.float(), atol=atol, rtol=rtol)


def test_rmsnorm():
    torch.manual_seed(0)
    x = torch.randn(64, 2048, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(2048, dtype=torch.bfloat16, device=DEV)
    out = torch.empty_like(x)
    ops.rmsnorm(out, x, w, 1e-6)
    expect = ref.rmsnorm_ref(x, w)
    assert bf16_close(out, expect)


def test_fused_add_rmsnorm():
    torch.manual_seed(1)
    x = torch.randn(32, 2048, dtype=torch.bfloat16, device=DEV)
    res = torch.randn(32, 2048, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(2048, dtype=torch.bfloat16, device=DEV)
    res_ref = res.clone()
    out = torch.empty_like(x)
    ops.fused_add_rmsnorm(out, res, x, w, 1e-6)
    expect_out, expect_res = ref.fused_add_rmsnorm_ref(res_ref, x, w)
    assert bf16_close(res, expect_res)
    assert bf16_close(out, expect_out)


def test_qk_norm_rope():
    torch.manual_seed(2)
    T, Hq, Hk, D = 17, 32, 4, 128
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(T, Hk, D, dtype=torch.bfloat16, device=DEV)
    q_w = torch.randn(D, dtype=torch.bfloat16, device=DEV)
    k_w = torch.randn(D, dtype=torch.bfloat16, device=DEV)
    cos_t, sin_t = ref.rope_tables(512, D, 1e6)
    cos_t, sin_t = cos_t.to(DEV), sin_t.to(DEV)
    pos = torch.randint(0, 512, (T,), dtype=torch.int32, device=DEV)
    q_ref, k_ref = ref.qk_norm_rope_ref(q, k, q_w, k_w, cos_t, sin_t, pos)
    ops.qk_norm_rope(q, k, q_w, k_w, cos_t, sin_t, pos, Hq, Hk, D, 1e-6)
    assert bf16_close(q, q_ref, atol=3e-2)
    assert bf16_close(k, k_ref, atol=3e-2)


def test_silu_mul():
    torch.manual_seed(3)
    gu = torch.randn(40, 1536, dtype=torch.bfloat16, device=DEV)
    out = torch.empty(40, 768, dtype=torch.bfloat16, device=DEV)
    ops.silu_mul(out, gu)
    assert bf16_close(out, ref.silu_mul_ref(gu))


def _setup_cache(num_seqs, max_len, Hk=4, D=128, BS=16):
    max_blocks = (max_len + BS - 1) // BS
    nb = num_seqs * max_blocks + 1
    kcache = torch.randn(nb, Hk, BS, D, dtype=torch.bfloat16, device=DEV)
    vcache = torch.randn(nb, Hk, BS, D, dtype=torch.bfloat16, device=DEV)
    # unique block per (seq, i)
    bt = torch.arange(1, num_seqs * max_blocks + 1, dtype=torch.int32,
                      device=DEV).reshape(num_seqs, max_blocks)
    return kcache, vcache, bt


def test_paged_attention_decode():
    torch.manual_seed(4)
    B, Hq, Hk, D = 3, 32, 4, 128
    lens = [33, 7, 100]
    kcache, vcache, bt = _setup_cache(B, 128)
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=DEV)
    seq_ids = torch.arange(B, dtype=torch.int32, device=DEV)
    q_pos = torch.tensor([l - 1 for l in lens], dtype=torch.int32, device=DEV)
    out = torch.empty_like(q)
    scale = D ** -0.5
    ops.paged_attention(out, q, kcache, vcache, bt, seq_ids, q_pos, scale)
    expect = ref.paged_attention_ref(q, kcache, vcache, bt, seq_ids, q_pos, scale)
    assert bf16_close(out, expect, atol=3e-2)


def test_paged_attention_prefill_tokens():
    """Prefill = many query tokens per sequence, causal bound per token."""
    torch.manual_seed(5)
    Hq, Hk, D = 32, 4, 128
    kcache, vcache, bt = _setup_cache(2, 64)
    seq_ids, q_pos = [], []
    for s, l in [(0, 40), (1, 22)]:
        for p in range(l):
            seq_ids.append(s)
            q_pos.append(p)
    T = len(seq_ids)
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device=DEV)
    seq_ids = torch.tensor(seq_ids, dtype=torch.int32, device=DEV)
    q_pos = torch.tensor(q_pos, dtype=torch.int32, device=DEV)
    out = torch.empty_like(q)
    scale = D ** -0.5
    ops.paged_attention(out, q, kcache, vcache, bt, seq_ids, q_pos, scale)
    expect = ref.paged_attention_ref(q, kcache, vcache, bt, seq_ids, q_pos, scale)
    assert bf16_close(out, expect, atol=3e-2)


def test_write_kv_roundtrip():
    torch.manual_seed(6)
    Hk, D, BS = 4, 128, 16
    kcache = torch.zeros(8, Hk, BS, D, dtype=torch.bfloat16, device=DEV)
    vcache = torch.zeros(8, Hk, BS, D, dtype=torch.bfloat16, device=DEV)
    bt = torch.tensor([[1, 3, 5, 7]], dtype=torch.int32, device=DEV)
    T = 50
    k = torch.randn(T, Hk, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(T, Hk, D, dtype=torch.bfloat16, device=DEV)
    seq_ids = torch.zeros(T, dtype=torch.int32, device=DEV)
    q_pos = torch.arange(T, dtype=torch.int32, device=DEV)
    ops.write_kv(kcache, vcache, k, v, bt, seq_ids, q_pos)
    for pos in [0, 15, 16, 31, 49]:
        blk = int(bt[0, pos // BS])
        assert torch.equal(kcache[blk, :, pos % BS, :], k[pos])
        assert torch.equal(vcache[blk, :, pos % BS, :], v[pos])


def test_moe_router():
    torch.manual_seed(7)
    T, E, K = 11, 128, 8
    logits = torch.randn(T, E, dtype=torch.float32, device=DEV) * 2
    ids, w = ops.moe_router(logits, K)
    ids_ref, w_ref = ref.moe_router_ref(logits, K)
    for t in range(T):
        assert set(ids[t].tolist()) == set(ids_ref[t].tolist())
        assert w[t].sum().item() == pytest.approx(1.0, abs=1e-4)
        # weights matched by expert id
        m = {int(i): float(x) for i, x in zip(ids[t], w[t])}
        mr = {int(i): float(x) for i, x in zip(ids_ref[t], w_ref[t])}
        for e in m:
            assert m[e] == pytest.approx(mr[e], abs=1e-4)


def test_router_topk_fused():
    """Fused router (GEMV + softmax + top-k) vs fp32 torch reference."""
    torch.manual_seed(11)
    for T, E, H, K in [(5, 128, 2048, 8), (1, 16, 512, 4)]:
        x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV) * 0.5
        wr = torch.randn(E, H, dtype=torch.bfloat16, device=DEV) * 0.05
        ids, w = ops.router_topk(x, wr, K)
        logits_ref = (x.float() @ wr.float().t())
        ids_ref, w_ref = ref.moe_router_ref(logits_ref, K)
        for t in range(T):
            assert set(ids[t].tolist()) == set(ids_ref[t].tolist()), \
                f"T={T} t={t}: {ids[t].tolist()} vs {ids_ref[t].tolist()}"
            assert w[t].sum().item() == pytest.approx(1.0, abs=1e-3)
            m = {int(i): float(v) for i, v in zip(ids[t], w[t])}
            mr = {int(i): float(v) for i, v in zip(ids_ref[t], w_ref[t])}
            for e in m:
                assert m[e] == pytest.approx(mr[e], abs=5e-3)


def _moe_setup(T=5, E=16, H=2048, I=768, K=8):
    x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV) * 0.5
    w13 = torch.randn(E, 2 * I, H, dtype=torch.bfloat16, device=DEV) * 0.02
    w2 = torch.randn(E, H, I, dtype=torch.bfloat16, device=DEV) * 0.02
    logits = torch.randn(T, E, dtype=torch.float32, device=DEV)
    ids, w = ref.moe_router_ref(logits, K)
    return x, w13, w2, ids.to(DEV), w.to(DEV)


def test_moe_gemv_path():
    torch.manual_seed(8)
    x, w13, w2, ids, w = _moe_setup()
    T, K = ids.shape
    H, I = x.size(1), w2.size(2)
    pair_token = torch.arange(T, device=DEV).repeat_interleave(K).int()
    pair_expert = ids.flatten().int()
    pair_w = w.flatten().float()
    P = pair_token.numel()
    h = torch.empty(P, I, dtype=torch.bfloat16, device=DEV)
    ops.moe_gemv_h(h, x, w13, pair_token, pair_expert)
    out = torch.zeros(T, H, dtype=torch.float32, device=DEV)
    ops.moe_gemv_down(out, h, w2, pair_w, pair_token, pair_expert)
    expect = ref.moe_ref(x, w13, w2, ids, w)
    assert bf16_close(out, expect, atol=5e-2, rtol=5e-2)


def test_moe_grouped_gemm_vs_matmul():
    """MFMA fragment-layout check: asymmetric random A and B (guide G9)."""
    torch.manual_seed(9)
    E, H, N = 4, 2048, 1536
    T = 37
    x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV) * 0.3
    w = torch.randn(E, N, H, dtype=torch.bfloat16, device=DEV) * 0.05
    pair_expert = torch.randint(0, E, (T,), device=DEV).int()
    order = torch.argsort(pair_expert)
    pair_expert = pair_expert[order].contiguous()
    pair_token = order.int().contiguous()
    tile_desc = ops.build_moe_tile_desc(pair_expert, N // 64, E)
    out = torch.empty(T, N, dtype=torch.bfloat16, device=DEV)
    ops.moe_grouped_gemm(out, x, w, pair_token, tile_desc)
    for p in range(T):
        t, e = int(pair_token[p]), int(pair_expert[p])
        expect = (x[t].float() @ w[e].float().T).to(torch.bfloat16)
        assert bf16_close(out[p], expect, atol=6e-2, rtol=6e-2), f"pair {p}"


def test_moe_grouped_path_full():
    """Full grouped path (sorted pairs → gateup gemm → silu_mul → down gemm →
    combine) vs reference MoE."""
    torch.manual_seed(10)
    x, w13, w2, ids, w = _moe_setup(T=23)
    T, K = ids.shape
    H, I = x.size(1), w2.size(2)
    flat_expert = ids.flatten().int()
    order = torch.argsort(flat_expert)
    pair_expert = flat_expert[order].contiguous()
    pair_token = (torch.arange(T, device=DEV).repeat_interleave(K).int())[order].contiguous()
    pair_w = w.flatten().float()[order].contiguous()
    P = pair_token.numel()

    gateup = torch.empty(P, 2 * I, dtype=torch.bfloat16, device=DEV)
    desc1 = ops.build_moe_tile_desc(pair_expert, (2 * I) // 64, 16)
    ops.moe_grouped_gemm(gateup, x, w13, pair_token, desc1)
    h = torch.empty(P, I, dtype=torch.bfloat16, device=DEV)
    ops.silu_mul(h, gateup)
    z = torch.empty(P, H, dtype=torch.bfloat16, device=DEV)
    desc2 = ops.build_moe_tile_desc(pair_expert, H // 64, 16)
    # down gemm reads h rows per sorted pair: identity token mapping
    ops.moe_grouped_gemm(z, h, w2, torch.arange(P, device=DEV).int(), desc2)
    out = torch.zeros(T, H, dtype=torch.float32, device=DEV)
    ops.moe_combine(out, z, pair_w, pair_token)
    expect = ref.moe_ref(x, w13, w2, ids, w)
    assert bf16_close(out, expect, atol=6e-2, rtol=6e-2)


def test_sampling_greedy_and_topk_membership():
    torch.manual_seed(11)
    B, V = 4, 151936
    logits = torch.randn(B, V, dtype=torch.float32, device=DEV)
    seeds = torch.randint(1, 2**62, (B,), dtype=torch.int64, device=DEV)
    # greedy: temperature 0 → argmax
    toks = ops.sample_tokens(logits, seeds, top_k=40, temperature=0.0, top_p=1.0)
    assert torch.equal(toks.long().cpu(), logits.argmax(-1).cpu())
    # sampled tokens must come from the true top-k set
    k = 16
    topk_sets = [set(logits[b].topk(k).indices.tolist()) for b in range(B)]
    for trial in range(5):
        seeds = torch.randint(1, 2**62, (B,), dtype=torch.int64, device=DEV)
        toks = ops.sample_tokens(logits, seeds, top_k=k, temperature=1.0, top_p=1.0)
        for b in range(B):
            assert int(toks[b]) in topk_sets[b]


def test_sampling_top_p_cut():
    # one dominant logit with top_p small → always picks it
    B, V = 2, 50000
    logits = torch.full((B, V), -10.0, dtype=torch.float32, device=DEV)
    logits[:, 123] = 10.0
    logits[:, 456] = 5.0
    for trial in range(3):
        seeds = torch.randint(1, 2**62, (B,), dtype=torch.int64, device=DEV)
        toks = ops.sample_tokens(logits, seeds, top_k=40, temperature=1.0, top_p=0.5)
        assert toks.tolist() == [123, 123]


def test_vector_store_topk():
    torch.manual_seed(12)
    N, D, k = 100_000, 384, 10
    mat = torch.randn(N, D, dtype=torch.float32, device=DEV)
    mat = torch.nn.functional.normalize(mat, dim=-1).to(torch.bfloat16)
    query = torch.nn.functional.normalize(
        torch.randn(D, dtype=torch.float32, device=DEV), dim=-1)
    v, i = ops.vs_topk(mat, query, k)
    v_ref, i_ref = ref.vs_topk_ref(mat, query, k)
    # bf16 dot rounding can swap near-ties: check score agreement
    assert torch.allclose(v.cpu(), v_ref.cpu(), atol=1e-2)
    overlap = len(set(i.tolist()) & set(i_ref.tolist()))
    assert overlap >= k - 2


def test_gemv_vs_matmul():
    torch.manual_seed(13)
    # cover every BN template specialization (1..8)
    for B, H, N in [(1, 2048, 5120), (2, 2048, 512), (3, 512, 1024),
                    (4, 4096, 512), (5, 4096, 2048), (6, 2048, 512),
                    (7, 512, 512), (8, 2048, 128)]:
        x = torch.randn(B, H, dtype=torch.bfloat16, device=DEV)
        w = torch.randn(N, H, dtype=torch.bfloat16, device=DEV) * 0.05
        y = torch.empty(B, N, dtype=torch.bfloat16, device=DEV)
        ops.gemv(y, x, w)
        expect = (x.float() @ w.float().T)
        assert bf16_close(y, expect.to(torch.bfloat16), atol=5e-2, rtol=5e-2)
        yf = torch.empty(B, N, dtype=torch.float32, device=DEV)
        ops.gemv(yf, x, w)
        assert torch.allclose(yf, expect, atol=5e-2, rtol=5e-2)


def test_paged_attention_split_matches_ref():
    torch.manual_seed(14)
    B, Hq, Hk, D = 4, 32, 4, 128
    lens = [1, 33, 250, 500]
    kcache, vcache, bt = _setup_cache(B, 512)
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=DEV)
    seq_ids = torch.arange(B, dtype=torch.int32, device=DEV)
    q_pos = torch.tensor([l - 1 for l in lens], dtype=torch.int32, device=DEV)
    out = torch.empty_like(q)
    nsp = ops.attn_nsplits()
    part = torch.empty(B, Hq, nsp, D, dtype=torch.float32, device=DEV)
    part_ml = torch.empty(B, Hq, nsp, 2, dtype=torch.float32, device=DEV)
    scale = D ** -0.5
    ops.paged_attention_split(out, q, kcache, vcache, bt, seq_ids, q_pos,
                              part, part_ml, scale)
    expect = ref.paged_attention_ref(q, kcache, vcache, bt, seq_ids, q_pos, scale)
    assert bf16_close(out, expect, atol=3e-2)
    # 64-split band (long-context decode graphs) must match too
    out64 = torch.empty_like(q)
    ops.paged_attention_split(out64, q, kcache, vcache, bt, seq_ids, q_pos,
                              part, part_ml, scale, splits=64)
    assert bf16_close(out64, expect, atol=3e-2)


def test_moe_grouped_gemm128_vs_matmul():
    torch.manual_seed(15)
    E, H, N = 4, 2048, 1536
    T = 300  # several 128-row tiles plus a ragged tail
    x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV) * 0.3
    w = torch.randn(E, N, H, dtype=torch.bfloat16, device=DEV) * 0.05
    pair_expert = torch.randint(0, E, (T,), device=DEV).int()
    order = torch.argsort(pair_expert)
    pair_expert = pair_expert[order].contiguous()
    pair_token = order.int().contiguous()
    tile_desc = ops.moe_build_desc_device(pair_expert, E, bm=128)
    out = torch.empty(T, N, dtype=torch.bfloat16, device=DEV)
    ops.moe_grouped_gemm128(out, x, w, pair_token, tile_desc)
    for p in [0, 1, 100, 150, 299]:
        t, e = int(pair_token[p]), int(pair_expert[p])
        expect = (x[t].float() @ w[e].float().T).to(torch.bfloat16)
        assert bf16_close(out[p], expect, atol=6e-2, rtol=6e-2), f"pair {p}"


def test_flash_prefill_vs_ref():
    torch.manual_seed(16)
    Hq, Hk, D = 32, 4, 128
    kcache, vcache, bt = _setup_cache(2, 128)
    # two sequences: 50 tokens from pos 0, 37 tokens continuing from pos 20
    segments, seq_ids, q_pos = [], [], []
    row = 0
    for s, start, n in [(0, 0, 50), (1, 20, 37)]:
        segments.append((row, n))
        for p in range(start, start + n):
            seq_ids.append(s)
            q_pos.append(p)
        row += n
    T = len(seq_ids)
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device=DEV)
    seq_ids = torch.tensor(seq_ids, dtype=torch.int32, device=DEV)
    q_pos = torch.tensor(q_pos, dtype=torch.int32, device=DEV)
    desc = ops.build_qtile_desc(segments, DEV)
    out = torch.empty_like(q)
    scale = D ** -0.5
    ops.flash_prefill(out, q, kcache, vcache, bt, seq_ids, q_pos, desc, scale)
    expect = ref.paged_attention_ref(q, kcache, vcache, bt, seq_ids, q_pos, scale)
    assert bf16_close(out, expect, atol=4e-2, rtol=4e-2)


def test_moe_gemv_dedup_matches_ref():
    torch.manual_seed(17)
    x, w13, w2, ids, w = _moe_setup(T=6)
    T, H = x.shape
    scratch = {}
    out = torch.zeros(T, H, dtype=torch.float32, device=DEV)
    ops.moe_gemv_dedup(out, x, w13, w2, ids, w, scratch)
    expect = ref.moe_ref(x, w13, w2, ids, w)
    assert bf16_close(out, expect, atol=5e-2, rtol=5e-2)
    # second call must reset counts correctly (persistent scratch)
    out2 = torch.zeros(T, H, dtype=torch.float32, device=DEV)
    ops.moe_gemv_dedup(out2, x, w13, w2, ids, w, scratch)
    assert torch.allclose(out, out2, atol=1e-3)


def test_qk_rope_write_kv_fused_matches_unfused():
    torch.manual_seed(18)
    T, Hq, Hk, D = 9, 32, 4, 128
    kcache, vcache, bt = _setup_cache(1, 160)
    kcache2, vcache2 = kcache.clone(), vcache.clone()
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(T, Hk, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(T, Hk, D, dtype=torch.bfloat16, device=DEV)
    q2, k2 = q.clone(), k.clone()
    q_w = torch.randn(D, dtype=torch.bfloat16, device=DEV)
    k_w = torch.randn(D, dtype=torch.bfloat16, device=DEV)
    cos_t, sin_t = ref.rope_tables(512, D, 1e6)
    cos_t, sin_t = cos_t.to(DEV), sin_t.to(DEV)
    seq_ids = torch.zeros(T, dtype=torch.int32, device=DEV)
    pos = torch.arange(40, 40 + T, dtype=torch.int32, device=DEV)
    # unfused reference path
    ops.qk_norm_rope(q2, k2, q_w, k_w, cos_t, sin_t, pos, Hq, Hk, D)
    ops.write_kv(kcache2, vcache2, k2, v, bt, seq_ids, pos)
    # fused, on the packed qkv buffer
    qkv = torch.cat([q.reshape(T, -1), k.reshape(T, -1), v.reshape(T, -1)],
                    dim=1).contiguous()
    ops.qk_rope_write_kv(qkv, Hq, kcache, vcache, q_w, k_w, cos_t, sin_t,
                         bt, seq_ids, pos)
    q_out = qkv[:, :Hq * D].reshape(T, Hq, D)
    assert torch.equal(q_out, q2)
    assert torch.equal(vcache, vcache2)
    assert bf16_close(kcache, kcache2, atol=1e-3)


@pytest.mark.parametrize("B", [1, 2, 5])
def test_gemv_addnorm_matches_composition(B):
    torch.manual_seed(19)
    H, N = 2048, 1024
    x = torch.randn(B, H, dtype=torch.bfloat16, device=DEV)
    gamma = torch.randn(H, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(N, H, dtype=torch.bfloat16, device=DEV) * 0.05
    for delta in (torch.randn(B, H, dtype=torch.float32, device=DEV) * 0.1,
                  torch.randn(B, H, dtype=torch.bfloat16, device=DEV) * 0.1,
                  torch.empty(0, dtype=torch.float32, device=DEV)):
        x_in = x.clone()
        x_out = torch.empty_like(x)
        y = torch.empty(B, N, dtype=torch.bfloat16, device=DEV)
        ops.gemv_addnorm(y, x_in, delta, x_out, gamma, w, 1e-6)
        # reference composition: fused_add_rmsnorm (or rmsnorm) then gemv
        if delta.numel():
            res = x.clone()
            hb = torch.empty_like(x)
            ops.fused_add_rmsnorm(hb, res, delta.to(delta.dtype), gamma, 1e-6)
            assert bf16_close(x_out, res)           # residual update matches
        else:
            hb = torch.empty_like(x)
            ops.rmsnorm(hb, x, gamma, 1e-6)
        y_ref = torch.empty(B, N, dtype=torch.bfloat16, device=DEV)
        ops.gemv(y_ref, hb, w)
        assert bf16_close(y, y_ref, atol=6e-2, rtol=6e-2)
        # f32 output variant
        yf = torch.empty(B, N, dtype=torch.float32, device=DEV)
        ops.gemv_addnorm(yf, x.clone(), delta, torch.empty_like(x), gamma, w)
        assert bf16_close(yf, y_ref, atol=6e-2, rtol=6e-2)


def test_router_gemv_topk_split():
    """H-split router GEMV + top-k vs fp32 torch reference."""
    torch.manual_seed(13)
    for T, E, H, K in [(5, 128, 2048, 8), (1, 16, 512, 4), (8, 128, 2048, 8)]:
        x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV) * 0.5
        wr = torch.randn(E, H, dtype=torch.bfloat16, device=DEV) * 0.05
        ids, w = ops.router_gemv_topk(x, wr, K)
        ids_ref, w_ref = ref.moe_router_ref(x.float() @ wr.float().t(), K)
        for t in range(T):
            assert set(ids[t].tolist()) == set(ids_ref[t].tolist())
            m = {int(i): float(v) for i, v in zip(ids[t], w[t])}
            mr = {int(i): float(v) for i, v in zip(ids_ref[t], w_ref[t])}
            for e in m:
                assert m[e] == pytest.approx(mr[e], abs=5e-3)


@pytest.mark.parametrize("B", [1, 5, 8])
def test_attn_merge_o_matches_composition(B):
    """Fused split-merge + O-projection vs the two-kernel path
    (paged_attention_split's merge + gemv), which is itself ref-validated."""
    torch.manual_seed(23)
    Hq, Hk, D = 32, 4, 128
    H = 2048
    lens = [1 + 97 * i for i in range(B)]
    kcache, vcache, bt = _setup_cache(B, 1024)
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=DEV)
    wo = torch.randn(H, Hq * D, dtype=torch.bfloat16, device=DEV) * 0.05
    seq_ids = torch.arange(B, dtype=torch.int32, device=DEV)
    q_pos = torch.tensor([l - 1 for l in lens], dtype=torch.int32, device=DEV)
    nsp = ops.attn_nsplits()
    part = torch.empty(B, Hq, nsp, D, dtype=torch.float32, device=DEV)
    part_ml = torch.empty(B, Hq, nsp, 2, dtype=torch.float32, device=DEV)
    scale = D ** -0.5
    # fused path
    o_accum = torch.full((B, H), 777.0, dtype=torch.float32, device=DEV)
    ops.paged_attention_splitk(part, part_ml, q, kcache, vcache, bt,
                               seq_ids, q_pos, scale, o_accum)
    ops.attn_merge_o(o_accum, part, part_ml, wo)
    # two-kernel reference path
    attn = torch.empty_like(q)
    ops.paged_attention_split(attn, q, kcache, vcache, bt, seq_ids, q_pos,
                              part, part_ml, scale)
    o_ref = torch.empty(B, H, dtype=torch.float32, device=DEV)
    ops.gemv(o_ref, attn.reshape(B, Hq * D), wo)
    assert bf16_close(o_accum, o_ref, atol=6e-2, rtol=6e-2)
    # fp32 torch reference end-to-end
    merged = ref.paged_attention_ref(q, kcache, vcache, bt, seq_ids, q_pos,
                                     scale)
    o_t = merged.reshape(B, Hq * D).float() @ wo.float().T
    assert bf16_close(o_accum, o_t, atol=0.25, rtol=6e-2)


@pytest.mark.parametrize("B,E", [(1, 128), (5, 128), (8, 16)])
def test_router_addnorm_matches_composition(B, E):
    torch.manual_seed(24)
    H = 2048 if E == 128 else 512
    x = torch.randn(B, H, dtype=torch.bfloat16, device=DEV)
    delta = torch.randn(B, H, dtype=torch.float32, device=DEV) * 0.1
    gamma = torch.randn(H, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(E, H, dtype=torch.bfloat16, device=DEV) * 0.05
    x_out = torch.empty_like(x)
    xn_out = torch.empty_like(x)
    y = torch.empty(B, E, dtype=torch.float32, device=DEV)
    ops.router_addnorm(y, x, delta, x_out, xn_out, gamma, w, 1e-6)
    # composition reference: fused_add_rmsnorm then f32 gemv
    res = x.clone()
    hb = torch.empty_like(x)
    ops.fused_add_rmsnorm(hb, res, delta, gamma, 1e-6)
    assert bf16_close(x_out, res)
    assert bf16_close(xn_out, hb)
    y_ref = torch.empty(B, E, dtype=torch.float32, device=DEV)
    ops.gemv(y_ref, hb, w)
    assert bf16_close(y, y_ref, atol=6e-2, rtol=6e-2)
    # fp32 torch reference
    s = (x.float() + delta).to(torch.bfloat16).float()
    xn_t = s * torch.rsqrt(s.pow(2).mean(-1, keepdim=True) + 1e-6) * gamma.float()
    y_t = xn_t @ w.float().T
    assert bf16_close(y, y_t, atol=0.3, rtol=6e-2)


def test_moe_grouped_gemm_bm256_vs_matmul():
    """BM=256 m-tiles (half the expert-panel re-reads) match the reference."""
    torch.manual_seed(25)
    E, H, N = 4, 2048, 1536
    T = 700  # >1 256-row tile per expert plus ragged tails
    x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV) * 0.3
    w = torch.randn(E, N, H, dtype=torch.bfloat16, device=DEV) * 0.05
    pair_expert = torch.randint(0, E, (T,), device=DEV).int()
    order = torch.argsort(pair_expert)
    pair_expert = pair_expert[order].contiguous()
    pair_token = order.int().contiguous()
    tile_desc = ops.moe_build_desc_device(pair_expert, E, bm=256)
    out = torch.empty(T, N, dtype=torch.bfloat16, device=DEV)
    ops.moe_grouped_gemm128(out, x, w, pair_token, tile_desc, bm=256)
    for p in [0, 1, 200, 399, 550, 699]:
        t, e = int(pair_token[p]), int(pair_expert[p])
        expect = (x[t].float() @ w[e].float().T).to(torch.bfloat16)
        assert bf16_close(out[p], expect, atol=6e-2, rtol=6e-2), f"pair {p}"
