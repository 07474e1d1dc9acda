"""Per-op decode-step attribution on the real 30B layer shapes.

Times each op of one decode layer at batch B with torch.cuda.Event, 100 iters.
Identifies where the 13ms/step goes before optimizing blind.
"""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.nn.functional as F

from room_amd import ops

DEV = "cuda"
H, QD, KVD = 2048, 4096, 512
E, K, I = 128, 8, 768
V = 151936
B = int(sys.argv[1]) if len(sys.argv) > 1 else 5
ITERS = 100

torch.manual_seed(0)


def bench(name, fn, iters=ITERS, bytes_moved=None):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    us = (time.time() - t0) / iters * 1e6
    bw = f" {bytes_moved/us/1e3:.2f} TB/s" if bytes_moved else ""
    print(f"{name:28s} {us:9.1f} µs{bw}", flush=True)
    return us


x = torch.randn(B, H, dtype=torch.bfloat16, device=DEV)
wqkv = torch.randn(QD + 2 * KVD, H, dtype=torch.bfloat16, device=DEV) * 0.02
wo = torch.randn(H, QD, dtype=torch.bfloat16, device=DEV) * 0.02
wr = torch.randn(E, H, dtype=torch.float32, device=DEV) * 0.02
w13 = torch.randn(E, 2 * I, H, dtype=torch.bfloat16, device=DEV) * 0.02
w2 = torch.randn(E, H, I, dtype=torch.bfloat16, device=DEV) * 0.02
lm_head = torch.randn(V, H, dtype=torch.bfloat16, device=DEV) * 0.02
norm_w = torch.ones(H, dtype=torch.bfloat16, device=DEV)
d_w = torch.ones(128, dtype=torch.bfloat16, device=DEV)

total = 0.0
out = torch.empty_like(x)
total += bench("rmsnorm", lambda: ops.rmsnorm(out, x, norm_w),
               bytes_moved=2 * B * H * 2)
res = x.clone()
total += bench("fused_add_rmsnorm",
               lambda: ops.fused_add_rmsnorm(out, res, x, norm_w),
               bytes_moved=4 * B * H * 2)
qkv_out = torch.empty(B, QD + 2 * KVD, dtype=torch.bfloat16, device=DEV)
total += bench("qkv gemv", lambda: ops.gemv(qkv_out, x, wqkv),
               bytes_moved=wqkv.numel() * 2)
bench("qkv linear (hipBLASLt)", lambda: F.linear(x, wqkv),
      bytes_moved=wqkv.numel() * 2)

q = torch.randn(B, 32, 128, dtype=torch.bfloat16, device=DEV)
k = torch.randn(B, 4, 128, dtype=torch.bfloat16, device=DEV)
from room_amd.ops.reference import rope_tables
cos_t, sin_t = rope_tables(8192, 128, 1e7)
cos_t, sin_t = cos_t.to(DEV), sin_t.to(DEV)
pos = torch.full((B,), 512, dtype=torch.int32, device=DEV)
total += bench("qk_norm_rope",
               lambda: ops.qk_norm_rope(q, k, d_w, d_w, cos_t, sin_t, pos, 32, 4))

NB = 4096
kcache = torch.randn(NB, 4, 16, 128, dtype=torch.bfloat16, device=DEV)
vcache = torch.randn(NB, 4, 16, 128, dtype=torch.bfloat16, device=DEV)
bt = torch.arange(B * 64, dtype=torch.int32, device=DEV).reshape(B, 64) % NB
seq_ids = torch.arange(B, dtype=torch.int32, device=DEV)
qpos = torch.full((B,), 512, dtype=torch.int32, device=DEV)
total += bench("write_kv",
               lambda: ops.write_kv(kcache, vcache, k, k, bt, seq_ids, qpos))
attn = torch.empty_like(q)
part = torch.empty(B, 32, 32, 128, dtype=torch.float32, device=DEV)
part_ml = torch.empty(B, 32, 32, 2, dtype=torch.float32, device=DEV)
total += bench("paged_attn_split seq=512",
               lambda: ops.paged_attention_split(attn, q, kcache, vcache, bt,
                                                 seq_ids, qpos, part, part_ml,
                                                 0.0884),
               bytes_moved=B * 4 * 512 * 128 * 2 * 2)
bench("paged_attention(old) seq=512",
      lambda: ops.paged_attention(attn, q, kcache, vcache, bt, seq_ids,
                                  qpos, 0.0884),
      bytes_moved=B * 4 * 512 * 128 * 2 * 2)
o_out = torch.empty(B, H, dtype=torch.bfloat16, device=DEV)
attn_flat = attn.reshape(B, QD).contiguous()
total += bench("o gemv", lambda: ops.gemv(o_out, attn_flat, wo),
               bytes_moved=wo.numel() * 2)
wr_bf = wr.to(torch.bfloat16)
r_out = torch.empty(B, E, dtype=torch.float32, device=DEV)
total += bench("router gemv f32out", lambda: ops.gemv(r_out, x, wr_bf),
               bytes_moved=wr_bf.numel() * 2)
rl = r_out
total += bench("moe_router", lambda: ops.moe_router(rl, K))

ids, w = ops.moe_router(rl, K)
pair_token = torch.arange(B, device=DEV, dtype=torch.int32).repeat_interleave(K)
pair_expert = ids.flatten().contiguous()
pair_w = w.flatten().contiguous()
P = B * K
h = torch.empty(P, I, dtype=torch.bfloat16, device=DEV)
# bytes: weights actually touched = P × (2I×H + H×I) × 2B  (upper bound)
t_h = bench("moe_gemv_h", lambda: ops.moe_gemv_h(h, x, w13, pair_token, pair_expert),
            bytes_moved=P * 2 * I * H * 2)
outf = torch.zeros(B, H, dtype=torch.float32, device=DEV)
t_d = bench("moe_gemv_down",
            lambda: ops.moe_gemv_down(outf, h, w2, pair_w, pair_token, pair_expert),
            bytes_moved=P * H * I * 2)
total += t_h + t_d

print(f"{'—'*50}\nper-layer total ≈ {total:.0f} µs → 48 layers = {total*48/1000:.2f} ms")
lm_out = torch.empty(B, V, dtype=torch.float32, device=DEV)
lm = bench("lm_head gemv", lambda: ops.gemv(lm_out, x, lm_head),
           bytes_moved=lm_head.numel() * 2)
bench("lm_head hipBLASLt", lambda: F.linear(x, lm_head),
      bytes_moved=lm_head.numel() * 2)
smp_logits = lm_out
seeds = torch.randint(1, 2**62, (B,), dtype=torch.int64, device=DEV)
bench("sample_tokens", lambda: ops.sample_tokens(smp_logits, seeds))
out_i = torch.empty(B, dtype=torch.int32, device=DEV)
bench("sample_v3", lambda: ops._require().sample_tokens_v3(out_i, smp_logits, seeds, 40, 0.7, 0.95))
bench("sample_scan_probe", lambda: ops._require().sample_scan_probe(out_i, smp_logits))
# v3 greedy correctness inline check
ops._require().sample_tokens_v3(out_i, smp_logits, seeds, 40, 0.0, 1.0)
assert torch.equal(out_i.long().cpu(), smp_logits.argmax(-1).cpu()), "v3 greedy mismatch"
print("v3 greedy OK")
print(f"estimated decode step = {(total*48 + lm)/1000:.2f} ms (+sampling)")
