"""Microbench: split-KV decode attention at bench-like context lengths.

Times room_amd.ops.paged_attention_split for B decode tokens against per-seq
KV lengths L (the bench's hot shape: B=5 agents, sessions ~4-6k tokens).
Cold-cache rotation over independent KV pools so L2 doesn't flatter the loop.
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from room_amd import ops  # noqa: E402

HK, HQ, D, BS = 4, 32, 128, 16
NSPLITS = ops.attn_nsplits()


def run(B: int, L: int, iters: int = 50) -> float:
    torch.manual_seed(0)
    dev = "cuda"
    nblk_seq = (L + BS - 1) // BS + 1
    # several independent pools (> L2 512MB) rotated per iter
    pools = []
    npools = 6
    for _ in range(npools):
        nb = B * nblk_seq + 1
        kc = torch.randn(nb, HK, BS, D, device=dev, dtype=torch.bfloat16)
        vc = torch.randn(nb, HK, BS, D, device=dev, dtype=torch.bfloat16)
        bt = torch.arange(1, nb, device=dev, dtype=torch.int32).reshape(B, nblk_seq)
        pools.append((kc, vc, bt))
    q = torch.randn(B, HQ, D, device=dev, dtype=torch.bfloat16)
    seq_ids = torch.arange(B, device=dev, dtype=torch.int32)
    q_pos = torch.full((B,), L - 1, device=dev, dtype=torch.int32)
    out = torch.empty(B, HQ, D, device=dev, dtype=torch.bfloat16)
    part = torch.empty(B, HQ, NSPLITS, D, device=dev, dtype=torch.float32)
    part_ml = torch.empty(B, HQ, NSPLITS, 2, device=dev, dtype=torch.float32)
    scale = D ** -0.5

    for i in range(10):  # warmup
        kc, vc, bt = pools[i % npools]
        ops.paged_attention_split(out, q, kc, vc, bt, seq_ids, q_pos,
                                  part, part_ml, scale)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(iters):
        kc, vc, bt = pools[i % npools]
        ops.paged_attention_split(out, q, kc, vc, bt, seq_ids, q_pos,
                                  part, part_ml, scale)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    kv_mb = B * L * HK * D * 2 * 2 / 1e6
    print(f"B={B} L={L}: {us:7.1f} us/call  (KV {kv_mb:.0f} MB -> "
          f"{kv_mb / us:.2f} TB/s effective)")
    return us


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--batches", type=int, nargs="+", default=[1, 5])
    ap.add_argument("--lens", type=int, nargs="+", default=[1024, 5000, 16384])
    args = ap.parse_args()
    for B in args.batches:
        for L in args.lens:
            run(B, L)
