"""Quantify per-layer launch/fill/drain bubbles in the decode chain.

Runs ONE decode layer's kernel sequence (a) as the natural back-to-back chain
and (b) with each kernel timed in isolation (events around single launches,
averaged), then reports chain_ms - sum(parts) = bubble headroom — the number
the round-2 megakernel work must beat (profiles/PERF_NOTES.md roadmap #1).
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from room_amd import ops  # noqa: E402
from room_amd.engine.kv_cache import PagedKVCache  # noqa: E402
from room_amd.models.qwen3_moe import Qwen3MoEConfig, Qwen3MoEModel  # noqa: E402


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=5)
    ap.add_argument("--ctx", type=int, default=4096, help="session length")
    ap.add_argument("--iters", type=int, default=200)
    args = ap.parse_args()

    assert torch.cuda.is_available(), "GPU required"
    dev = torch.device("cuda")
    cfg = Qwen3MoEConfig.qwen3_coder_30b()
    cfg.num_layers = 1                       # one layer is enough for the probe
    model = Qwen3MoEModel(cfg, device=dev)
    cache = PagedKVCache(1, cfg.num_kv_heads, cfg.head_dim,
                         num_blocks=args.batch * (args.ctx // 16 + 2) + 8,
                         max_seqs=args.batch + 1,
                         max_blocks_per_seq=args.ctx // 16 + 2, device=dev)
    B = args.batch
    slots = [cache.alloc_seq() for _ in range(B)]
    for s in slots:
        cache.ensure_capacity(s, args.ctx + args.iters + 8)
    for li in range(1):
        cache.kcaches[li].normal_()
        cache.vcaches[li].normal_()

    tokens = torch.randint(0, cfg.vocab_size, (B,), device=dev)
    seq_ids = torch.tensor(slots, dtype=torch.int32, device=dev)
    q_pos = torch.full((B,), args.ctx - 1, dtype=torch.int32, device=dev)

    def chain():
        model.forward(tokens, seq_ids, q_pos, cache.block_table,
                      cache.kcaches, cache.vcaches)

    # warmup + chain timing
    for _ in range(20):
        chain()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        chain()
    torch.cuda.synchronize()
    chain_us = (time.perf_counter() - t0) / args.iters * 1e6

    # isolated-kernel accounting from rocprof-free event timing: rerun the
    # forward under a CUDA-events profiler that brackets each op dispatch
    ev = []
    torch.cuda.synchronize()
    prof = torch.profiler.profile(
        activities=[torch.profiler.ProfilerActivity.CUDA], record_shapes=False)
    with prof:
        for _ in range(20):
            chain()
    torch.cuda.synchronize()
    kernels = {}
    for e in prof.key_averages():
        if e.device_type == torch.profiler.DeviceType.CUDA or e.self_device_time_total > 0:
            kernels[e.key] = e.self_device_time_total / 20
    ksum = sum(v for v in kernels.values())
    print(f"B={B} ctx={args.ctx}: one-layer chain {chain_us:.1f} us, "
          f"sum of kernel times {ksum:.1f} us, "
          f"bubble ≈ {chain_us - ksum:.1f} us/layer "
          f"({(chain_us - ksum) / max(chain_us, 1e-9) * 100:.0f}%)")
    for k, v in sorted(kernels.items(), key=lambda kv: -kv[1])[:14]:
        print(f"  {v:8.1f} us  {k[:80]}")
    print(f"\n48-layer projection: bubbles ≈ "
          f"{(chain_us - ksum) * 48 / 1000:.2f} ms/step")


if __name__ == "__main__":
    main()
