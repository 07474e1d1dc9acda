"""Pure-engine throughput probe (prefill tok/s, decode tok/s by batch size).

Run on a GPU box: python scripts/gpu_bench_engine.py [--tiny]
The numbers here isolate the inference engine from the agent-loop host work;
use with rocprofv3 --stats for per-kernel attribution.
"""
import argparse
import sys
import threading
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from room_amd.engine.llm import LocalEngine
from room_amd.models.qwen3_moe import Qwen3MoEConfig


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--tiny", action="store_true")
    ap.add_argument("--prefill-tokens", type=int, default=2048)
    ap.add_argument("--decode-tokens", type=int, default=64)
    ap.add_argument("--batches", type=int, nargs="+", default=[1, 5, 16])
    args = ap.parse_args()

    cfg = Qwen3MoEConfig.tiny() if args.tiny else Qwen3MoEConfig.qwen3_coder_30b()
    t0 = time.time()
    eng = LocalEngine(cfg=cfg)
    torch.cuda.synchronize()
    print(f"engine init {time.time()-t0:.1f}s, "
          f"{eng.model.num_params()/1e9:.1f}B params, "
          f"kv blocks {eng.cache.num_blocks}", flush=True)

    V = cfg.vocab_size
    import random
    rng = random.Random(7)

    # prefill throughput (single long prompt)
    prompt = [rng.randrange(V) for _ in range(args.prefill_tokens)]
    t0 = time.time()
    req = eng.generate(prompt, max_new_tokens=1, temperature=0.0)
    dt = time.time() - t0
    print(f"prefill: {args.prefill_tokens} tok in {dt:.3f}s = "
          f"{args.prefill_tokens/dt:.0f} tok/s", flush=True)

    # decode throughput by batch size
    for B in args.batches:
        s0 = dict(eng.stats)
        t0 = time.time()
        threads = []
        for i in range(B):
            p = [rng.randrange(V) for _ in range(64)]

            def run(pp=p, i=i):
                eng.generate(pp, max_new_tokens=args.decode_tokens,
                             temperature=0.7, session_key=None)

            th = threading.Thread(target=run)
            th.start()
            threads.append(th)
        for th in threads:
            th.join(300)
        dt = time.time() - t0
        dsteps = eng.stats["decode_steps"] - s0["decode_steps"]
        dtok = eng.stats["decode_tokens"] - s0["decode_tokens"]
        dtime = eng.stats["decode_time"] - s0["decode_time"]
        print(f"B={B}: {dtok} decode tok, {dsteps} steps, wall {dt:.2f}s, "
              f"decode-time {dtime:.2f}s → {dtok/max(dtime,1e-9):.1f} tok/s "
              f"({1000*dtime/max(dsteps,1):.1f} ms/step)", flush=True)

    eng.shutdown()


if __name__ == "__main__":
    main()
