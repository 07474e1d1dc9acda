"""Attribute prefill time: profile one T-token prefill forward of the 30B model."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from room_amd.engine.kv_cache import PagedKVCache
from room_amd.models.qwen3_moe import Qwen3MoEConfig, Qwen3MoEModel

T = int(sys.argv[1]) if len(sys.argv) > 1 else 2048
cfg = Qwen3MoEConfig.qwen3_coder_30b()
dev = torch.device("cuda")
model = Qwen3MoEModel(cfg, dev)
cache = PagedKVCache(cfg.num_layers, cfg.num_kv_heads, cfg.head_dim,
                     num_blocks=2048, max_seqs=8,
                     max_blocks_per_seq=cfg.max_position // 16, device=dev)
slot = cache.alloc_seq()
cache.ensure_capacity(slot, T)
tokens = torch.randint(0, cfg.vocab_size, (T,), device=dev)
seq_ids = torch.full((T,), slot, dtype=torch.int32, device=dev)
q_pos = torch.arange(T, dtype=torch.int32, device=dev)
rows = torch.tensor([T - 1], device=dev)
from room_amd import ops
qtiles = ops.build_qtile_desc([(0, T)], dev)

def fwd():
    return model.forward(tokens, seq_ids, q_pos, cache.block_table,
                         cache.kcaches, cache.vcaches, logits_rows=rows,
                         qtile_desc=qtiles)

# warm
fwd(); torch.cuda.synchronize()
t0 = time.time()
fwd(); torch.cuda.synchronize()
print(f"prefill T={T}: {time.time()-t0:.3f}s", flush=True)

from torch.profiler import ProfilerActivity, profile

with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
    fwd()
    torch.cuda.synchronize()
print(prof.key_averages().table(sort_by="self_cuda_time_total", row_limit=28))
