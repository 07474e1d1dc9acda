"""Minimal engine bring-up probe for a GPU box (run via gpurun)."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

print("cuda:", torch.cuda.is_available(), torch.cuda.get_device_name(0))

from room_amd.engine import tokenizer as tok
from room_amd.engine.llm import LocalEngine
from room_amd.models.qwen3_moe import Qwen3MoEConfig, Qwen3MoEModel
from room_amd.engine.kv_cache import PagedKVCache

cfg = Qwen3MoEConfig.tiny()
dev = torch.device("cuda")

# step 1: direct model forward (no scheduler thread)
print("== direct forward ==", flush=True)
model = Qwen3MoEModel(cfg, dev)
cache = PagedKVCache(cfg.num_layers, cfg.num_kv_heads, cfg.head_dim,
                     num_blocks=512, max_seqs=8,
                     max_blocks_per_seq=cfg.max_position // 16, device=dev)
slot = cache.alloc_seq()
T = 37
cache.ensure_capacity(slot, T)
tokens = torch.randint(0, cfg.vocab_size, (T,), device=dev)
seq_ids = torch.full((T,), slot, dtype=torch.int32, device=dev)
q_pos = torch.arange(T, dtype=torch.int32, device=dev)
t0 = time.time()
logits = model.forward(tokens, seq_ids, q_pos, cache.block_table,
                       cache.kcaches, cache.vcaches,
                       logits_rows=torch.tensor([T - 1], device=dev))
torch.cuda.synchronize()
print("prefill forward ok", logits.shape, f"{time.time()-t0:.2f}s", flush=True)
assert torch.isfinite(logits).all(), "non-finite logits"

# decode step
tokens = torch.randint(0, cfg.vocab_size, (1,), device=dev)
seq_ids = torch.tensor([slot], dtype=torch.int32, device=dev)
q_pos = torch.tensor([T], dtype=torch.int32, device=dev)
cache.ensure_capacity(slot, T + 1)
logits = model.forward(tokens, seq_ids, q_pos, cache.block_table,
                       cache.kcaches, cache.vcaches)
torch.cuda.synchronize()
print("decode forward ok", logits.shape, flush=True)

# step 2: engine with scheduler thread
print("== engine ==", flush=True)
eng = LocalEngine(cfg=Qwen3MoEConfig.tiny(), kv_gb=2.0, max_seqs=16)
prompt = tok.encode("hello swarm " * 30)
t0 = time.time()
req = eng.generate(prompt, max_new_tokens=8, timeout=60)
print("generate ok:", req.out_tokens, f"{time.time()-t0:.2f}s", flush=True)
eng.shutdown()
print("ALL OK", flush=True)
