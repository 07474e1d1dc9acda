"""room_amd — MI355X-native autonomous-agent swarm runtime.

A from-scratch reimplementation of the capabilities of quoroom-ai/room
(reference: /root/reference, 100% TypeScript, LLM work delegated to external
providers) redesigned MI355X-first:

- Rooms / Queen / Worker / Quorum semantics and the SQLite on-disk format are
  preserved (reference: src/shared/schema.ts:9-480, src/shared/quorum.ts).
- The LLM (qwen3-coder-30b, MoE) runs IN-PROCESS on PyTorch-ROCm with
  hand-written CDNA4 (gfx950) HIP kernels: RMSNorm, RoPE, paged attention
  (prefill+decode), sampling, MoE grouped GEMM (MFMA + LDS tiling).
- The 384-dim semantic memory store is a HIP batched cosine/top-k kernel
  resident in HBM (reference used sqlite-vec's CPU scan,
  src/shared/embeddings.ts:16-27).
- Workers shard one-per-GPU over RCCL/xGMI; quorum votes and goal/skill
  broadcasts are RCCL collectives overlapped with decode.
"""

__version__ = "0.1.0"
