"""Paged KV cache manager sized for 288 GB HBM3E.

Block pool shared by all sequences on this GPU; sessions retain their blocks
across agent cycles (session continuity becomes KV-cache
residency, with the agent_sessions SQLite row as the durable fallback).
"""
from __future__ import annotations

import torch

BLOCK_SIZE = 16


class PagedKVCache:
    def __init__(self, num_layers: int, num_kv_heads: int, head_dim: int,
                 num_blocks: int, max_seqs: int, max_blocks_per_seq: int,
                 device: torch.device):
        self.block_size = BLOCK_SIZE
        self.num_blocks = num_blocks
        self.max_seqs = max_seqs
        self.max_blocks_per_seq = max_blocks_per_seq
        self.kcaches = [torch.zeros(num_blocks, num_kv_heads, BLOCK_SIZE, head_dim,
                                    dtype=torch.bfloat16, device=device)
                        for _ in range(num_layers)]
        self.vcaches = [torch.zeros(num_blocks, num_kv_heads, BLOCK_SIZE, head_dim,
                                    dtype=torch.bfloat16, device=device)
                        for _ in range(num_layers)]
        self.block_table = torch.zeros(max_seqs, max_blocks_per_seq,
                                       dtype=torch.int32, device=device)
        self._bt_host = [[0] * max_blocks_per_seq for _ in range(max_seqs)]
        self.free_blocks = list(range(num_blocks - 1, 0, -1))  # block 0 reserved
        self.free_slots = list(range(max_seqs - 1, -1, -1))
        self.seq_len: dict[int, int] = {}
        self.seq_nblocks: dict[int, int] = {}

    @staticmethod
    def bytes_per_block(num_layers: int, num_kv_heads: int, head_dim: int) -> int:
        return num_layers * 2 * num_kv_heads * BLOCK_SIZE * head_dim * 2

    def alloc_seq(self) -> int:
        if not self.free_slots:
            raise RuntimeError("KV cache: no free sequence slots")
        slot = self.free_slots.pop()
        self.seq_len[slot] = 0
        self.seq_nblocks[slot] = 0
        return slot

    def free_seq(self, slot: int) -> None:
        n = self.seq_nblocks.pop(slot, 0)
        for i in range(n):
            self.free_blocks.append(self._bt_host[slot][i])
            self._bt_host[slot][i] = 0
        if n:
            self.block_table[slot, :n] = 0
        self.seq_len.pop(slot, None)
        self.free_slots.append(slot)

    def ensure_capacity(self, slot: int, new_len: int) -> None:
        """Extend the block list of `slot` to cover new_len tokens."""
        need = (new_len + BLOCK_SIZE - 1) // BLOCK_SIZE
        have = self.seq_nblocks[slot]
        if need <= have:
            return
        if need > self.max_blocks_per_seq:
            raise RuntimeError(f"sequence exceeds max length "
                               f"({new_len} > {self.max_blocks_per_seq * BLOCK_SIZE})")
        add = need - have
        if add > len(self.free_blocks):
            raise RuntimeError("KV cache: out of blocks")
        new_blocks = [self.free_blocks.pop() for _ in range(add)]
        for i, b in enumerate(new_blocks):
            self._bt_host[slot][have + i] = b
        self.block_table[slot, have:need] = torch.tensor(
            new_blocks, dtype=torch.int32, device=self.block_table.device)
        self.seq_nblocks[slot] = need

    def blocks_free(self) -> int:
        return len(self.free_blocks)
