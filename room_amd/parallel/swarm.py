"""Multi-GPU swarm collectives: RCCL over xGMI.

The reference has zero GPU-level distribution: quorum votes are
SQLite rows, wakes are in-process. Scaling the swarm across an 8×MI355X node
makes each rank (one process per GPU, torch.distributed backend "nccl" = RCCL
on ROCm) own a worker shard + model replica + KV; the cross-GPU primitives:

- quorum_allgather: per-rank vote vectors → all ranks (payload: bytes; xGMI
  p2p latency-bound, so a single small all_gather beats anything clever)
- broadcast_blob: goal/skill/WIP context refresh from the queen rank
- topk_merge: per-GPU memory-store top-k → global top-k on all ranks

Collectives run on a dedicated side stream so the next decode step's kernels
overlap with communication; SQLite on rank 0 stays the durable record.
"""
from __future__ import annotations

import os
import pickle
from typing import Optional

import torch
import torch.distributed as dist


class SwarmContext:
    def __init__(self, rank: int, world_size: int, device: torch.device,
                 group: Optional[dist.ProcessGroup] = None):
        self.rank = rank
        self.world_size = world_size
        self.device = device
        self.group = group
        self.is_distributed = world_size > 1 and dist.is_initialized()
        self.comm_stream = (torch.cuda.Stream(device)
                            if device.type == "cuda" else None)

    # ------------------------------------------------------------ init

    @staticmethod
    def from_env(device: torch.device | None = None) -> "SwarmContext":
        """Initialize from torchrun env (RANK/WORLD_SIZE/MASTER_*); single-rank
        context when not launched distributed."""
        world = int(os.environ.get("WORLD_SIZE", "1"))
        rank = int(os.environ.get("RANK", "0"))
        local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
        if device is None:
            if torch.cuda.is_available():
                torch.cuda.set_device(local_rank)
                device = torch.device("cuda", local_rank)
            else:
                device = torch.device("cpu")
        if world > 1 and not dist.is_initialized():
            backend = "nccl" if device.type == "cuda" else "gloo"
            dist.init_process_group(backend=backend, rank=rank, world_size=world)
        return SwarmContext(rank, world, device)

    def barrier(self) -> None:
        if self.is_distributed:
            dist.barrier(group=self.group)

    # ------------------------------------------------------------ quorum

    def quorum_allgather(self, votes: torch.Tensor) -> torch.Tensor:
        """votes: [n_local_agents] int8 (-1 object / 0 abstain / 1 approve).
        Returns [world, n] on every rank. Runs on the side stream on GPU."""
        if not self.is_distributed:
            return votes.unsqueeze(0)
        v = votes.to(self.device, dtype=torch.int8, non_blocking=True).contiguous()
        out = torch.empty(self.world_size * v.numel(), dtype=torch.int8,
                          device=self.device)
        if self.comm_stream is not None:
            self.comm_stream.wait_stream(torch.cuda.current_stream(self.device))
            with torch.cuda.stream(self.comm_stream):
                dist.all_gather_into_tensor(out, v, group=self.group)
            torch.cuda.current_stream(self.device).wait_stream(self.comm_stream)
        else:
            dist.all_gather_into_tensor(out, v, group=self.group)
        return out.view(self.world_size, -1)

    def quorum_tally(self, votes: torch.Tensor) -> dict:
        allv = self.quorum_allgather(votes).cpu()
        return {
            "yes": int((allv == 1).sum()),
            "no": int((allv == -1).sum()),
            "abstain": int((allv == 0).sum()),
            "total": allv.numel(),
        }

    # ------------------------------------------------------------ broadcast

    def broadcast_blob(self, obj, src: int = 0):
        """Goal/skill/WIP context refresh: queen rank broadcasts a small python
        object (KBs). Uses broadcast of a length-prefixed byte tensor."""
        if not self.is_distributed:
            return obj
        if self.rank == src:
            payload = pickle.dumps(obj)
            size = torch.tensor([len(payload)], dtype=torch.int64, device=self.device)
        else:
            size = torch.zeros(1, dtype=torch.int64, device=self.device)
        dist.broadcast(size, src=src, group=self.group)
        n = int(size.item())
        if self.rank == src:
            buf = torch.frombuffer(bytearray(payload), dtype=torch.uint8).to(self.device)
        else:
            buf = torch.empty(n, dtype=torch.uint8, device=self.device)
        dist.broadcast(buf, src=src, group=self.group)
        if self.rank == src:
            return obj
        return pickle.loads(bytes(buf.cpu().numpy().tobytes()))

    # ------------------------------------------------------------ memory merge

    def topk_merge(self, local_v: torch.Tensor, local_i: torch.Tensor,
                   k: int) -> tuple[torch.Tensor, torch.Tensor]:
        """Per-GPU vector-store top-k → global top-k (scores desc). local_i
        carries globally-unique ids (caller offsets per shard)."""
        if not self.is_distributed:
            v, order = local_v.sort(descending=True)
            return v[:k], local_i[order][:k]
        vs = torch.empty(self.world_size * local_v.numel(), dtype=local_v.dtype,
                         device=self.device)
        is_ = torch.empty(self.world_size * local_i.numel(), dtype=local_i.dtype,
                          device=self.device)
        dist.all_gather_into_tensor(vs, local_v.to(self.device).contiguous(),
                                    group=self.group)
        dist.all_gather_into_tensor(is_, local_i.to(self.device).contiguous(),
                                    group=self.group)
        flat_v, flat_i = vs.flatten(), is_.flatten()
        v, order = flat_v.sort(descending=True)
        return v[:k], flat_i[order][:k]
