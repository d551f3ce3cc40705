"""Intra-node pipeline parallelism over RCCL/xGMI.

The MI355X-native replacement for the reference's TCP `rpc_push` hand-off when
adjacent spans live on one 8xMI355X node (SURVEY §2.2 PP): activations move
rank->rank with torch.distributed send/recv (backend "nccl" == RCCL on ROCm),
never touching the host.

Used by bench.py and by co-located server deployments; the TCP path in
`server/handler.py` remains the cross-node fallback.
"""

from __future__ import annotations

import os
from typing import List, Sequence

import torch
import torch.distributed as dist


def init_process_group_from_env() -> tuple[int, int]:
    """Initialize RCCL process group from torchrun env vars. Returns (rank, world)."""
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1 and not dist.is_initialized():
        import datetime

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29571")
        backend = "nccl" if torch.cuda.is_available() else "gloo"  # nccl == RCCL on ROCm
        # bounded timeout: a wedged collective should abort the job, not hang it
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world, timeout=datetime.timedelta(seconds=300)
        )
        if torch.cuda.is_available():
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
    return rank, world


def split_blocks(num_blocks: int, world: int) -> List[range]:
    """Contiguous spans, larger spans on later ranks (rank0 also runs embeddings/head)."""
    base = num_blocks // world
    extra = num_blocks % world
    spans = []
    start = 0
    for r in range(world):
        n = base + (1 if r >= world - extra else 0)
        spans.append(range(start, start + n))
        start += n
    assert start == num_blocks
    return spans


class PipelineStage:
    """One rank's span of blocks + RCCL hand-off to neighbors."""

    def __init__(self, blocks: Sequence[torch.nn.Module], rank: int, world: int, device, hidden_size: int,
                 dtype=torch.bfloat16):
        self.blocks = list(blocks)
        self.rank, self.world = rank, world
        self.device = device
        self.hidden_size = hidden_size
        self.dtype = dtype
        self.prev_rank = (rank - 1) % world
        self.next_rank = (rank + 1) % world

    def recv(self, batch: int, seq: int) -> torch.Tensor:
        buf = torch.empty(batch, seq, self.hidden_size, device=self.device, dtype=self.dtype)
        dist.recv(buf, src=self.prev_rank)
        return buf

    def send(self, h: torch.Tensor) -> None:
        dist.send(h.contiguous(), dst=self.next_rank)

    def forward_span(self, h: torch.Tensor, kv_caches, prefix_length: int) -> torch.Tensor:
        for block, (k, v) in zip(self.blocks, kv_caches):
            h = block(h, kv_cache=(k, v), prefix_length=prefix_length)
        return h
